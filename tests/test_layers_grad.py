"""Finite-difference gradient checks for every differentiable layer (CPU
fp64 where possible; fp32 with loose tolerance elsewhere)."""

import pytest
import torch

from poseidon_amd.core.blob import Blob
from poseidon_amd.core.layer import create_layer
from poseidon_amd.proto import Message, parse_text
from poseidon_amd.utils.grad_check import grad_check

TRAIN = 0


def L(text: str) -> Message:
    return parse_text("LayerParameter", text)


def blob(shape, scale=1.0, offset=0.0, seed=0, dtype=torch.float64):
    torch.manual_seed(seed)
    b = Blob(shape, dtype=dtype)
    b.data = torch.randn(shape, dtype=dtype) * scale + offset
    return b


def label_blob(n, num_classes, seed=1):
    torch.manual_seed(seed)
    b = Blob((n,), dtype=torch.float64)
    b.data = torch.randint(0, num_classes, (n,), dtype=torch.float64)
    return b


def test_conv_grad():
    layer = create_layer(L("""name: "c" type: CONVOLUTION blobs_lr: 1 blobs_lr: 1
        convolution_param { num_output: 4 kernel_size: 3 stride: 2 pad: 1
            weight_filler { type: "gaussian" std: 0.3 } }"""), TRAIN)
    grad_check(layer, [blob((2, 3, 7, 7))], eps=1e-5, rtol=1e-4)


def test_conv_grouped_grad():
    layer = create_layer(L("""name: "c" type: CONVOLUTION
        convolution_param { num_output: 4 kernel_size: 3 group: 2
            weight_filler { type: "gaussian" std: 0.3 } }"""), TRAIN)
    grad_check(layer, [blob((2, 4, 5, 5))], eps=1e-5, rtol=1e-4)


def test_inner_product_grad():
    layer = create_layer(L("""name: "ip" type: INNER_PRODUCT
        inner_product_param { num_output: 5
            weight_filler { type: "xavier" } }"""), TRAIN)
    grad_check(layer, [blob((3, 4, 2, 2))], eps=1e-5, rtol=1e-4)


def test_pool_max_grad():
    layer = create_layer(L("""name: "p" type: POOLING
        pooling_param { pool: MAX kernel_size: 3 stride: 2 pad: 1 }"""), TRAIN)
    # distinct values avoid argmax ties under perturbation
    b = Blob((2, 2, 7, 7), dtype=torch.float64)
    g = torch.Generator().manual_seed(3)
    b.data = torch.randperm(2 * 2 * 7 * 7, generator=g).double().reshape(2, 2, 7, 7)
    grad_check(layer, [b], eps=1e-3, rtol=1e-4)


def test_pool_ave_grad():
    layer = create_layer(L("""name: "p" type: POOLING
        pooling_param { pool: AVE kernel_size: 3 stride: 2 pad: 1 }"""), TRAIN)
    grad_check(layer, [blob((2, 2, 6, 6))], eps=1e-5, rtol=1e-4)


def test_lrn_grad():
    layer = create_layer(L("""name: "n" type: LRN
        lrn_param { local_size: 5 alpha: 0.001 beta: 0.75 }"""), TRAIN)
    grad_check(layer, [blob((2, 7, 3, 3))], eps=1e-5, rtol=1e-4)


def test_lrn_within_channel_grad():
    layer = create_layer(L("""name: "n" type: LRN
        lrn_param { local_size: 3 alpha: 0.001 beta: 0.75
                    norm_region: WITHIN_CHANNEL }"""), TRAIN)
    grad_check(layer, [blob((2, 2, 5, 5))], eps=1e-5, rtol=1e-4)


@pytest.mark.parametrize("ltype", ["SIGMOID", "TANH", "BNLL", "ABSVAL"])
def test_simple_neuron_grads(ltype):
    layer = create_layer(L(f'name: "x" type: {ltype}'), TRAIN)
    grad_check(layer, [blob((2, 3, 4, 4), offset=0.5)], eps=1e-5, rtol=1e-4)


def test_relu_grad():
    layer = create_layer(L('name: "r" type: RELU relu_param { negative_slope: 0.1 }'),
                         TRAIN)
    b = blob((2, 3, 4, 4))
    b.data = b.data + torch.sign(b.data) * 0.05  # keep away from kink
    grad_check(layer, [b], eps=1e-5, rtol=1e-4)


def test_power_grad():
    layer = create_layer(L("""name: "pw" type: POWER
        power_param { power: 2.0 scale: 0.5 shift: 1.5 }"""), TRAIN)
    grad_check(layer, [blob((2, 3, 2, 2), scale=0.3)], eps=1e-5, rtol=1e-4)


def test_softmax_grad():
    layer = create_layer(L('name: "s" type: SOFTMAX'), TRAIN)
    grad_check(layer, [blob((3, 5, 2, 2))], eps=1e-5, rtol=1e-4)


def test_softmax_loss_grad():
    layer = create_layer(L('name: "sl" type: SOFTMAX_LOSS'), TRAIN)
    x = blob((4, 6))
    y = label_blob(4, 6)
    grad_check(layer, [x, y], check_bottoms=[0], eps=1e-5, rtol=1e-4)


def test_euclidean_loss_grad():
    layer = create_layer(L('name: "el" type: EUCLIDEAN_LOSS'), TRAIN)
    grad_check(layer, [blob((4, 3), seed=0), blob((4, 3), seed=5)],
               eps=1e-5, rtol=1e-4)


def test_sigmoid_ce_loss_grad():
    layer = create_layer(L('name: "sce" type: SIGMOID_CROSS_ENTROPY_LOSS'), TRAIN)
    x = blob((4, 5))
    t = Blob((4, 5), dtype=torch.float64)
    torch.manual_seed(9)
    t.data = torch.rand(4, 5, dtype=torch.float64)
    grad_check(layer, [x, t], check_bottoms=[0], eps=1e-5, rtol=1e-4)


def test_hinge_loss_grad():
    layer = create_layer(L('name: "h" type: HINGE_LOSS hinge_loss_param { norm: L2 }'),
                         TRAIN)
    x = blob((4, 5))
    y = label_blob(4, 5)
    grad_check(layer, [x, y], check_bottoms=[0], eps=1e-5, rtol=1e-4)


def test_contrastive_loss_grad():
    layer = create_layer(L('name: "cl" type: CONTRASTIVE_LOSS'), TRAIN)
    a = blob((4, 3), seed=2, scale=0.4)
    b = blob((4, 3), seed=7, scale=0.4)
    y = Blob((4,), dtype=torch.float64)
    y.data = torch.tensor([1.0, 0.0, 1.0, 0.0], dtype=torch.float64)
    grad_check(layer, [a, b, y], check_bottoms=[0, 1], eps=1e-5, rtol=1e-4)


def test_mvn_grad():
    layer = create_layer(L('name: "m" type: MVN'), TRAIN)
    grad_check(layer, [blob((2, 3, 4, 4))], eps=1e-5, rtol=1e-3)


def test_eltwise_grads():
    for op in ("SUM", "PROD", "MAX"):
        layer = create_layer(
            L(f'name: "e" type: ELTWISE eltwise_param {{ operation: {op} }}'), TRAIN)
        b0 = blob((2, 3, 2, 2), seed=1, offset=1.5)
        b1 = blob((2, 3, 2, 2), seed=2, offset=1.5)
        grad_check(layer, [b0, b1], eps=1e-5, rtol=1e-4)


def test_concat_slice_flatten_split_grads():
    layer = create_layer(L('name: "cc" type: CONCAT'), TRAIN)
    grad_check(layer, [blob((2, 3, 2, 2), seed=1), blob((2, 5, 2, 2), seed=2)],
               eps=1e-5, rtol=1e-4)
    layer = create_layer(
        L('name: "sl" type: SLICE slice_param { slice_dim: 1 slice_point: 2 }'),
        TRAIN)
    grad_check(layer, [blob((2, 5, 2, 2))], eps=1e-5, rtol=1e-4, n_tops=2)
    layer = create_layer(L('name: "f" type: FLATTEN'), TRAIN)
    grad_check(layer, [blob((2, 3, 2, 2))], eps=1e-5, rtol=1e-4)
    layer = create_layer(L('name: "sp" type: SPLIT'), TRAIN)
    grad_check(layer, [blob((2, 3, 2, 2))], eps=1e-5, rtol=1e-4, n_tops=2)


def test_dropout_grad():
    layer = create_layer(L('name: "d" type: DROPOUT dropout_param { dropout_ratio: 0.4 }'),
                         TRAIN)
    b = blob((3, 4))
    tops = [Blob()]
    layer.setup([b], tops)
    layer.forward([b], tops)
    mask = layer._mask.clone()
    w = torch.randn_like(tops[0].data)
    tops[0].diff = w.clone()
    layer.backward(tops, [True], [b])
    expected = w * mask.to(w.dtype) / (1 - 0.4)
    assert torch.allclose(b.diff, expected)
