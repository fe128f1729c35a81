"""bf16 compute path on GPU: kernels vs fp32 CPU reference with
bf16-appropriate tolerances (8-bit mantissa -> ~1e-2 relative), plus an
end-to-end bf16 training step."""

import pytest
import torch

import poseidon_amd as pa
from poseidon_amd.ops import functional as ops

pytestmark = pytest.mark.gpu
DEV = "cuda"


def rnd(*shape, seed=0, scale=1.0):
    g = torch.Generator().manual_seed(seed)
    return torch.randn(*shape, generator=g) * scale


def close_bf16(a, b, rtol=3e-2, atol=3e-2, what=""):
    a = a.detach().cpu().float()
    b = b.detach().cpu().float()
    assert a.shape == b.shape, f"{what}: {a.shape} vs {b.shape}"
    err = (a - b).abs()
    denom = b.abs().clamp(min=1.0)
    rel = (err / denom).max().item()
    assert rel <= rtol or err.max().item() <= atol, \
        f"{what}: max abs {err.max():.3e} rel {rel:.3e}"


@pytest.mark.parametrize("ak,bk", [(True, True), (True, False), (False, False)])
def test_gemm_bf16_layouts(ak, bk):
    from poseidon_amd.ops._backend import load
    ext = load()
    M, N, K = 190, 70, 333
    opA = rnd(M, K, seed=1)
    opB = rnd(K, N, seed=2)
    ref = opA @ opB
    A = (opA if ak else opA.t()).contiguous().to(DEV, torch.bfloat16)
    B = (opB.t() if bk else opB).contiguous().to(DEV, torch.bfloat16)
    out = ext.gemm(A, B, M, N, K, ak, bk)
    assert out.dtype == torch.float32
    # bf16 inputs quantize: compare vs bf16-quantized reference
    ref_q = (opA.to(torch.bfloat16).float() @ opB.to(torch.bfloat16).float())
    close_bf16(out, ref_q, rtol=1e-2, what=f"gemm bf16 ak={ak} bk={bk}")


@pytest.mark.parametrize("implicit", [False, True])
@pytest.mark.parametrize("cfg", [
    dict(C=8, Co=12, k=5, s=1, p=2, g=2, hw=13),    # materialized (Cg=4)
    dict(C=16, Co=12, k=3, s=1, p=1, g=1, hw=11),   # implicit (Cg=16)
    dict(C=64, Co=32, k=3, s=2, p=1, g=1, hw=14),   # implicit + glds (Cg=64)
    dict(C=128, Co=24, k=3, s=1, p=1, g=2, hw=9),   # implicit grouped Cg=64
])
def test_conv_bf16(cfg, implicit):
    ops.set_implicit_gemm(implicit)
    try:
        _run_conv_bf16(cfg)
    finally:
        ops.set_implicit_gemm(False)


def _run_conv_bf16(cfg):
    x = rnd(2, cfg["C"], cfg["hw"], cfg["hw"], seed=11)
    w = rnd(cfg["Co"], cfg["C"] // cfg["g"], cfg["k"], cfg["k"], seed=12, scale=0.2)
    b = rnd(cfg["Co"], seed=13)
    stride, pad, g = (cfg["s"], cfg["s"]), (cfg["p"], cfg["p"]), cfg["g"]
    # reference from bf16-QUANTIZED inputs: the GPU kernel sees bf16 x/w,
    # so comparing against unquantized fp32 mixes input-quantization noise
    # (which grows with K) into the kernel check
    xq = x.to(torch.bfloat16).float()
    wq = w.to(torch.bfloat16).float()
    y_ref, _ = ops.conv2d_forward_ex(xq, wq, b, stride, pad, g)
    xg = x.to(DEV, torch.bfloat16)
    y, cache = ops.conv2d_forward_ex(xg, w.to(DEV), b.to(DEV), stride, pad, g)
    colT, _wkT = cache
    assert y.dtype == torch.bfloat16
    close_bf16(y, y_ref, rtol=1e-2, atol=1e-2, what="conv bf16 fwd")

    dy = rnd(*y_ref.shape, seed=14)
    dyq = dy.to(torch.bfloat16).float()
    dx_ref = ops.conv2d_backward_input(wq, dyq, x.shape, stride, pad, g)
    dx = ops.conv2d_backward_input(w.to(DEV), dy.to(DEV, torch.bfloat16),
                                   x.shape, stride, pad, g)
    close_bf16(dx, dx_ref, what="conv bf16 dgrad")  # dcolT is bf16 on GPU: ~1.5% vs the fp32-intermediate reference

    # wgrad reference from bf16-QUANTIZED inputs (isolates kernel bugs from
    # input quantization; the GPU kernel sees bf16 x/dy)
    xq = x.to(torch.bfloat16).float()
    dyq = dy.to(torch.bfloat16).float()
    dw_ref = torch.zeros_like(w)
    db_ref = torch.zeros(cfg["Co"])
    ops.conv2d_backward_weight_acc(xq, None, dyq, dw_ref, db_ref, stride, pad, g)
    dw = torch.zeros_like(w).to(DEV)
    db = torch.zeros(cfg["Co"]).to(DEV)
    ops.conv2d_backward_weight_acc(xg, colT, dy.to(DEV, torch.bfloat16),
                                   dw, db, stride, pad, g)
    assert dw.dtype == torch.float32
    close_bf16(dw, dw_ref, rtol=1e-2, atol=1e-2, what="conv bf16 wgrad")
    close_bf16(db, db_ref, rtol=1e-2, atol=1e-2, what="conv bf16 bgrad")


def test_linear_and_softmax_loss_bf16():
    M, K, N = 37, 130, 75
    x, w, b = rnd(M, K, seed=3), rnd(N, K, seed=4, scale=0.2), rnd(N, seed=5)
    y_ref = ops.linear_forward(x, w, b)
    xg = x.to(DEV, torch.bfloat16)
    y = ops.linear_forward(xg, w.to(DEV), b.to(DEV))
    assert y.dtype == torch.bfloat16
    close_bf16(y, y_ref, what="linear bf16")

    logits = rnd(33, 500, seed=6, scale=2.0)  # >256 classes: label exactness
    labels = torch.randint(0, 500, (33,)).float()
    loss_ref, prob_ref = ops.softmax_loss_forward(logits, labels)
    loss, prob = ops.softmax_loss_forward(logits.to(DEV, torch.bfloat16),
                                          labels.to(DEV))
    close_bf16(loss.reshape(1), loss_ref.reshape(1), what="smloss bf16")
    dx_ref = ops.softmax_loss_backward(prob_ref, labels, 1.0)
    dx = ops.softmax_loss_backward(prob, labels.to(DEV), 1.0)
    close_bf16(dx, dx_ref, atol=5e-3, what="smloss bwd bf16")


def test_alexnet_step_bf16():
    from poseidon_amd.core.net import Net, TRAIN
    from poseidon_amd.models import zoo
    pa.init(device="cuda", seed=3, compute_dtype=torch.bfloat16)
    try:
        net = Net(zoo.alexnet(batch=8, num_classes=100), phase=TRAIN)
        loss = net.forward()
        assert torch.isfinite(torch.tensor(loss)), loss
        assert net.blobs["conv1"].data.dtype == torch.bfloat16
        net.zero_param_diffs()
        net.backward()
        for ps_ in net.learnable_params:
            assert ps_.blob.diff.dtype == torch.float32
            assert torch.isfinite(ps_.blob.diff).all(), ps_.blob.name
            assert float(ps_.blob.diff.abs().sum()) > 0, ps_.blob.name
    finally:
        pa.init(device="cpu", compute_dtype=torch.float32)


def test_bf16_training_converges():
    from poseidon_amd.solver.solver import SGDSolver
    from poseidon_amd.proto import Message
    from poseidon_amd.models import zoo
    pa.init(device="cuda", seed=5, compute_dtype=torch.bfloat16)
    try:
        sp = Message("SolverParameter", base_lr=0.01, lr_policy="fixed",
                     momentum=0.9, weight_decay=0.0005, max_iter=1000)
        sp.net_param = zoo.cifar10_quick(batch=64, num_classes=4)
        solver = SGDSolver(sp, verbose=False)
        g = torch.Generator().manual_seed(9)
        protos = torch.randn(4, 3, 32, 32, generator=g)
        labels = torch.randint(0, 4, (64,), generator=g)
        imgs = protos[labels] + 0.2 * torch.randn(64, 3, 32, 32, generator=g)
        dev = pa.ctx().torch_device
        solver.net.blobs["data"].data = imgs.to(dev, torch.bfloat16)
        solver.net.blobs["label"].data = labels.float().to(dev)
        solver.net.layers[0]._filled = True
        solver.net.layers[0].refill = [False, False]
        first = float(solver.net.forward())
        solver.step(150)
        last = float(solver.net.forward_async().item())
        assert last < first * 0.75, (first, last)
    finally:
        pa.init(device="cpu", compute_dtype=torch.float32)


@pytest.mark.parametrize("M,N,K", [
    (64, 576, 25088),    # tr16 split-K (VGG conv1_2 wgrad class)
    (128, 832, 6400),    # tr16, 64x128 tile
    (96, 640, 4096),     # tr16 thin-M 32-row tile
    (130, 577, 8192),    # tr16 interior + generic edge strips? under floor -> generic
])
def test_gemm_bf16_tn_tr16(M, N, K):
    """TN bf16->f32 shapes routed through the ds_read_b64_tr_b16 kernel
    (gemm.hip try_gemm_tn_tr) vs the fp32 torch reference."""
    from poseidon_amd.ops._backend import load
    ext = load()
    opA = rnd(M, K, seed=5) * 0.3
    opB = rnd(K, N, seed=6) * 0.3
    A = opA.t().contiguous().to(DEV, torch.bfloat16)  # [K, M] K-major
    B = opB.contiguous().to(DEV, torch.bfloat16)      # [K, N] K-major
    out = ext.gemm(A, B, M, N, K, False, False)
    assert out.dtype == torch.float32
    ref_q = opA.to(torch.bfloat16).float() @ opB.to(torch.bfloat16).float()
    close_bf16(out, ref_q, rtol=2e-2, what=f"tn tr16 {M}x{N}x{K}")
