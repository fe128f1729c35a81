"""Net graph mechanics: insert_splits, in-place layers, state filtering,
param sharing, checkpoint roundtrip, model zoo builds."""

import numpy as np
import torch

import poseidon_amd as pa
from poseidon_amd.core.net import Net, filter_net, TRAIN, TEST
from poseidon_amd.core.insert_splits import insert_splits
from poseidon_amd.models import zoo
from poseidon_amd.proto import Message, parse_text


def test_insert_splits():
    net = parse_text("NetParameter", """
        name: "t"
        layers { name: "d" type: DUMMY_DATA top: "x"
                 dummy_data_param { num: 2 channels: 3 height: 4 width: 4 } }
        layers { name: "a" type: RELU bottom: "x" top: "a" }
        layers { name: "b" type: SIGMOID bottom: "x" top: "b" }
    """)
    out = insert_splits(net)
    names = [l.name for l in out.layers]
    assert "x_d_0_split" in names
    relu = next(l for l in out.layers if l.name == "a")
    sig = next(l for l in out.layers if l.name == "b")
    assert relu.bottom[0] == "x_d_0_split_0"
    assert sig.bottom[0] == "x_d_0_split_1"


def test_filter_net_phase():
    net = parse_text("NetParameter", """
        layers { name: "train_data" type: DUMMY_DATA top: "x"
                 dummy_data_param { num: 2 channels: 1 height: 2 width: 2 }
                 include { phase: TRAIN } }
        layers { name: "test_data" type: DUMMY_DATA top: "x"
                 dummy_data_param { num: 2 channels: 1 height: 2 width: 2 }
                 include { phase: TEST } }
        layers { name: "act" type: RELU bottom: "x" top: "y" }
    """)
    st_train = Message("NetState", phase=TRAIN)
    st_test = Message("NetState", phase=TEST)
    tr = filter_net(net, st_train)
    te = filter_net(net, st_test)
    assert [l.name for l in tr.layers] == ["train_data", "act"]
    assert [l.name for l in te.layers] == ["test_data", "act"]


def test_split_gradient_accumulation():
    """A blob feeding two consumers must accumulate both gradients."""
    net_param = parse_text("NetParameter", """
        name: "t" force_backward: true
        layers { name: "d" type: DUMMY_DATA top: "x"
                 dummy_data_param { num: 2 channels: 3 height: 1 width: 1
                                    data_filler { type: "gaussian" std: 1.0 } } }
        layers { name: "p1" type: POWER bottom: "x" top: "a"
                 power_param { power: 2.0 } loss_weight: 1.0 }
        layers { name: "p2" type: POWER bottom: "x" top: "b"
                 power_param { power: 3.0 } loss_weight: 1.0 }
    """)
    net = Net(net_param, phase=TRAIN)
    net.forward()
    net.backward()
    x = net.blobs["x_d_0_split"] if "x_d_0_split" in net.blobs else net.blobs["x"]
    xd = net.blobs["x"]
    expected = 2 * xd.data + 3 * xd.data ** 2
    assert torch.allclose(xd.diff, expected, rtol=1e-5, atol=1e-6)


def test_param_sharing():
    net_param = parse_text("NetParameter", """
        name: "t"
        layers { name: "d" type: DUMMY_DATA top: "x" top: "label"
                 dummy_data_param { num: 4 num: 4 channels: 8 channels: 1
                                    height: 1 height: 1 width: 1 width: 1
                                    data_filler { type: "gaussian" std: 1.0 }
                                    data_filler { type: "constant" value: 1.0 } } }
        layers { name: "ip1" type: INNER_PRODUCT bottom: "x" top: "h1"
                 param: "shared_w" param: "shared_b"
                 inner_product_param { num_output: 8
                     weight_filler { type: "xavier" } } }
        layers { name: "ip2" type: INNER_PRODUCT bottom: "h1" top: "h2"
                 param: "shared_w" param: "shared_b"
                 inner_product_param { num_output: 8
                     weight_filler { type: "xavier" } } }
        layers { name: "loss" type: EUCLIDEAN_LOSS bottom: "h2" bottom: "x"
                 top: "l" }
    """)
    net = Net(net_param, phase=TRAIN)
    by_name = {l.name: l for l in net.layers}
    l1, l2 = by_name["ip1"], by_name["ip2"]
    assert l1.blobs[0].data.data_ptr() == l2.blobs[0].data.data_ptr()
    assert len(net.learnable_params) == 2  # shared_w, shared_b (owners only)
    net.zero_param_diffs()
    net.forward()
    net.backward()
    # both layers contributed to the shared diff
    assert float(l1.blobs[0].diff.abs().sum()) > 0


def test_zoo_builds_and_shapes():
    for name, batch in [("lenet", 4), ("cifar10_quick", 4), ("alexnet", 2)]:
        net = Net(zoo.build_net(name, batch=batch), phase=TRAIN)
        loss = net.forward()
        assert np.isfinite(loss), name
        net.backward()


def test_googlenet_builds():
    net = Net(zoo.googlenet(batch=2, num_classes=50), phase=TRAIN)
    loss = net.forward()
    assert np.isfinite(loss)
    net.backward()
    # three losses: two aux (0.3) + main
    assert len(net._loss_tops) == 3


def test_checkpoint_roundtrip(tmp_path):
    from poseidon_amd.proto import write_proto_binary, read_proto_binary
    net = Net(zoo.lenet(batch=2), phase=TRAIN)
    net.forward()
    proto = net.to_proto()
    path = str(tmp_path / "m.caffemodel")
    write_proto_binary(proto, path)
    net2 = Net(zoo.lenet(batch=2), phase=TRAIN)
    before = net2.layers[1].blobs[0].data.clone()
    net2.copy_trained_layers_from(read_proto_binary(path, "NetParameter"))
    after = net2.layers[1].blobs[0].data
    assert torch.allclose(after, net.layers[1].blobs[0].data)
    assert not torch.allclose(before, after)


def test_relu_fusion_matches_unfused():
    """conv/IP + in-place ReLU folds into the GEMM epilogue (net.py
    _fuse_relu_epilogues); forward/backward must match the unfused net."""
    import numpy as np
    import torch
    import poseidon_amd as pa
    from poseidon_amd.core.net import Net, TRAIN
    from poseidon_amd.proto import parse_text

    txt = """
        name: "fuse"
        layers { name: "data" type: DUMMY_DATA top: "data" top: "label"
                 dummy_data_param { num: 4 channels: 3 height: 8 width: 8
                     num: 4 channels: 1 height: 1 width: 1
                     data_filler { type: "gaussian" std: 1.0 } } }
        layers { name: "conv" type: CONVOLUTION bottom: "data" top: "conv"
                 convolution_param { num_output: 6 kernel_size: 3
                     weight_filler { type: "xavier" }
                     bias_filler { type: "constant" value: 0.1 } } }
        layers { name: "relu1" type: RELU bottom: "conv" top: "conv" }
        layers { name: "ip" type: INNER_PRODUCT bottom: "conv" top: "ip"
                 inner_product_param { num_output: 10
                     weight_filler { type: "xavier" } } }
        layers { name: "relu2" type: RELU bottom: "ip" top: "ip" }
        layers { name: "fc" type: INNER_PRODUCT bottom: "ip" top: "fc"
                 inner_product_param { num_output: 5
                     weight_filler { type: "xavier" } } }
        layers { name: "loss" type: SOFTMAX_LOSS bottom: "fc" bottom: "label"
                 top: "loss" }
    """

    def run(disable_fusion):
        pa.init(device="cpu", seed=11)
        net = Net(parse_text("NetParameter", txt), phase=TRAIN)
        fused = [l for l in net.layers if getattr(l, "fuse_relu", False)]
        if disable_fusion:
            for l in net.layers:
                if hasattr(l, "fuse_relu"):
                    l.fuse_relu = False
                if hasattr(l, "fused"):
                    l.fused = False
        else:
            assert len(fused) == 2  # conv+relu1, ip+relu2 (fc feeds loss)
        loss = net.forward()
        net.zero_param_diffs()
        net.backward()
        grads = [p.blob.diff.clone() for p in net.params]
        return loss, grads

    l1, g1 = run(disable_fusion=False)
    l0, g0 = run(disable_fusion=True)
    assert np.isclose(l1, l0)
    for a, b in zip(g1, g0):
        assert torch.allclose(a, b), "fused/unfused gradients differ"


def test_state_rules_levels_and_stages():
    """NetStateRule level/stage/not_stage matching (net.cpp:423-468)."""
    from poseidon_amd.core.net import state_meets_rule
    from poseidon_amd.proto import Message

    st = Message("NetState", phase=0, level=2)
    st.stage.append("deploy")
    st.stage.append("quantized")

    r = Message("NetStateRule", min_level=1, max_level=3)
    assert state_meets_rule(st, r)
    r = Message("NetStateRule", min_level=3)
    assert not state_meets_rule(st, r)
    r = Message("NetStateRule", max_level=1)
    assert not state_meets_rule(st, r)

    r = Message("NetStateRule")
    r.stage.append("deploy")
    assert state_meets_rule(st, r)
    r.stage.append("missing")
    assert not state_meets_rule(st, r)

    r = Message("NetStateRule")
    r.not_stage.append("quantized")
    assert not state_meets_rule(st, r)
    r = Message("NetStateRule")
    r.not_stage.append("other")
    assert state_meets_rule(st, r)

    r = Message("NetStateRule", phase=1)
    assert not state_meets_rule(st, r)


def test_filter_net_rejects_include_and_exclude():
    import pytest as _pt
    from poseidon_amd.core.net import filter_net
    from poseidon_amd.proto import parse_text, Message
    np_ = parse_text("NetParameter", """
        name: "x"
        layers { name: "d" type: DUMMY_DATA top: "d"
                 dummy_data_param { num: 1 channels: 1 height: 1 width: 1 }
                 include { phase: TRAIN } exclude { phase: TEST } }
    """)
    with _pt.raises(ValueError, match="include OR exclude"):
        filter_net(np_, Message("NetState", phase=0))
