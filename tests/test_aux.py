"""Aux subsystems: V0 proto upgrade, per-layer stats, net-output CSV."""



import poseidon_amd as pa
from poseidon_amd.core.net import Net, TRAIN
from poseidon_amd.proto import parse_text
from poseidon_amd.proto.upgrade import net_needs_upgrade, upgrade_v0_net
from poseidon_amd.utils.stats import LayerStats
from poseidon_amd.models import zoo


V0_NET = """
name: "legacy"
layers {
  layer { name: "data" type: "data" source: "/nope" batchsize: 4 }
  top: "data"
  top: "label"
}
layers {
  layer { name: "conv1" type: "conv" num_output: 4 kernelsize: 3 stride: 1
          weight_filler { type: "gaussian" std: 0.01 } }
  bottom: "data"
  top: "conv1"
}
layers {
  layer { name: "pool1" type: "pool" kernelsize: 2 stride: 2 pool: MAX }
  bottom: "conv1"
  top: "pool1"
}
layers {
  layer { name: "ip1" type: "innerproduct" num_output: 10
          weight_filler { type: "xavier" } }
  bottom: "pool1"
  top: "ip1"
}
layers {
  layer { name: "loss" type: "softmax_loss" }
  bottom: "ip1"
  bottom: "label"
  top: "loss"
}
"""


def test_v0_upgrade():
    net = parse_text("NetParameter", V0_NET)
    assert net_needs_upgrade(net)
    up = upgrade_v0_net(net)
    assert not net_needs_upgrade(up)
    types = [l.enum_name("type") for l in up.layers]
    assert types == ["DATA", "CONVOLUTION", "POOLING", "INNER_PRODUCT",
                     "SOFTMAX_LOSS"]
    conv = up.layers[1]
    assert conv.convolution_param.num_output == 4
    assert conv.convolution_param.kernel_size == 3
    assert up.layers[2].pooling_param.kernel_size == 2
    assert up.layers[0].data_param.batch_size == 4


def test_v0_padding_fusion():
    net = parse_text("NetParameter", """
        layers { layer { name: "pad1" type: "padding" pad: 2 }
                 bottom: "data" top: "padded" }
        layers { layer { name: "conv1" type: "conv" num_output: 2
                         kernelsize: 5 } bottom: "padded" top: "conv1" }
    """)
    up = upgrade_v0_net(net)
    assert len(up.layers) == 1
    assert up.layers[0].convolution_param.pad == 2


def test_layer_stats(tmp_path):
    pa.init(device="cpu", seed=1)
    net = Net(zoo.lenet(batch=4), phase=TRAIN)
    stats = LayerStats(net, use_events=False)
    with stats.timed():
        net.forward()
        net.backward()
    rows = stats.table()
    assert len(rows) == len(net.layers)
    assert sum(r[1] for r in rows) > 0
    path = str(tmp_path / "stats.yaml")
    stats.dump(path)
    import yaml
    with open(path) as f:
        doc = yaml.safe_load(f)
    assert "poseidon_stats" in doc
    assert doc["poseidon_stats"]["total_forward_ms"] > 0
    assert stats.report()
