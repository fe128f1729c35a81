"""WINDOW_DATA layer: window-file parsing, fg/bg sampling, crop+warp."""

import numpy as np

import poseidon_amd as pa
from poseidon_amd.core.net import Net, TRAIN
from poseidon_amd.proto import parse_text


def _make_window_dataset(tmp_path, n_imgs=3):
    from PIL import Image
    rng = np.random.default_rng(7)
    lines = []
    for i in range(n_imgs):
        path = str(tmp_path / f"img{i}.png")
        arr = rng.integers(0, 255, (48, 64, 3), dtype=np.uint8)
        Image.fromarray(arr).save(path)
        lines += [f"# {i}", path, "3 48 64", "4",
                  "1 0.80 5 5 30 30",     # fg (overlap .8)
                  "2 0.65 10 8 40 40",    # fg
                  "0 0.10 0 0 20 20",     # bg
                  "0 0.05 30 20 60 45"]   # bg
    wf = str(tmp_path / "windows.txt")
    with open(wf, "w") as f:
        f.write("\n".join(lines) + "\n")
    return wf


def test_window_data_layer(tmp_path):
    pa.init(device="cpu", seed=3)
    wf = _make_window_dataset(tmp_path)
    net_param = parse_text("NetParameter", f"""
        name: "wd"
        layers {{ name: "data" type: WINDOW_DATA top: "data" top: "label"
                 window_data_param {{ source: "{wf}" batch_size: 8
                     crop_size: 24 fg_threshold: 0.5 bg_threshold: 0.3
                     fg_fraction: 0.5 context_pad: 2 mirror: true }} }}
        layers {{ name: "conv" type: CONVOLUTION bottom: "data" top: "conv"
                 convolution_param {{ num_output: 4 kernel_size: 3
                     weight_filler {{ type: "xavier" }} }} }}
        layers {{ name: "pool" type: POOLING bottom: "conv" top: "pool"
                 pooling_param {{ pool: MAX kernel_size: 22 stride: 1 }} }}
        layers {{ name: "loss" type: SOFTMAX_LOSS bottom: "pool" bottom: "label"
                 top: "loss" }}
    """)
    net = Net(net_param, phase=TRAIN)
    loss = net.forward()
    assert np.isfinite(loss)
    data = net.blobs["data"]
    labels = net.blobs["label"].data.numpy()
    assert data.shape == (8, 3, 24, 24)
    # fg_fraction 0.5 -> first 4 are fg classes {1,2}, rest bg class 0
    assert set(labels[:4]).issubset({1.0, 2.0})
    assert set(labels[4:]) == {0.0}
    net.backward()
