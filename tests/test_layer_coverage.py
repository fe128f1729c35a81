"""Coverage for the layer types not exercised elsewhere: ARGMAX, IM2COL,
IMAGE_DATA, INFOGAIN_LOSS, MULTINOMIAL_LOGISTIC_LOSS, SILENCE, THRESHOLD
(HDF5_DATA/HDF5_OUTPUT have their own suite in test_hdf5.py)."""

import numpy as np
import pytest
import torch

import poseidon_amd as pa
from poseidon_amd.core.net import Net, TRAIN
from poseidon_amd.proto import parse_text


def _net(txt):
    pa.init(device="cpu", seed=5)
    return Net(parse_text("NetParameter", txt), phase=TRAIN)


def test_argmax_threshold_silence():
    net = _net("""
        name: "aux"
        layers { name: "d" type: DUMMY_DATA top: "d"
                 dummy_data_param { num: 4 channels: 6 height: 1 width: 1
                     data_filler { type: "gaussian" std: 1.0 } } }
        layers { name: "am" type: ARGMAX bottom: "d" top: "am"
                 argmax_param { top_k: 2 out_max_val: true } }
        layers { name: "th" type: THRESHOLD bottom: "d" top: "th"
                 threshold_param { threshold: 0.25 } }
        layers { name: "sil" type: SILENCE bottom: "am" }
        layers { name: "sil2" type: SILENCE bottom: "th" }
    """)
    net.forward()
    d = net.blobs["d"].data.view(4, 6)
    am = net.blobs["am"].data
    # argmax with out_max_val: (N, 2, top_k) = indices then values
    assert am.shape[0] == 4
    top1 = d.argmax(dim=1).float()
    assert torch.equal(am.reshape(4, 2, -1)[:, 0, 0], top1)
    th = net.blobs["th"].data.view(4, 6)
    assert torch.equal(th, (d > 0.25).float())


def test_im2col_layer():
    net = _net("""
        name: "im"
        layers { name: "d" type: DUMMY_DATA top: "d"
                 dummy_data_param { num: 2 channels: 3 height: 5 width: 5
                     data_filler { type: "gaussian" std: 1.0 } } }
        layers { name: "col" type: IM2COL bottom: "d" top: "col"
                 convolution_param { kernel_size: 3 stride: 1 pad: 1 } }
    """)
    net.forward()
    col = net.blobs["col"].data
    x = net.blobs["d"].data
    ref = torch.nn.functional.unfold(x, 3, padding=1, stride=1)
    assert col.reshape(2, 27, 25).allclose(ref, atol=1e-6)


def test_multinomial_and_infogain_losses(tmp_path):
    # identity infogain matrix H as a binary BlobProto file
    from poseidon_amd.core.blob import Blob
    from poseidon_amd.proto import write_proto_binary
    hb = Blob((1, 1, 4, 4))
    hb.data = torch.eye(4)
    hpath = tmp_path / "H.binaryproto"
    write_proto_binary(hb.to_proto(), str(hpath))
    net = _net(f"""
        name: "ml"
        layers {{ name: "d" type: DUMMY_DATA top: "p" top: "l"
                 dummy_data_param {{ num: 5 channels: 4 height: 1 width: 1
                     num: 5 channels: 1 height: 1 width: 1
                     data_filler {{ type: "uniform" min: 0.05 max: 1.0 }} }} }}
        layers {{ name: "ml" type: MULTINOMIAL_LOGISTIC_LOSS
                 bottom: "p" bottom: "l" top: "ml" }}
        layers {{ name: "ig" type: INFOGAIN_LOSS
                 bottom: "p" bottom: "l" top: "ig"
                 infogain_loss_param {{ source: "{hpath}" }} }}
    """)
    # labels must be valid class ids
    net.blobs["l"].data = torch.tensor([0., 1., 2., 3., 0.])
    loss = net.forward()
    p = net.blobs["p"].data.view(5, 4)
    lbl = net.blobs["l"].data.long()
    ref = -torch.log(p[torch.arange(5), lbl].clamp(min=1e-20)).mean()
    # identity infogain H == multinomial logistic
    assert abs(float(net.blobs["ml"].data) - float(ref)) < 1e-5
    assert abs(float(net.blobs["ig"].data) - float(ref)) < 1e-5
    assert np.isfinite(loss)
    net.backward()


def test_image_data_layer(tmp_path):
    PIL = pytest.importorskip("PIL")
    from PIL import Image
    rng = np.random.default_rng(3)
    lines = []
    for i in range(4):
        p = tmp_path / f"im{i}.png"
        Image.fromarray(rng.integers(0, 255, (12, 10, 3), dtype=np.uint8)
                        ).save(str(p))
        lines.append(f"{p} {i % 2}")
    lst = tmp_path / "list.txt"
    lst.write_text("\n".join(lines) + "\n")
    net = _net(f"""
        name: "img"
        layers {{ name: "d" type: IMAGE_DATA top: "data" top: "label"
                 image_data_param {{ source: "{lst}" batch_size: 4 }} }}
    """)
    net.forward()
    assert net.blobs["data"].shape == (4, 3, 12, 10)
    assert sorted(net.blobs["label"].data.tolist()) == [0.0, 0.0, 1.0, 1.0]


def test_hdf5_missing_source_errors():
    # HDF5 layers are fully implemented (data/hdf5_io.py, no h5py needed);
    # a missing source list must fail loudly, not silently no-op
    with pytest.raises(Exception):
        _net("""
            name: "h5"
            layers { name: "d" type: HDF5_DATA top: "data"
                     hdf5_data_param { source: "/nonexistent.txt" batch_size: 2 } }
        """)
