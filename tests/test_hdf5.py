"""Pure-python HDF5 v0 subset (data/hdf5_io.py) + the HDF5_DATA /
HDF5_OUTPUT layers (reference hdf5_{data,output}_layer.cpp parity)."""

import numpy as np
import pytest
import torch

import poseidon_amd as pa
from poseidon_amd.data.hdf5_io import Hdf5Reader, Hdf5Writer
from poseidon_amd.proto import parse_text


def test_hdf5_roundtrip(tmp_path):
    p = str(tmp_path / "a.h5")
    rng = np.random.default_rng(3)
    data = rng.standard_normal((12, 3, 5, 5)).astype(np.float32)
    lab64 = rng.standard_normal(12)  # float64
    ids = np.arange(12, dtype=np.int32)
    with Hdf5Writer(p) as w:
        w.put("data", data)
        w.put("label", lab64)
        w.put("ids", ids)
    r = Hdf5Reader(p)
    assert sorted(r.keys()) == ["data", "ids", "label"]
    np.testing.assert_array_equal(r.get("data"), data)
    np.testing.assert_array_equal(r.get("label"), lab64)
    assert r.get("label").dtype == np.float64
    np.testing.assert_array_equal(r.get("ids"), ids)
    assert r.get("ids").dtype == np.int32


def test_hdf5_format_invariants(tmp_path):
    """Spot-check the on-disk bytes against the published v0 layout."""
    p = str(tmp_path / "b.h5")
    with Hdf5Writer(p) as w:
        w.put("x", np.zeros(4, dtype=np.float32))
    raw = open(p, "rb").read()
    assert raw[:8] == b"\x89HDF\r\n\x1a\n"
    assert raw[8] == 0 and raw[13] == 8 and raw[14] == 8
    assert b"HEAP" in raw and b"TREE" in raw and b"SNOD" in raw


def test_hdf5_data_layer_cycles_files(tmp_path):
    pa.init(device="cpu", seed=5)
    files = []
    for fi in range(2):
        p = str(tmp_path / f"part{fi}.h5")
        with Hdf5Writer(p) as w:
            base = fi * 10
            w.put("data", np.arange(base, base + 10, dtype=np.float32)
                  .reshape(10, 1, 1, 1).repeat(4, axis=2).repeat(4, axis=3))
            w.put("label", np.arange(base, base + 10, dtype=np.float32))
        files.append(p)
    src = tmp_path / "list.txt"
    src.write_text("\n".join(files) + "\n")
    np_param = parse_text("NetParameter", f"""
        name: "h5net"
        layers {{ name: "data" type: HDF5_DATA top: "data" top: "label"
                 hdf5_data_param {{ source: "{src}" batch_size: 4 }} }}
    """)
    from poseidon_amd.core.net import Net, TRAIN
    net = Net(np_param, phase=TRAIN, verbose=False)
    seen = []
    for _ in range(6):  # 24 rows -> crosses the file boundary
        net.forward()
        seen.extend(float(v) for v in net.blobs["label"].data)
    assert seen == [float(i % 20) for i in range(24)]
    assert tuple(net.blobs["data"].data.shape) == (4, 1, 4, 4)


def test_hdf5_output_layer(tmp_path):
    pa.init(device="cpu", seed=7)
    out = str(tmp_path / "out.h5")
    np_param = parse_text("NetParameter", f"""
        name: "h5out"
        layers {{ name: "data" type: DUMMY_DATA top: "data" top: "label"
                 dummy_data_param {{ num: 3 channels: 2 height: 2 width: 2
                     num: 3 channels: 1 height: 1 width: 1
                     data_filler {{ type: "gaussian" std: 1.0 }}
                     data_filler {{ type: "constant" value: 5 }} }} }}
        layers {{ name: "save" type: HDF5_OUTPUT bottom: "data" bottom: "label"
                 hdf5_output_param {{ file_name: "{out}" }} }}
    """)
    from poseidon_amd.core.net import Net, TRAIN
    net = Net(np_param, phase=TRAIN, verbose=False)
    net.forward()
    net.forward()
    save = next(l for l in net.layers if l.name == "save")
    save.finalize()
    r = Hdf5Reader(out)
    assert r.get("data").shape == (6, 2, 2, 2)
    assert np.all(r.get("label") == 5.0)
