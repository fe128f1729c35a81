"""Pure-python LMDB reader/writer (data/lmdb_io.py): format round-trip,
overflow values, multi-level B+tree, and the DATA layer reading a Datum
LMDB fixture (reference data_layer.cpp LMDB backend parity)."""

import os
import struct

import numpy as np
import pytest
import torch

import poseidon_amd as pa
from poseidon_amd.data.lmdb_io import (LmdbReader, LmdbWriter, MAGIC,
                                       PAGEHDRSZ, P_META)
from poseidon_amd.proto import Message, parse_text


def _write(tmp_path, records, psize=4096):
    env = str(tmp_path / "env")
    with LmdbWriter(env, psize=psize) as w:
        for k, v in records:
            w.put(k, v)
    return env


def test_lmdb_roundtrip_small(tmp_path):
    recs = [(b"%08d" % i, bytes([i % 251]) * (10 + i % 40)) for i in range(500)]
    env = _write(tmp_path, recs)
    r = LmdbReader(env)
    assert len(r) == 500
    for i, (k, v) in enumerate(recs):
        assert r.key(i) == k
        assert r.get_raw(i) == v
    assert r.get_by_key(b"%08d" % 123) == recs[123][1]
    assert r.get_by_key(b"nope") is None
    r.close()


def test_lmdb_overflow_values(tmp_path):
    # values >= nodemax (~2040 for 4K pages) go to overflow page chains
    recs = [(b"%08d" % i, bytes(range(256)) * (20 + i)) for i in range(40)]
    env = _write(tmp_path, recs)
    r = LmdbReader(env)
    assert len(r) == 40
    for i, (k, v) in enumerate(recs):
        assert r.get_raw(i) == v, f"record {i} ({len(v)} bytes)"
    r.close()


def test_lmdb_multilevel_tree(tmp_path):
    # small pages force several branch levels
    recs = [(b"%08d" % i, b"v" * 50) for i in range(3000)]
    env = _write(tmp_path, recs, psize=512)
    r = LmdbReader(env)
    assert len(r) == 3000
    assert r.depth >= 3
    for i in (0, 1, 999, 1500, 2999):
        assert r.get_raw(i) == b"v" * 50
        assert r.key(i) == b"%08d" % i
    r.close()


def test_lmdb_meta_layout(tmp_path):
    """The bytes we write must match the canonical mdb.c meta layout the
    reference's liblmdb parses: magic at +16, psize in dbs[0].pad, live
    meta = higher txnid."""
    env = _write(tmp_path, [(b"k1", b"a"), (b"k2", b"b")])
    raw = open(os.path.join(env, "data.mdb"), "rb").read()
    for page in (0, 1):
        off = page * 4096
        flags = struct.unpack_from("<H", raw, off + 10)[0]
        assert flags & P_META
        assert struct.unpack_from("<I", raw, off + PAGEHDRSZ)[0] == MAGIC
        assert struct.unpack_from("<I", raw, off + PAGEHDRSZ + 24)[0] == 4096
    t0 = struct.unpack_from("<Q", raw, PAGEHDRSZ + 24 + 96 + 8)[0]
    t1 = struct.unpack_from("<Q", raw, 4096 + PAGEHDRSZ + 24 + 96 + 8)[0]
    assert (t0, t1) == (0, 1)


def test_data_layer_reads_lmdb(tmp_path):
    """DATA layer pulls Datum records straight from an LMDB env
    (data_layer.cpp:143-261 parity, backend LMDB)."""
    pa.init(device="cpu", seed=3)
    n, C, H, W = 64, 3, 8, 8
    rng = np.random.default_rng(7)
    env = str(tmp_path / "train_lmdb")
    raws = []
    with LmdbWriter(env) as w:
        for i in range(n):
            d = Message("Datum", channels=C, height=H, width=W,
                        label=i % 5)
            d.data = rng.integers(0, 256, C * H * W).astype(np.uint8).tobytes()
            raw = d.encode()
            raws.append(raw)
            w.put(b"%08d" % i, raw)
    np_param = parse_text("NetParameter", f"""
        name: "lmdbnet"
        layers {{ name: "data" type: DATA top: "data" top: "label"
                 data_param {{ source: "{env}" backend: LMDB batch_size: 16 }}
                 transform_param {{ scale: 0.00390625 }} }}
    """)
    from poseidon_amd.core.net import Net, TRAIN
    net = Net(np_param, phase=TRAIN, verbose=False)
    net.forward()
    data = net.blobs["data"].data
    labels = net.blobs["label"].data
    assert tuple(data.shape) == (16, C, H, W)
    # first record check: scaled uint8 payload
    d0 = Message.decode("Datum", raws[0])
    want = np.frombuffer(d0.data, dtype=np.uint8).reshape(C, H, W)
    got = data[0].numpy() * 256.0
    assert np.allclose(got, want, atol=0.51)
    assert float(labels[0]) == 0.0


def test_convert_db_roundtrip(tmp_path):
    """PDB -> LMDB -> PDB via the converter keeps records byte-identical
    (reference LMDB shards become runnable, and vice versa)."""
    from poseidon_amd.data.pdb import PDBWriter, PDBReader
    from poseidon_amd.tools.datasets import convert_db
    src = str(tmp_path / "a.pdb")
    rng = np.random.default_rng(11)
    with PDBWriter(src) as w:
        for i in range(30):
            w.put_raw(rng.integers(0, 256, 100 + 97 * i).astype(np.uint8)
                      .tobytes())
    env = str(tmp_path / "b_lmdb")
    convert_db([src, env, "--to-lmdb"])
    back = str(tmp_path / "c.pdb")
    convert_db([env, back])
    a, c = PDBReader(src), PDBReader(back)
    assert len(a) == len(c) == 30
    for i in range(30):
        assert a.get_raw(i) == c.get_raw(i)
