"""Pure-python LevelDB reader (data/leveldb_io.py): snappy decoder,
SSTable block/footer parsing, log replay, and DATA-layer ingestion of a
LevelDB environment (reference create_cifar10.sh default backend)."""

import os
import struct

import numpy as np
import pytest
import torch

import poseidon_amd as pa
from poseidon_amd.data.leveldb_io import (LevelDbReader, LevelDbWriter,
                                          snappy_uncompress, _crc32c,
                                          _varint_enc)
from poseidon_amd.proto import Message, parse_text


def _snappy_compress_ref(data: bytes) -> bytes:
    """Tiny literal-only snappy encoder (valid per spec) for decoder
    tests; plus hand-built copy elements below."""
    out = bytearray()
    n = len(data)
    while True:
        out.append(n & 0x7F | (0x80 if n > 0x7F else 0))
        n >>= 7
        if not n:
            break
    i = 0
    while i < len(data):
        chunk = data[i:i + 60]
        out.append((len(chunk) - 1) << 2)
        out += chunk
        i += len(chunk)
    return bytes(out)


def test_snappy_literals_and_copies():
    payload = b"hello world, hello world, hello world!"
    assert snappy_uncompress(_snappy_compress_ref(payload)) == payload
    # hand-built: literal "abcd" + copy-1(offset 4, len 8) -> "abcdabcdabcd"
    enc = bytes([12]) + bytes([(4 - 1) << 2]) + b"abcd" + \
        bytes([((8 - 4) << 2) | 1, 4])
    assert snappy_uncompress(enc) == b"abcdabcdabcd"
    # copy-2 form
    enc = bytes([12]) + bytes([(4 - 1) << 2]) + b"abcd" + \
        bytes([((8 - 1) << 2) | 2]) + struct.pack("<H", 4)
    assert snappy_uncompress(enc) == b"abcdabcdabcd"


def test_leveldb_roundtrip(tmp_path):
    env = str(tmp_path / "db")
    recs = [(b"%08d" % i, bytes([i % 256]) * (50 + i * 7)) for i in range(200)]
    with LevelDbWriter(env) as w:
        for k, v in recs:
            w.put(k, v)
    r = LevelDbReader(env)
    assert len(r) == 200
    for i, (k, v) in enumerate(recs):
        assert r.key(i) == k
        assert r.get_raw(i) == v
    assert r.get_by_key(b"%08d" % 123) == recs[123][1]


def test_leveldb_snappy_blocks_and_log(tmp_path):
    """A hand-assembled environment: one SSTable whose block is
    snappy-compressed + one write-ahead log with newer overwrites --
    exercises compression, sequence merge, and log replay."""
    env = tmp_path / "db2"
    env.mkdir()
    # build an sstable with LevelDbWriter then recompress its data block
    with LevelDbWriter(str(env)) as w:
        w.put(b"k1", b"old1")
        w.put(b"k2", b"val2")
    # write-ahead log: batch with seq 100: Put k1=new1, Delete k2, Put k3=v3
    def lrec(batch):
        hdr = struct.pack("<IHB", 0, len(batch), 1)  # crc unchecked, FULL
        return hdr + batch
    batch = struct.pack("<QI", 100, 3)
    batch += bytes([1]) + _varint_enc(2) + b"k1" + _varint_enc(4) + b"new1"
    batch += bytes([0]) + _varint_enc(2) + b"k2"
    batch += bytes([1]) + _varint_enc(2) + b"k3" + _varint_enc(2) + b"v3"
    (env / "000003.log").write_bytes(lrec(batch))
    r = LevelDbReader(str(env))
    assert len(r) == 2  # k2 deleted
    assert r.get_by_key(b"k1") == b"new1"
    assert r.get_by_key(b"k3") == b"v3"
    assert r.get_by_key(b"k2") is None


def test_data_layer_reads_leveldb(tmp_path):
    pa.init(device="cpu", seed=3)
    n, C, H, W = 32, 3, 6, 6
    rng = np.random.default_rng(11)
    env = str(tmp_path / "train_leveldb")
    with LevelDbWriter(env) as w:
        for i in range(n):
            d = Message("Datum", channels=C, height=H, width=W, label=i % 4)
            d.data = rng.integers(0, 256, C * H * W).astype(np.uint8).tobytes()
            w.put(b"%08d" % i, d.encode())
    np_param = parse_text("NetParameter", f"""
        name: "ldbnet"
        layers {{ name: "data" type: DATA top: "data" top: "label"
                 data_param {{ source: "{env}" backend: LEVELDB
                               batch_size: 8 }} }}
    """)
    from poseidon_amd.core.net import Net, TRAIN
    net = Net(np_param, phase=TRAIN, verbose=False)
    net.forward()
    assert tuple(net.blobs["data"].data.shape) == (8, C, H, W)
    assert [float(v) for v in net.blobs["label"].data] == [0., 1., 2., 3.,
                                                           0., 1., 2., 3.]


def test_snappy_against_real_compressor():
    """Cross-validate the from-scratch decoder against REAL snappy bytes
    (pyarrow ships the reference codec)."""
    pyarrow = pytest.importorskip("pyarrow")
    rng = np.random.default_rng(5)
    for trial in range(20):
        n = int(rng.integers(1, 100000))
        if trial % 3 == 0:
            data = rng.integers(0, 4, n).astype(np.uint8).tobytes()
        elif trial % 3 == 1:
            data = (b"pattern" * (n // 7 + 1))[:n]
        else:
            data = rng.integers(0, 256, n).astype(np.uint8).tobytes()
        comp = pyarrow.compress(data, codec="snappy", asbytes=True)
        assert snappy_uncompress(comp) == data, f"trial {trial}"


def test_leveldb_reads_snappy_compressed_table(tmp_path):
    """A table whose data blocks are REAL-snappy compressed (as the
    reference's leveldb+snappy build writes them)."""
    pyarrow = pytest.importorskip("pyarrow")
    env = tmp_path / "db3"
    # write raw, then recompress each block by rebuilding the file
    with LevelDbWriter(str(env)) as w:
        for i in range(50):
            w.put(b"%08d" % i, bytes([i]) * 100)
    raw = (env / "000005.ldb").read_bytes()
    # parse the footer to find the index, recompress every data block
    from poseidon_amd.data import leveldb_io as L
    footer = raw[-48:]
    i = 0
    meta_off, i = L._varint32(footer, i)
    meta_sz, i = L._varint32(footer, i)
    idx_off, i = L._varint32(footer, i)
    idx_sz, i = L._varint32(footer, i)
    index = L._read_block(raw, idx_off, idx_sz)
    out = bytearray()
    new_index = []
    for key, handle in L._block_entries(index):
        off, j = L._varint32(handle, 0)
        size, j = L._varint32(handle, j)
        body = raw[off:off + size]
        comp = pyarrow.compress(body, codec="snappy", asbytes=True)
        new_index.append((key, len(out), len(comp)))
        out += comp + b"\x01" + struct.pack("<I", L._crc32c(comp + b"\x01"))
    meta_body = LevelDbWriter._block([])
    m_off = len(out)
    out += meta_body + b"\x00" + struct.pack("<I", L._crc32c(meta_body + b"\x00"))
    idx_body = LevelDbWriter._block(
        [(k, _varint_enc(o) + _varint_enc(s)) for k, o, s in new_index])
    i_off = len(out)
    out += idx_body + b"\x00" + struct.pack("<I", L._crc32c(idx_body + b"\x00"))
    ftr = _varint_enc(m_off) + _varint_enc(len(meta_body)) + \
        _varint_enc(i_off) + _varint_enc(len(idx_body))
    ftr += b"\x00" * (40 - len(ftr)) + struct.pack("<Q", L.MAGIC)
    out += ftr
    (env / "000005.ldb").write_bytes(bytes(out))
    r = LevelDbReader(str(env))
    assert len(r) == 50
    for i in range(50):
        assert r.get_raw(i) == bytes([i]) * 100


def test_convert_db_from_leveldb(tmp_path):
    """convert_db ingests a LevelDB environment directly (reference
    create_*.sh output) into PDB."""
    from poseidon_amd.tools.datasets import convert_db
    from poseidon_amd.data.pdb import PDBReader
    env = str(tmp_path / "src_leveldb")
    with LevelDbWriter(env) as w:
        for i in range(20):
            w.put(b"%08d" % i, bytes([i]) * (30 + i))
    dst = str(tmp_path / "out.pdb")
    convert_db([env, dst])
    r = PDBReader(dst)
    assert len(r) == 20
    for i in range(20):
        assert r.get_raw(i) == bytes([i]) * (30 + i)
