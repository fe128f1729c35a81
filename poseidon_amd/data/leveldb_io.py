"""Pure-Python LevelDB read support (+ fixture writer) for reference
datasets.

The reference's dataset tools default to the LevelDB backend
(/root/reference/examples/cifar10/create_cifar10.sh BACKEND="leveldb",
src/caffe/layers/data_layer.cpp LevelDB branch). This image ships no
libleveldb/snappy, so the on-disk formats are implemented directly:

- SSTable (.ldb/.sst): footer (metaindex + index BlockHandles, magic
  0xdb4775248b80fb57), prefix-compressed blocks with restart arrays,
  per-block snappy or raw compression.
- Write-ahead log (.log): 32 KiB blocks of fragmented records; each
  record is a WriteBatch (sequence, count, then Put/Delete ops).
- Snappy: a from-scratch decompressor (literals + copy elements).

The reader does a read-only FULL-SCAN merge: every live key-value from
all table files plus the log, newest sequence number winning -- no
MANIFEST parsing needed (a dataset DB has no overwrites, but the merge
is still sequence-correct if it did). LevelDbWriter emits a single
uncompressed SSTable plus CURRENT/MANIFEST-free minimal layout for test
fixtures; it is a fixture writer, not a general LevelDB implementation.
"""

from __future__ import annotations

import os
import struct
from typing import Dict, Iterator, List, Optional, Tuple

MAGIC = 0xDB4775248B80FB57


# ---------------------------------------------------------------------------
# snappy
# ---------------------------------------------------------------------------

def snappy_uncompress(data: bytes) -> bytes:
    """Decode one snappy-compressed buffer (format spec: varint length,
    then literal / copy-1 / copy-2 / copy-4 elements)."""
    # preamble: uncompressed length varint
    n = 0
    shift = 0
    i = 0
    while True:
        b = data[i]
        n |= (b & 0x7F) << shift
        i += 1
        if not b & 0x80:
            break
        shift += 7
    out = bytearray()
    ln = len(data)
    while i < ln:
        tag = data[i]
        i += 1
        t = tag & 3
        if t == 0:  # literal
            size = (tag >> 2) + 1
            if size > 60:
                nb = size - 60
                size = int.from_bytes(data[i:i + nb], "little") + 1
                i += nb
            out += data[i:i + size]
            i += size
        else:
            if t == 1:  # copy with 1-byte offset
                size = ((tag >> 2) & 7) + 4
                off = ((tag >> 5) << 8) | data[i]
                i += 1
            elif t == 2:  # 2-byte offset
                size = (tag >> 2) + 1
                off = int.from_bytes(data[i:i + 2], "little")
                i += 2
            else:  # 4-byte offset
                size = (tag >> 2) + 1
                off = int.from_bytes(data[i:i + 4], "little")
                i += 4
            if off == 0:
                raise ValueError("snappy: zero copy offset")
            start = len(out) - off
            if start < 0:
                raise ValueError("snappy: offset before start")
            # overlapping copies are byte-serial by definition
            for _ in range(size):
                out.append(out[start])
                start += 1
    if len(out) != n:
        raise ValueError(f"snappy: expected {n} bytes, got {len(out)}")
    return bytes(out)


def _varint32(data: bytes, i: int) -> Tuple[int, int]:
    n = 0
    shift = 0
    while True:
        b = data[i]
        n |= (b & 0x7F) << shift
        i += 1
        if not b & 0x80:
            return n, i
        shift += 7


# ---------------------------------------------------------------------------
# SSTable reading
# ---------------------------------------------------------------------------

def _read_block(raw, offset: int, size: int) -> bytes:
    """BlockHandle points at block contents; a 5-byte trailer follows:
    1 byte compression type (0 raw, 1 snappy) + 4 byte crc. `raw` may be
    bytes or an mmap."""
    ctype = raw[offset + size]
    body = bytes(raw[offset:offset + size])
    if ctype == 0:
        return body
    if ctype == 1:
        return snappy_uncompress(body)
    raise ValueError(f"unsupported block compression {ctype}")


def _block_entries(block: bytes) -> Iterator[Tuple[bytes, bytes]]:
    """Iterate (key, value) of a prefix-compressed block."""
    if len(block) < 4:
        return
    n_restarts = struct.unpack_from("<I", block, len(block) - 4)[0]
    data_end = len(block) - 4 - 4 * n_restarts
    i = 0
    key = b""
    while i < data_end:
        shared, i = _varint32(block, i)
        non_shared, i = _varint32(block, i)
        vlen, i = _varint32(block, i)
        key = key[:shared] + block[i:i + non_shared]
        i += non_shared
        yield key, block[i:i + vlen]
        i += vlen


def _iter_table(path: str) -> Iterator[Tuple[bytes, int, int, bytes]]:
    """Yield (user_key, sequence, type, value) from one .ldb/.sst file.
    Internal keys carry an 8-byte (sequence << 8 | type) suffix."""
    raw = open(path, "rb").read()
    if len(raw) < 48:
        return
    footer = raw[-48:]
    if struct.unpack_from("<Q", footer, 40)[0] != MAGIC:
        raise ValueError(f"{path}: bad sstable magic")
    i = 0
    _, i = _varint32(footer, i)       # metaindex offset
    _, i = _varint32(footer, i)       # metaindex size
    idx_off, i = _varint32(footer, i)
    idx_size, i = _varint32(footer, i)
    index = _read_block(raw, idx_off, idx_size)
    for _, handle in _block_entries(index):
        off, j = _varint32(handle, 0)
        size, j = _varint32(handle, j)
        block = _read_block(raw, off, size)
        for ikey, value in _block_entries(block):
            if len(ikey) < 8:
                continue
            tag = struct.unpack_from("<Q", ikey, len(ikey) - 8)[0]
            yield ikey[:-8], tag >> 8, tag & 0xFF, value


# ---------------------------------------------------------------------------
# write-ahead log reading
# ---------------------------------------------------------------------------

def _iter_log(path: str) -> Iterator[Tuple[bytes, int, int, bytes]]:
    """Replay a LevelDB .log: reassemble fragmented records, decode each
    WriteBatch (8B sequence, 4B count, then ops: 1B type + varint-length
    key [+ varint-length value])."""
    raw = open(path, "rb").read()
    record = b""
    pos = 0
    while pos + 7 <= len(raw):
        block_rem = 32768 - (pos % 32768)
        if block_rem < 7:
            pos += block_rem  # trailer padding
            continue
        crc, length, rtype = struct.unpack_from("<IHB", raw, pos)
        pos += 7
        if length == 0 and crc == 0 and rtype == 0:
            break  # zero padding at EOF
        frag = raw[pos:pos + length]
        pos += length
        if rtype == 1:      # FULL
            batches = [frag]
        elif rtype == 2:    # FIRST
            record = frag
            continue
        elif rtype == 3:    # MIDDLE
            record += frag
            continue
        elif rtype == 4:    # LAST
            batches = [record + frag]
            record = b""
        else:
            continue
        for batch in batches:
            if len(batch) < 12:
                continue
            seq = struct.unpack_from("<Q", batch, 0)[0]
            count = struct.unpack_from("<I", batch, 8)[0]
            i = 12
            for k in range(count):
                if i >= len(batch):
                    break
                op = batch[i]
                i += 1
                klen, i = _varint32(batch, i)
                key = batch[i:i + klen]
                i += klen
                if op == 1:  # Put
                    vlen, i = _varint32(batch, i)
                    yield key, seq + k, 1, batch[i:i + vlen]
                    i += vlen
                else:        # Delete
                    yield key, seq + k, 0, b""


# ---------------------------------------------------------------------------
# reader
# ---------------------------------------------------------------------------

class LevelDbReader:
    """Read-only merge over every table + log file of an environment dir.
    Records exposed positionally in key order (Caffe writes zero-padded
    decimal keys, so key order == insertion order).

    Memory model: the index pass stores (file, block, entry) LOCATORS for
    table records -- values are fetched lazily per block with a small LRU
    of decompressed blocks, so an ImageNet-sized (tens of GB) LevelDB
    only costs its key index in RAM. Log records (normally a tail of
    recent writes) are held in memory."""

    def __init__(self, path: str, block_cache: int = 32):
        self.path = path
        if not os.path.isdir(path):
            raise ValueError(f"{path}: LevelDB environments are directories")
        # key -> (seq, type, locator); locator = bytes (log value) or
        # (file_idx, block_off, block_size, entry_idx)
        best: Dict[bytes, Tuple[int, int, object]] = {}
        self._files: List[str] = []
        self._mms: List[object] = []
        from collections import OrderedDict
        self._cache: "OrderedDict" = OrderedDict()
        self._cache_max = block_cache
        n_files = 0
        for name in sorted(os.listdir(path)):
            full = os.path.join(path, name)
            try:
                if name.endswith((".ldb", ".sst")):
                    fi = len(self._files)
                    self._files.append(full)
                    f = open(full, "rb")
                    import mmap as _mmap
                    self._mms.append(_mmap.mmap(f.fileno(), 0,
                                                access=_mmap.ACCESS_READ))
                    for key, seq, typ, loc in self._index_table(fi):
                        cur = best.get(key)
                        if cur is None or seq >= cur[0]:
                            best[key] = (seq, typ, loc)
                elif name.endswith(".log"):
                    for key, seq, typ, val in _iter_log(full):
                        cur = best.get(key)
                        if cur is None or seq >= cur[0]:
                            best[key] = (seq, typ, val)
                else:
                    continue
                n_files += 1
            except ValueError as e:
                raise ValueError(f"{full}: {e}") from e
        if n_files == 0:
            raise ValueError(
                f"{path}: no .ldb/.sst/.log files (not a LevelDB dir?)")
        self._keys = sorted(k for k, (s, t, v) in best.items() if t == 1)
        self._loc = {k: best[k][2] for k in self._keys}

    def _index_table(self, fi: int):
        raw = self._mms[fi]
        if len(raw) < 48:
            return
        footer = bytes(raw[-48:])
        if struct.unpack_from("<Q", footer, 40)[0] != MAGIC:
            raise ValueError("bad sstable magic")
        i = 0
        _, i = _varint32(footer, i)
        _, i = _varint32(footer, i)
        idx_off, i = _varint32(footer, i)
        idx_size, i = _varint32(footer, i)
        index = _read_block(raw, idx_off, idx_size)
        for _, handle in _block_entries(index):
            off, j = _varint32(handle, 0)
            size, j = _varint32(handle, j)
            block = self._block(fi, off, size)
            for ei, (ikey, _val) in enumerate(_block_entries(block)):
                if len(ikey) < 8:
                    continue
                tag = struct.unpack_from("<Q", ikey, len(ikey) - 8)[0]
                yield ikey[:-8], tag >> 8, tag & 0xFF, (fi, off, size, ei)

    def _block(self, fi: int, off: int, size: int) -> bytes:
        ck = (fi, off)
        blk = self._cache.get(ck)
        if blk is None:
            blk = _read_block(self._mms[fi], off, size)
            self._cache[ck] = blk
            if len(self._cache) > self._cache_max:
                self._cache.popitem(last=False)
        else:
            self._cache.move_to_end(ck)
        return blk

    def __len__(self) -> int:
        return len(self._keys)

    def key(self, i: int) -> bytes:
        return self._keys[i]

    def _fetch(self, loc) -> bytes:
        if isinstance(loc, bytes):
            return loc  # log-resident value
        fi, off, size, ei = loc
        block = self._block(fi, off, size)
        for j, (_k, v) in enumerate(_block_entries(block)):
            if j == ei:
                return v
        raise ValueError("entry index out of range")

    def get_raw(self, i: int) -> bytes:
        return self._fetch(self._loc[self._keys[i]])

    def get(self, i: int):
        from ..proto import Message
        return Message.decode("Datum", self.get_raw(i))

    def get_by_key(self, key: bytes) -> Optional[bytes]:
        loc = self._loc.get(key)
        return None if loc is None else self._fetch(loc)

    def __iter__(self) -> Iterator[Tuple[bytes, bytes]]:
        for k in self._keys:
            yield k, self._fetch(self._loc[k])


# ---------------------------------------------------------------------------
# fixture writer (single uncompressed SSTable)
# ---------------------------------------------------------------------------

def _varint_enc(n: int) -> bytes:
    out = bytearray()
    while True:
        b = n & 0x7F
        n >>= 7
        if n:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def _crc32c(data: bytes) -> int:
    # CRC32C (Castagnoli), then LevelDB's mask
    poly = 0x82F63B78
    crc = 0xFFFFFFFF
    for byte in data:
        crc ^= byte
        for _ in range(8):
            crc = (crc >> 1) ^ (poly if crc & 1 else 0)
    crc ^= 0xFFFFFFFF
    return ((crc >> 15) | (crc << 17)) + 0xA282EAD8 & 0xFFFFFFFF


class LevelDbWriter:
    """Writes records (ascending keys) as one uncompressed .ldb table --
    enough for LevelDbReader round-trips and for testing the block/
    restart/footer parsing against independently-written bytes."""

    def __init__(self, path: str, block_size: int = 4096):
        os.makedirs(path, exist_ok=True)
        self.path = path
        self.block_size = block_size
        self._items: List[Tuple[bytes, bytes]] = []

    def put(self, key: bytes, value: bytes) -> None:
        if isinstance(key, str):
            key = key.encode()
        if self._items and key <= self._items[-1][0]:
            raise ValueError("ascending keys required")
        self._items.append((key, value))

    @staticmethod
    def _block(entries: List[Tuple[bytes, bytes]],
               restart_every: int = 16) -> bytes:
        out = bytearray()
        restarts = []
        prev = b""
        for j, (k, v) in enumerate(entries):
            if j % restart_every == 0:
                restarts.append(len(out))
                shared = 0
            else:
                shared = 0
                while (shared < len(prev) and shared < len(k)
                       and prev[shared] == k[shared]):
                    shared += 1
            out += _varint_enc(shared) + _varint_enc(len(k) - shared) + \
                _varint_enc(len(v)) + k[shared:] + v
            prev = k
        for r in restarts:
            out += struct.pack("<I", r)
        out += struct.pack("<I", len(restarts))
        return bytes(out)

    def close(self) -> None:
        seq = 1
        blocks: List[bytes] = []
        index: List[Tuple[bytes, int, int]] = []  # (last key, off, size)
        raw = bytearray()
        cur: List[Tuple[bytes, bytes]] = []
        cur_bytes = 0

        def flush():
            nonlocal cur, cur_bytes
            if not cur:
                return
            body = self._block(cur)
            off = len(raw)
            raw.extend(body)
            raw.append(0)  # compression: raw
            raw.extend(struct.pack("<I", _crc32c(body + b"\x00")))
            index.append((cur[-1][0], off, len(body)))
            cur, cur_bytes = [], 0

        for i, (k, v) in enumerate(self._items):
            ikey = k + struct.pack("<Q", (seq + i) << 8 | 1)
            cur.append((ikey, v))
            cur_bytes += len(ikey) + len(v)
            if cur_bytes >= self.block_size:
                flush()
        flush()
        # metaindex (empty block)
        meta_body = self._block([])
        meta_off = len(raw)
        raw.extend(meta_body)
        raw.append(0)
        raw.extend(struct.pack("<I", _crc32c(meta_body + b"\x00")))
        # index block: key = last data key (+max tag), value = BlockHandle
        idx_entries = []
        for last, off, size in index:
            ikey = last + struct.pack("<Q", 0xFFFFFFFFFF01)
            idx_entries.append((ikey, _varint_enc(off) + _varint_enc(size)))
        idx_body = self._block(idx_entries)
        idx_off = len(raw)
        raw.extend(idx_body)
        raw.append(0)
        raw.extend(struct.pack("<I", _crc32c(idx_body + b"\x00")))
        footer = _varint_enc(meta_off) + _varint_enc(len(meta_body)) + \
            _varint_enc(idx_off) + _varint_enc(len(idx_body))
        footer += b"\x00" * (40 - len(footer))
        footer += struct.pack("<Q", MAGIC)
        raw.extend(footer)
        with open(os.path.join(self.path, "000005.ldb"), "wb") as f:
            f.write(bytes(raw))
        with open(os.path.join(self.path, "CURRENT"), "w") as f:
            f.write("MANIFEST-000004\n")

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()
