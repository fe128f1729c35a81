"""Pure-Python LMDB (Lightning Memory-Mapped Database) file access.

The reference's DataLayer reads image datasets from LMDB environments
created by tools/convert_imageset (/root/reference/src/caffe/layers/
data_layer.cpp:143-261, database backend "lmdb"). This image ships no
liblmdb and no python-lmdb, so this module implements the on-disk format
directly: a read-only B+tree walker (LmdbReader) plus a bulk writer
(LmdbWriter) able to produce files the reference's liblmdb can open --
enough to ingest a reference-produced data.mdb unmodified and to build
test fixtures / converted shards.

On-disk layout (little-endian, 64-bit lmdb, canonical mdb.c):

  page header (16 B):  u64 pgno | u16 pad | u16 flags | u16 lower | u16 upper
                       (overflow pages: u32 page-count overlays lower/upper)
  meta page (pages 0-1, flags P_META): page header then
      u32 magic 0xBEEFC0DE | u32 version(1) | u64 address | u64 mapsize
      MDB_db[2] (48 B each: u32 pad | u16 flags | u16 depth | u64 branch_pages
                 | u64 leaf_pages | u64 overflow_pages | u64 entries | u64 root)
      u64 last_pg | u64 txnid
      -- page size lives in dbs[0].pad; the main (unnamed) DB is dbs[1];
      the live meta is the one with the larger txnid.
  leaf/branch pages: page header, then u16 node offsets [lower-16)/2 entries,
      nodes allocated downward from `upper`.
  node (8 B header): u16 lo | u16 hi | u16 flags | u16 ksize | key bytes ...
      leaf:   data size = lo | hi<<16; flags bit 0 (F_BIGDATA) means the
              key is followed by a u64 overflow pgno instead of inline data
      branch: child pgno = lo | hi<<16 | flags<<32; node 0 has ksize 0
"""

from __future__ import annotations

import mmap
import os
import struct
from typing import Iterator, List, Optional, Tuple

MAGIC = 0xBEEFC0DE
VERSION = 1

P_BRANCH = 0x01
P_LEAF = 0x02
P_OVERFLOW = 0x04
P_META = 0x08
P_DIRTY = 0x10
P_LEAF2 = 0x20

F_BIGDATA = 0x01

PAGEHDRSZ = 16
NODESZ = 8
P_INVALID = 0xFFFFFFFFFFFFFFFF


def _even(n: int) -> int:
    return (n + 1) & ~1


class LmdbReader:
    """Read-only walker over an LMDB data file. Accepts either the
    environment directory (containing data.mdb) or the data file itself.
    Records are exposed positionally in key order (Caffe writes zero-padded
    decimal keys, so key order == insertion order)."""

    def __init__(self, path: str):
        if os.path.isdir(path):
            path = os.path.join(path, "data.mdb")
        self.path = path
        self._f = open(path, "rb")
        self._mm = mmap.mmap(self._f.fileno(), 0, access=mmap.ACCESS_READ)
        meta = self._pick_meta()
        (self.psize, self.depth, self.entries, self.root) = meta
        # Positional index: list of (kind, offset, length) with kind 0 =
        # inline bytes at offset, 1 = overflow starting at page `offset`.
        self._index: List[Tuple[int, int, int]] = []
        self._keys: List[bytes] = []
        if self.root != P_INVALID:
            self._walk(self.root)
        if len(self._index) != self.entries:
            raise ValueError(
                f"{path}: walked {len(self._index)} entries, meta says "
                f"{self.entries}")

    # -- meta -----------------------------------------------------------
    def _parse_meta(self, off: int):
        mm = self._mm
        flags = struct.unpack_from("<H", mm, off + 10)[0]
        if not flags & P_META:
            raise ValueError(f"{self.path}: page at {off} is not a meta page")
        m = off + PAGEHDRSZ
        magic, version = struct.unpack_from("<II", mm, m)
        if magic != MAGIC:
            raise ValueError(f"{self.path}: bad LMDB magic {magic:#x}")
        if version != VERSION:
            raise ValueError(f"{self.path}: unsupported LMDB version {version}")
        psize = struct.unpack_from("<I", mm, m + 24)[0]  # dbs[0].md_pad
        main = m + 24 + 48  # dbs[1]
        depth = struct.unpack_from("<H", mm, main + 6)[0]
        entries = struct.unpack_from("<Q", mm, main + 32)[0]
        root = struct.unpack_from("<Q", mm, main + 40)[0]
        txnid = struct.unpack_from("<Q", mm, m + 24 + 96 + 8)[0]
        return txnid, (psize, depth, entries, root)

    def _pick_meta(self):
        t0, m0 = self._parse_meta(0)
        psize = m0[0]
        t1, m1 = self._parse_meta(psize)
        return m1 if t1 > t0 else m0

    # -- tree walk ------------------------------------------------------
    def _page(self, pgno: int) -> int:
        return pgno * self.psize

    def _walk(self, pgno: int) -> None:
        mm = self._mm
        off = self._page(pgno)
        flags, lower = struct.unpack_from("<HH", mm, off + 10)
        nkeys = (lower - PAGEHDRSZ) >> 1
        ptrs = struct.unpack_from(f"<{nkeys}H", mm, off + PAGEHDRSZ)
        if flags & P_BRANCH:
            for p in ptrs:
                n = off + p
                lo, hi, nflags = struct.unpack_from("<HHH", mm, n)
                child = lo | (hi << 16) | (nflags << 32)
                self._walk(child)
        elif flags & P_LEAF:
            if flags & P_LEAF2:
                raise ValueError("LEAF2 (fixed-size dupsort) pages unsupported")
            for p in ptrs:
                n = off + p
                lo, hi, nflags, ksize = struct.unpack_from("<HHHH", mm, n)
                dsize = lo | (hi << 16)
                key = bytes(mm[n + NODESZ:n + NODESZ + ksize])
                self._keys.append(key)
                if nflags & F_BIGDATA:
                    ovp = struct.unpack_from("<Q", mm, n + NODESZ + ksize)[0]
                    self._index.append((1, ovp, dsize))
                else:
                    self._index.append((0, n + NODESZ + ksize, dsize))
        else:
            raise ValueError(f"page {pgno}: unexpected flags {flags:#x}")

    # -- access ---------------------------------------------------------
    def __len__(self) -> int:
        return len(self._index)

    def key(self, i: int) -> bytes:
        return self._keys[i]

    def get_raw(self, i: int) -> bytes:
        kind, off, length = self._index[i]
        if kind == 0:
            return bytes(self._mm[off:off + length])
        start = self._page(off) + PAGEHDRSZ
        return bytes(self._mm[start:start + length])

    def get(self, i: int):
        from ..proto import Message
        return Message.decode("Datum", self.get_raw(i))

    def get_by_key(self, key: bytes) -> Optional[bytes]:
        import bisect
        j = bisect.bisect_left(self._keys, key)
        if j < len(self._keys) and self._keys[j] == key:
            return self.get_raw(j)
        return None

    def __iter__(self) -> Iterator[Tuple[bytes, bytes]]:
        for i in range(len(self._index)):
            yield self._keys[i], self.get_raw(i)

    def close(self) -> None:
        self._mm.close()
        self._f.close()


class LmdbWriter:
    """Bulk writer producing a fresh single-snapshot LMDB environment
    (directory with data.mdb) from records fed in ASCENDING key order --
    the shape Caffe's convert_imageset produces. Builds the leaf level,
    then branch levels bottom-up, then the two meta pages."""

    def __init__(self, path: str, psize: int = 4096):
        if not path.endswith(".mdb"):
            os.makedirs(path, exist_ok=True)
            path = os.path.join(path, "data.mdb")
        self.path = path
        self.psize = psize
        # mdb: values >= nodemax go to overflow pages
        self.nodemax = ((psize - PAGEHDRSZ) // 2) & ~1
        # pages stream straight to disk (ImageNet-scale conversions must
        # not buffer the dataset); the two meta pages are back-patched at
        # close. Page allocation is strictly sequential.
        self._f = open(path, "wb")
        self._f.write(b"\x00" * (2 * psize))  # meta placeholders
        self._next_pgno = 2
        self._leaf: List[Tuple[bytes, Tuple[int, bytes]]] = []  # pending nodes
        self._leaf_fill = 0
        self._leaf_firsts: List[Tuple[bytes, int]] = []  # (first key, pgno)
        self._entries = 0
        self._overflow_pages = 0
        self._last_key: Optional[bytes] = None

    def _new_pgno(self) -> int:
        return self._next_pgno

    def _emit_page(self, flags: int, nodes: List[Tuple[bytes, bytes]],
                   pgno: int) -> None:
        """nodes: list of (node_header_and_key_and_data bytes, '') pairs
        pre-rendered; lay out ptrs ascending, node bodies downward."""
        nk = len(nodes)
        lower = PAGEHDRSZ + 2 * nk
        body = b"".join(b for b, _ in nodes)
        upper = self.psize - len(body)
        assert upper >= lower, "page overflow"
        ptrs = []
        pos = upper
        for b, _ in nodes:
            ptrs.append(pos)
            pos += len(b)
        page = bytearray(self.psize)
        struct.pack_into("<QHHHH", page, 0, pgno, 0, flags, lower, upper)
        struct.pack_into(f"<{nk}H", page, PAGEHDRSZ, *ptrs)
        page[upper:upper + len(body)] = body
        self._write_page(pgno, bytes(page))

    def put(self, key: bytes, value: bytes) -> None:
        if isinstance(key, str):
            key = key.encode()
        if self._last_key is not None and key <= self._last_key:
            raise ValueError("LmdbWriter requires strictly ascending keys")
        self._last_key = key
        big = len(value) >= self.nodemax
        if big:
            # reserve overflow pages now so data locality mimics mdb
            npages = (PAGEHDRSZ + len(value) + self.psize - 1) // self.psize
            ovp = self._new_pgno_reserve(npages)
            page = bytearray(npages * self.psize)
            struct.pack_into("<QHHI", page, 0, ovp, 0, P_OVERFLOW, npages)
            page[PAGEHDRSZ:PAGEHDRSZ + len(value)] = value
            self._set_pages(ovp, bytes(page), npages)
            self._overflow_pages += npages
            node = struct.pack("<HHHH", len(value) & 0xFFFF,
                               (len(value) >> 16) & 0xFFFF, F_BIGDATA,
                               len(key)) + key + struct.pack("<Q", ovp)
        else:
            node = struct.pack("<HHHH", len(value) & 0xFFFF,
                               (len(value) >> 16) & 0xFFFF, 0,
                               len(key)) + key + value
        node = node + b"\x00" * (_even(len(node)) - len(node))
        need = len(node) + 2  # node + ptr slot
        if self._leaf and PAGEHDRSZ + self._leaf_fill + need > self.psize:
            self._emit_leaf()
        self._leaf.append((node, b""))
        self._leaf_fill += need
        if len(self._leaf) == 1:
            self._pending_first = key
        self._entries += 1

    def _new_pgno_reserve(self, n: int) -> int:
        pgno = self._next_pgno
        self._next_pgno += n
        return pgno

    def _write_page(self, pgno: int, data: bytes) -> None:
        assert len(data) % self.psize == 0
        self._f.seek(pgno * self.psize)
        self._f.write(data)

    def _set_pages(self, pgno: int, data: bytes, npages: int) -> None:
        self._write_page(pgno, data)

    def _emit_leaf(self) -> None:
        pgno = self._new_pgno_reserve(1)
        self._emit_page(P_LEAF, self._leaf, pgno)
        self._leaf_firsts.append((self._pending_first, pgno))
        self._leaf = []
        self._leaf_fill = 0

    def close(self) -> None:
        if self._leaf:
            self._emit_leaf()
        # build branch levels bottom-up
        level = self._leaf_firsts  # [(first_key, pgno)]
        depth = 1
        branch_pages = 0
        root = P_INVALID if self._entries == 0 else level[0][1]
        while len(level) > 1:
            nxt: List[Tuple[bytes, int]] = []
            nodes: List[Tuple[bytes, bytes]] = []
            fill = 0
            first_key = None
            for idx, (k, child) in enumerate(level):
                ksz = 0 if not nodes else len(k)  # node 0: empty key
                body = struct.pack("<HHHH", child & 0xFFFF,
                                   (child >> 16) & 0xFFFF,
                                   (child >> 32) & 0xFFFF, ksz)
                body += k[:ksz]
                body += b"\x00" * (_even(len(body)) - len(body))
                need = len(body) + 2
                if nodes and PAGEHDRSZ + fill + need > self.psize:
                    pgno = self._new_pgno_reserve(1)
                    self._emit_page(P_BRANCH, nodes, pgno)
                    branch_pages += 1
                    nxt.append((first_key, pgno))
                    nodes, fill, first_key = [], 0, None
                    # restart this child as node 0 of the new page
                    body = struct.pack("<HHHH", child & 0xFFFF,
                                       (child >> 16) & 0xFFFF,
                                       (child >> 32) & 0xFFFF, 0)
                    need = len(body) + 2
                if first_key is None:
                    first_key = k
                nodes.append((body, b""))
                fill += need
            pgno = self._new_pgno_reserve(1)
            self._emit_page(P_BRANCH, nodes, pgno)
            branch_pages += 1
            nxt.append((first_key, pgno))
            level = nxt
            depth += 1
            root = level[0][1]
        leaf_pages = len(self._leaf_firsts)
        last_pg = self._next_pgno - 1
        # meta pages: page 0 stale (txnid 0), page 1 live (txnid 1)
        def meta(pgno: int, txnid: int) -> bytes:
            page = bytearray(self.psize)
            struct.pack_into("<QHHHH", page, 0, pgno, 0, P_META,
                             PAGEHDRSZ, PAGEHDRSZ)
            m = PAGEHDRSZ
            struct.pack_into("<II", page, m, MAGIC, VERSION)
            struct.pack_into("<QQ", page, m + 8, 0, self.psize * (last_pg + 64))
            # dbs[0] (free DB): pad carries psize, empty tree
            struct.pack_into("<IHHQQQQQ", page, m + 24, self.psize, 0, 0,
                             0, 0, 0, 0, P_INVALID)
            # dbs[1] (main): the tree we just wrote (txnid 0 meta: empty)
            if txnid == 0:
                struct.pack_into("<IHHQQQQQ", page, m + 24 + 48, 0, 0, 0,
                                 0, 0, 0, 0, P_INVALID)
            else:
                struct.pack_into("<IHHQQQQQ", page, m + 24 + 48, 0, 0,
                                 depth if self._entries else 0,
                                 branch_pages, leaf_pages,
                                 self._overflow_pages, self._entries, root)
            struct.pack_into("<QQ", page, m + 24 + 96, last_pg, txnid)
            return bytes(page)

        self._f.seek(0)
        self._f.write(meta(0, 0))
        self._f.write(meta(1, 1))
        self._f.close()

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()
