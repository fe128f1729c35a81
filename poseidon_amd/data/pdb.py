"""PDB ("Poseidon DB"): a simple indexed record file of serialized Datum
protos -- the MI355X build's dataset container, replacing the reference's
LevelDB/LMDB backends (which need native libs this image does not ship).

Layout (little-endian):
    magic "PSDB" | u32 version | u64 nrecords
    u64 offsets[nrecords+1]           (byte offsets of record starts; last = EOF)
    record bytes (concatenated serialized Datum protos)
"""

from __future__ import annotations

import os
import shutil
import struct
from typing import Iterator, List

import numpy as np

from ..proto import Message

_MAGIC = b"PSDB"
_VERSION = 1


class PDBWriter:
    """Streams records to a sidecar file and assembles header + offset
    index at close -- O(nrecords) memory for the index only, so
    ImageNet-scale conversions do not buffer the dataset in RAM (the
    reference streamed into LevelDB/LMDB batches the same way)."""

    def __init__(self, path: str):
        self.path = path
        self._tmp_path = path + ".tmp"
        self._tmp = open(self._tmp_path, "wb")
        self._sizes: List[int] = []

    def put(self, datum: Message) -> None:
        self.put_raw(datum.encode())

    def put_raw(self, raw: bytes) -> None:
        self._tmp.write(raw)
        self._sizes.append(len(raw))

    def close(self) -> None:
        self._tmp.close()
        n = len(self._sizes)
        header = _MAGIC + struct.pack("<IQ", _VERSION, n)
        base = len(header) + 8 * (n + 1)
        offsets = [base]
        for sz in self._sizes:
            offsets.append(offsets[-1] + sz)
        with open(self.path, "wb") as f:
            f.write(header)
            f.write(struct.pack(f"<{n + 1}Q", *offsets))
            with open(self._tmp_path, "rb") as src:
                shutil.copyfileobj(src, f, 4 << 20)
        os.remove(self._tmp_path)

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.close()


class PDBReader:
    def __init__(self, path: str):
        self.path = path
        self._f = open(path, "rb")
        header = self._f.read(16)
        if header[:4] != _MAGIC:
            raise ValueError(f"{path}: not a PDB file")
        version, n = struct.unpack("<IQ", header[4:])
        self.n = n
        self.offsets = np.frombuffer(self._f.read(8 * (n + 1)), dtype="<u8")

    def __len__(self) -> int:
        return self.n

    def get_raw(self, i: int) -> bytes:
        start, end = int(self.offsets[i]), int(self.offsets[i + 1])
        self._f.seek(start)
        return self._f.read(end - start)

    def get(self, i: int) -> Message:
        return Message.decode("Datum", self.get_raw(i))

    def __iter__(self) -> Iterator[Message]:
        for i in range(self.n):
            yield self.get(i)

    def close(self) -> None:
        self._f.close()


def datum_to_array(datum: Message) -> np.ndarray:
    """Decode a Datum into a float32 CHW array (io.cpp Datum semantics)."""
    c, h, w = datum.channels, datum.height, datum.width
    if datum.has("data") and len(datum.data):
        arr = np.frombuffer(datum.data, dtype=np.uint8).astype(np.float32)
    else:
        arr = np.asarray(datum.float_data, dtype=np.float32)
    return arr.reshape(c, h, w)


def array_to_datum(arr: np.ndarray, label: int) -> Message:
    d = Message("Datum", channels=arr.shape[0], height=arr.shape[1],
                width=arr.shape[2], label=int(label))
    if arr.dtype == np.uint8:
        d.data = arr.tobytes()
    else:
        d.float_data = arr.astype(np.float32).ravel()
    return d
