"""Blob: the N-d (canonically NCHW) tensor pairing data with its gradient.

Design-parity with the reference Blob (/root/reference/include/caffe/blob.hpp,
src/caffe/blob.cpp) minus all parameter-server plumbing: on MI355X the
data/diff live device-resident in HBM3E (288 GB/GPU) as torch tensors and
never make PS round-trips; gradient synchronization is RCCL collectives
driven by the solver (see parallel/comm.py), replacing
Blob::UpdatePSTable/SyncWithPSTable (blob.cpp:208-286).
"""

from __future__ import annotations

from typing import Optional, Sequence, Tuple

import numpy as np
import torch

from .context import ctx
from ..proto import Message


class Blob:
    __slots__ = ("_shape", "_data", "_diff", "dtype", "device", "name")

    def __init__(self, shape: Sequence[int] = (), dtype: Optional[torch.dtype] = None,
                 device: Optional[torch.device] = None, name: str = ""):
        self._shape: Tuple[int, ...] = tuple(int(s) for s in shape)
        self._data: Optional[torch.Tensor] = None
        self._diff: Optional[torch.Tensor] = None
        self.dtype = dtype or torch.float32
        self.device = device if device is not None else ctx().torch_device
        self.name = name

    # -- shape ------------------------------------------------------------
    @property
    def shape(self) -> Tuple[int, ...]:
        return self._shape

    @property
    def count(self) -> int:
        n = 1
        for s in self._shape:
            n *= s
        return n if self._shape else 0

    # Legacy 4-D accessors (NCHW; missing TRAILING dims read as 1, matching
    # Caffe's always-4D blobs: shape (N, K) == (N, K, 1, 1)).
    def _dim4(self, i: int) -> int:
        pad = self._shape + (1,) * (4 - len(self._shape))
        return pad[i] if len(self._shape) <= 4 else self._shape[i]

    @property
    def num(self) -> int: return self._dim4(0)
    @property
    def channels(self) -> int: return self._dim4(1)
    @property
    def height(self) -> int: return self._dim4(2)
    @property
    def width(self) -> int: return self._dim4(3)

    def reshape(self, *shape) -> "Blob":
        if len(shape) == 1 and isinstance(shape[0], (tuple, list, torch.Size)):
            shape = tuple(shape[0])
        shape = tuple(int(s) for s in shape)
        if shape == self._shape:
            return self
        old_count = self.count
        self._shape = shape
        new_count = self.count
        if self._data is not None:
            if new_count == old_count:
                self._data = self._data.view(shape)
                if self._diff is not None:
                    self._diff = self._diff.view(shape)
            else:
                self._data = None
                self._diff = None
        return self

    def reshape_like(self, other: "Blob") -> "Blob":
        return self.reshape(other.shape)

    # -- storage ----------------------------------------------------------
    @property
    def data(self) -> torch.Tensor:
        if self._data is None:
            self._data = torch.zeros(self._shape, dtype=self.dtype, device=self.device)
        return self._data

    @data.setter
    def data(self, t: torch.Tensor) -> None:
        self._data = t
        self._shape = tuple(t.shape)
        self.dtype = t.dtype
        self.device = t.device

    @property
    def diff(self) -> torch.Tensor:
        # diff dtype follows data: bf16 activations carry bf16 grads
        # (bandwidth); parameter blobs are fp32 masters, so their grads
        # accumulate in fp32.
        if self._diff is None:
            self._diff = torch.zeros(self._shape, dtype=self.dtype, device=self.device)
        return self._diff

    @diff.setter
    def diff(self, t: torch.Tensor) -> None:
        self._diff = t

    def has_diff(self) -> bool:
        return self._diff is not None

    def share_data(self, other: "Blob") -> None:
        self._data = other.data
        self._shape = other.shape
        self.dtype = other.dtype
        self.device = other.device

    def share_diff(self, other: "Blob") -> None:
        self._diff = other.diff

    def zero_diff(self) -> None:
        if self._diff is not None:
            self._diff.zero_()

    def update(self) -> None:
        """data -= diff (reference Blob::Update, blob.cpp:182-205)."""
        self.data.sub_(self.diff.to(self.dtype))

    # -- proto interop ----------------------------------------------------
    def to_proto(self, write_diff: bool = False) -> Message:
        p = Message("BlobProto", num=self.num, channels=self.channels,
                    height=self.height, width=self.width)
        p.data = self.data.detach().to(torch.float32).cpu().numpy().ravel()
        if write_diff and self._diff is not None:
            p.diff = self._diff.detach().to(torch.float32).cpu().numpy().ravel()
        return p

    def from_proto(self, p: Message, reshape: bool = True) -> None:
        shape = (p.num, p.channels, p.height, p.width)
        if reshape and self._shape != shape:
            if self.count == shape[0] * shape[1] * shape[2] * shape[3]:
                self.reshape(shape)  # same count: keeps storage (and sharers)
            else:
                self._shape = shape
                self._data = None
                self._diff = None
        arr = np.asarray(p.data, dtype=np.float32)
        if arr.size != self.count:
            raise ValueError(
                f"BlobProto data count {arr.size} != blob count {self.count}")
        t = torch.from_numpy(arr.copy()).view(self._shape)
        # Copy IN PLACE when storage already exists: nets that shared this
        # blob (test nets via _share_params, name-shared params) must keep
        # seeing the same tensor, exactly like the reference's
        # Blob::FromProto memcpy into mutable_cpu_data (blob.cpp:399-426).
        # Rebinding self._data would silently detach every sharer.
        if self._data is not None and self._data.numel() == arr.size:
            self._data.copy_(t.view(self._data.shape))
        else:
            self._data = t.to(device=self.device, dtype=self.dtype)
        d = np.asarray(p.diff, dtype=np.float32)
        if d.size == self.count:
            dt = torch.from_numpy(d.copy()).view(self._shape)
            if self._diff is not None and self._diff.numel() == d.size:
                self._diff.copy_(dt.view(self._diff.shape))
            else:
                self._diff = dt.to(device=self.device, dtype=torch.float32)

    def __repr__(self) -> str:
        return f"Blob({self.name or '?'}, shape={self._shape}, dtype={self.dtype})"
