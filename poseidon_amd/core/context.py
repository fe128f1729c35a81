"""Process-global runtime context.

Replaces the reference's util::Context gflags snapshot
(/root/reference/src/caffe/context.cpp:17-74) and the Caffe singleton's
mode/phase/device state (/root/reference/src/caffe/common.cpp). One process
drives ONE GPU (rank == device) -- the MI355X-native replacement for the
reference's N-app-threads-per-process model (caffe_main.cpp:157-161).
"""

from __future__ import annotations

import os
from dataclasses import dataclass, field
from typing import Optional

import torch


@dataclass
class _Context:
    device: str = "cpu"                 # "cpu" or "cuda"
    device_index: int = 0
    compute_dtype: torch.dtype = torch.float32  # bf16 on GPU for MFMA path
    rank: int = 0
    world_size: int = 1
    seed: int = 1
    use_sfb: bool = True                # sufficient-factor broadcast for IP layers
    sfb_min_elems: int = 1 << 20        # only factor-broadcast large FC grads
    distributed: bool = False
    phase_stack: list = field(default_factory=list)

    @property
    def torch_device(self) -> torch.device:
        if self.device == "cuda":
            return torch.device("cuda", self.device_index)
        return torch.device("cpu")

    def is_root(self) -> bool:
        return self.rank == 0


_ctx = _Context()


def ctx() -> _Context:
    return _ctx


def init(device: Optional[str] = None, rank: Optional[int] = None,
         world_size: Optional[int] = None, seed: Optional[int] = None,
         compute_dtype: Optional[torch.dtype] = None) -> _Context:
    """(Re)configure the global context. Reads torchrun env vars when args
    are omitted."""
    if rank is None:
        rank = int(os.environ.get("RANK", "0"))
    if world_size is None:
        world_size = int(os.environ.get("WORLD_SIZE", "1"))
    _ctx.rank = rank
    _ctx.world_size = world_size
    _ctx.distributed = world_size > 1
    if device is not None:
        _ctx.device = device
    if _ctx.device == "cuda":
        local = int(os.environ.get("LOCAL_RANK", str(rank)))
        # oversubscription (more ranks than GPUs, e.g. debug runs on a
        # 1-GPU box) wraps onto the available devices
        local %= max(1, torch.cuda.device_count())
        _ctx.device_index = local
        torch.cuda.set_device(local)
    if seed is not None:
        _ctx.seed = seed
        torch.manual_seed(seed + 1000 * rank)
    if compute_dtype is not None:
        _ctx.compute_dtype = compute_dtype
    return _ctx
