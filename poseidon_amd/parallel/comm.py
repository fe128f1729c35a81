"""RCCL-over-xGMI gradient synchronization: the MI355X-native replacement for
the whole Bösen parameter server.

Three channels, mapped from the reference's design:

P1/P2 (full-matrix sync + DWBP)  ->  per-layer gradient all-reduce launched
    on a dedicated side HIP stream as soon as that layer's backward finishes;
    backprop of layer l-1 proceeds on the compute stream concurrently.
    Small gradients are coalesced into buckets sized for xGMI's per-link ring
    bandwidth (~153 GB/s/link) so GoogLeNet's thousands of <1 MB tensors do
    not pay per-collective latency (reference spawns a std::thread per
    (layer,param): solver.cpp:430-446; here it is events, not threads, and
    gradients never leave HBM).

P3 (SFB/SVB)  ->  solver/sfb.py: all-gather of (top_diff, bottom_data)
    factor pairs + one local MFMA GEMM reconstructs the summed ∂W.

P5/P6/P7 (metrics, init broadcast, barrier)  ->  small all-reduce /
    broadcast / barrier below.

Reduction op is SUM, matching the PS semantics (every worker pushes its
update; the server adds them -- ssp_consistency_controller.cpp Inc path).
"""

from __future__ import annotations

import os
from typing import Dict, List, Optional

import torch
import torch.distributed as dist

from ..core.context import ctx


def init_distributed(backend: Optional[str] = None) -> bool:
    c = ctx()
    if c.world_size <= 1:
        return False
    if dist.is_initialized():
        return True
    if backend is None:
        backend = os.environ.get("PS_BACKEND") or (
            "nccl" if c.device == "cuda" else "gloo")
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29500")
    kw = {}
    if backend == "nccl":
        # hipGraph capture of RCCL collectives: the NCCL watchdog's event
        # polling must not abort captured-but-idle work, and binding the PG
        # to its device up front lets RCCL init communicators eagerly
        # (before capture begins) instead of lazily inside it.
        os.environ.setdefault("TORCH_NCCL_ASYNC_ERROR_HANDLING", "0")
        kw["device_id"] = c.torch_device
    try:
        dist.init_process_group(backend=backend, rank=c.rank,
                                world_size=c.world_size, **kw)
    except TypeError:
        # older/variant torch builds without the device_id kwarg
        dist.init_process_group(backend=backend, rank=c.rank,
                                world_size=c.world_size)
    return True


def broadcast_params(params: List) -> None:
    """P6: rank 0's initial weights to everyone (replaces client0/thread0
    FillPSTable, filler.hpp:62-79)."""
    if not dist.is_initialized():
        return
    for ps in params:
        dist.broadcast(ps.blob.data, src=0)


def allreduce_metrics(t: torch.Tensor) -> torch.Tensor:
    """Small host-side metric reduction. The nccl/RCCL backend only moves
    device tensors, so stage through the GPU when that is the backend."""
    if not dist.is_initialized():
        return t
    if dist.get_backend() == "nccl" and not t.is_cuda:
        d = t.to(ctx().torch_device)
        dist.all_reduce(d, op=dist.ReduceOp.SUM)
        t.copy_(d.cpu())
        return t
    dist.all_reduce(t, op=dist.ReduceOp.SUM)
    return t


def allreduce_max(t: torch.Tensor) -> torch.Tensor:
    if not dist.is_initialized():
        return t
    if dist.get_backend() == "nccl" and not t.is_cuda:
        d = t.to(ctx().torch_device)
        dist.all_reduce(d, op=dist.ReduceOp.MAX)
        t.copy_(d.cpu())
        return t
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    return t


def barrier() -> None:
    if dist.is_initialized():
        dist.barrier()


class GradReducer:
    """Bucketed, overlapped gradient all-reduce (DWBP as stream scheduling).

    Usage per iteration:
        reducer.begin_iter()
        net.backward(post_layer_cb=reducer.on_layer_done)
        reducer.flush()              # reduce whatever is still buffered
        reducer.wait()               # compute stream waits on comm events
    After wait(), every owned param's .diff holds the SUM over ranks.
    """

    def __init__(self, net, bucket_bytes: int = 0):
        self.net = net
        # default 25 MB; tune per fabric with PS_BUCKET_MB (xGMI ring
        # bandwidth is per-link, so bigger buckets amortize latency while
        # smaller ones start overlapping earlier)
        self.bucket_bytes = bucket_bytes or int(
            os.environ.get("PS_BUCKET_BYTES", "0")) or \
            (int(os.environ.get("PS_BUCKET_MB", "25")) << 20)
        c = ctx()
        self.enabled = c.distributed and dist.is_initialized()
        self.use_stream = c.device == "cuda" and self.enabled
        self.comm_stream = torch.cuda.Stream() if self.use_stream else None
        self._events: List[torch.cuda.Event] = []
        self._pending: List[torch.Tensor] = []
        self._pending_bytes = 0
        # layer_idx -> [ParamSpec owned by that layer, excluding SFB-deferred]
        self._by_layer: Dict[int, List] = {}
        for i, p in enumerate(net.params):
            if p.owner == i and p.lr_mult != 0.0:
                self._by_layer.setdefault(p.layer_idx, []).append(p)

    def begin_iter(self) -> None:
        self._events.clear()
        self._pending.clear()
        self._pending_bytes = 0

    def on_layer_done(self, layer_idx: int, layer) -> None:
        if not self.enabled:
            return
        specs = self._by_layer.get(layer_idx)
        if not specs:
            return
        if getattr(layer, "sfb_active", False):
            # weight grad travels as factors (sfb.py); bias still reduces here
            specs = [p for p in specs if p.param_idx != 0]
            if not specs:
                return
        for p in specs:
            t = p.blob.diff
            self._pending.append(t)
            self._pending_bytes += t.numel() * t.element_size()
        if self._pending_bytes >= self.bucket_bytes:
            self._reduce_pending()

    def flush(self) -> None:
        if self.enabled and self._pending:
            self._reduce_pending()

    def _reduce_pending(self) -> None:
        tensors = self._pending
        self._pending = []
        self._pending_bytes = 0
        if self.use_stream:
            ev = torch.cuda.Event()
            ev.record(torch.cuda.current_stream())
            with torch.cuda.stream(self.comm_stream):
                self.comm_stream.wait_event(ev)
                if len(tensors) == 1:
                    dist.all_reduce(tensors[0], op=dist.ReduceOp.SUM)
                else:
                    flat = torch._utils._flatten_dense_tensors(tensors)
                    dist.all_reduce(flat, op=dist.ReduceOp.SUM)
                    for t, r in zip(tensors,
                                    torch._utils._unflatten_dense_tensors(flat, tensors)):
                        t.copy_(r)
                done = torch.cuda.Event()
                done.record(self.comm_stream)
                self._events.append(done)
        else:
            for t in tensors:
                dist.all_reduce(t, op=dist.ReduceOp.SUM)

    def wait(self) -> None:
        if self.use_stream:
            cur = torch.cuda.current_stream()
            for ev in self._events:
                cur.wait_event(ev)
            self._events.clear()
