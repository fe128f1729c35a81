"""Per-layer timing + run statistics (the Bösen Stats analogue, SURVEY §5.1:
PETUUM_STATS macros -> YAML at --stats_path; here hipEvent timers around each
layer's forward/backward and the comm hooks, dumped as YAML).

Also emits rocTX ranges (torch.cuda.nvtx maps to rocTracer on ROCm) so
`rocprofv3 --marker-trace` attributes kernels to layers.

Usage:
    stats = LayerStats(net, roctx=True)
    with stats.timed():
        solver.step(20)
    stats.dump("caffe_stats.yaml")
"""

from __future__ import annotations

import contextlib
import time
from typing import Dict, List, Optional

import torch


class LayerStats:
    def __init__(self, net, roctx: bool = False, use_events: Optional[bool] = None):
        self.net = net
        self.roctx = roctx
        self.use_events = torch.cuda.is_available() if use_events is None \
            else use_events
        self.fwd_ms: Dict[str, float] = {}
        self.bwd_ms: Dict[str, float] = {}
        self.calls = 0
        self._wrapped = False

    # -- instrumentation -------------------------------------------------
    def _wrap(self) -> None:
        if self._wrapped:
            return
        self._wrapped = True
        for layer in self.net.layers:
            layer.forward = self._timed_call(layer, layer.forward, self.fwd_ms)
            layer.backward = self._timed_call(layer, layer.backward, self.bwd_ms)

    def _timed_call(self, layer, fn, book: Dict[str, float]):
        name = layer.name

        def wrapper(*a, **kw):
            if self.roctx:
                torch.cuda.nvtx.range_push(f"{layer.type_name}:{name}")
            if self.use_events:
                e0 = torch.cuda.Event(enable_timing=True)
                e1 = torch.cuda.Event(enable_timing=True)
                e0.record()
                out = fn(*a, **kw)
                e1.record()
                self._pending.append((name, book, e0, e1))
            else:
                t0 = time.perf_counter()
                out = fn(*a, **kw)
                book[name] = book.get(name, 0.0) + (time.perf_counter() - t0) * 1e3
            if self.roctx:
                torch.cuda.nvtx.range_pop()
            return out

        return wrapper

    @contextlib.contextmanager
    def timed(self):
        self._pending: List = []
        self._wrap()
        yield self
        if self.use_events:
            torch.cuda.synchronize()
            for name, book, e0, e1 in self._pending:
                book[name] = book.get(name, 0.0) + e0.elapsed_time(e1)
            self._pending = []

    # -- reporting -------------------------------------------------------
    def table(self) -> List[tuple]:
        names = [l.name for l in self.net.layers]
        rows = [(n, self.fwd_ms.get(n, 0.0), self.bwd_ms.get(n, 0.0))
                for n in names]
        return sorted(rows, key=lambda r: -(r[1] + r[2]))

    def dump(self, path: str) -> None:
        """YAML stats file (format parity with the reference's stats YAML)."""
        total_f = sum(self.fwd_ms.values())
        total_b = sum(self.bwd_ms.values())
        with open(path, "w") as f:
            f.write("poseidon_stats:\n")
            f.write(f"  total_forward_ms: {total_f:.3f}\n")
            f.write(f"  total_backward_ms: {total_b:.3f}\n")
            f.write("  layers:\n")
            for name, fm, bm in self.table():
                f.write(f"    - {{name: {name!r}, forward_ms: {fm:.3f}, "
                        f"backward_ms: {bm:.3f}}}\n")

    def report(self, top: int = 15) -> str:
        lines = [f"{'layer':<28} {'fwd ms':>9} {'bwd ms':>9}"]
        for name, fm, bm in self.table()[:top]:
            lines.append(f"{name:<28} {fm:>9.3f} {bm:>9.3f}")
        return "\n".join(lines)
