"""Finite-difference gradient checker for layers.

The reference fork deleted upstream Caffe's test suite, so
these are fresh: objective L = sum(w_t * top_t) for fixed random w; analytic
grads from layer.backward(top.diff = w) are compared element-by-element to
central differences.
"""

from __future__ import annotations

from typing import List, Optional, Sequence

import torch

from ..core.blob import Blob


def grad_check(layer, bottoms: List[Blob], check_bottoms: Optional[Sequence[int]] = None,
               check_params: bool = True, eps: float = 1e-4,
               rtol: float = 1e-2, atol: float = 1e-4, seed: int = 0,
               n_tops: int = 1) -> None:
    torch.manual_seed(seed)
    tops = [Blob(name=f"top{i}") for i in range(n_tops)]
    layer.setup(bottoms, tops)
    layer.forward(bottoms, tops)
    ws = [torch.randn_like(t.data.to(torch.float64)).to(t.data.dtype) for t in tops]

    def objective() -> float:
        layer.forward(bottoms, tops)
        return sum(float((t.data * w).sum().item()) for t, w in zip(tops, ws))

    # analytic
    for t, w in zip(tops, ws):
        t.diff = w.clone()
    for pb in layer.blobs:
        pb.zero_diff()
    propagate = [True] * len(bottoms) if check_bottoms is None \
        else [i in check_bottoms for i in range(len(bottoms))]
    layer.backward(tops, propagate, bottoms)

    targets = []
    if check_bottoms is None:
        check_bottoms = range(len(bottoms))
    for i in check_bottoms:
        targets.append((f"bottom{i}", bottoms[i]))
    if check_params:
        for j, pb in enumerate(layer.blobs):
            targets.append((f"param{j}", pb))

    for tag, blob in targets:
        data = blob.data
        analytic = blob.diff.detach().clone().reshape(-1)
        flat = data.reshape(-1)
        numeric = torch.zeros_like(analytic, dtype=torch.float64)
        for k in range(flat.numel()):
            orig = float(flat[k].item())
            flat[k] = orig + eps
            fp = objective()
            flat[k] = orig - eps
            fm = objective()
            flat[k] = orig
            numeric[k] = (fp - fm) / (2 * eps)
        a = analytic.to(torch.float64)
        scale = torch.maximum(a.abs(), numeric.abs()).clamp(min=1.0)
        err = (a - numeric).abs() / scale
        worst = float(err.max().item())
        if worst > max(rtol, atol):
            idx = int(err.argmax().item())
            raise AssertionError(
                f"{type(layer).__name__} {tag}: grad mismatch at flat[{idx}]: "
                f"analytic={a[idx]:.6g} numeric={numeric[idx]:.6g} "
                f"(rel err {worst:.3g})")
