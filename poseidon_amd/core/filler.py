"""Weight fillers (constant/uniform/gaussian/positive_unitball/xavier).

Parity with /root/reference/include/caffe/filler.hpp:25-394. The PS-table
FillPSTable variants have no analogue: rank 0 fills locally and the solver
broadcasts over RCCL (parallelism item P6 in SURVEY.md §2.4).
"""

from __future__ import annotations

import torch

from .blob import Blob
from ..proto import Message


def fill(blob: Blob, filler_param: Message) -> None:
    t = filler_param.type if filler_param is not None else "constant"
    data = blob.data
    if t == "constant":
        data.fill_(filler_param.value if filler_param is not None else 0.0)
    elif t == "uniform":
        data.uniform_(filler_param.min, filler_param.max)
    elif t == "gaussian":
        data.normal_(filler_param.mean, filler_param.std)
        sparse = filler_param.sparse
        if sparse >= 0:
            # zero weights with prob 1 - sparse/fan_in (filler.hpp:144-164)
            num_inputs = blob.count // blob.num
            non_zero_prob = min(1.0, sparse / max(1, num_inputs))
            mask = torch.bernoulli(
                torch.full_like(data.float(), non_zero_prob)).to(data.dtype)
            data.mul_(mask)
    elif t == "positive_unitball":
        data.uniform_(0, 1)
        flat = data.view(blob.num, -1)
        flat.div_(flat.sum(dim=1, keepdim=True))
    elif t == "xavier":
        fan_in = blob.count // max(1, blob.num)
        scale = (3.0 / fan_in) ** 0.5
        data.uniform_(-scale, scale)
    else:
        raise ValueError(f"unknown filler type {t!r}")
