"""Layer base class + registry.

Parity with /root/reference/include/caffe/layer.hpp (SetUp/Forward/Backward
contract, loss-weight plumbing) and src/caffe/layer_factory.cpp (the
LayerType switch becomes a registration decorator). PS-table binding
(SetUpBlobGlobalTable) is gone: parameter blobs are device tensors synced
by RCCL.

Device dispatch: layers call ops in poseidon_amd.ops.functional, which route
to hand-written HIP/CDNA4 kernels on GPU and to a plain fp32 torch reference
on CPU. There is no per-layer _cpu/_gpu method pair to keep in sync.
"""

from __future__ import annotations

from typing import Dict, List, Optional, Type

from .blob import Blob
from ..proto import Message, spec

LAYER_REGISTRY: Dict[str, Type["Layer"]] = {}


def register_layer(*type_names: str):
    def deco(cls):
        for t in type_names:
            assert t in spec.ENUMS["LayerType"], f"unknown LayerType {t}"
            LAYER_REGISTRY[t] = cls
        cls.layer_types = type_names
        return cls
    return deco


def create_layer(param: Message, phase: int) -> "Layer":
    type_name = param.enum_name("type")
    cls = LAYER_REGISTRY.get(type_name)
    if cls is None:
        raise NotImplementedError(f"layer type {type_name} not implemented")
    return cls(param, phase)


class Layer:
    """Base layer. Subclasses override layer_setup/reshape/forward/backward.

    forward() returns nothing; loss layers write their scalar loss into
    top[0].data and the Net applies loss_weight. backward() must ACCUMULATE
    into param blob diffs? No -- following the reference solver's per-iter
    flow, param diffs are overwritten each iteration (Caffe semantics: layer
    Backward writes param diffs, the solver consumes them before the next
    iteration; shared params accumulate in Net).
    """

    def __init__(self, param: Message, phase: int):
        self.param = param
        self.phase = phase  # 0 TRAIN / 1 TEST
        self.name = param.name or ""
        self.blobs: List[Blob] = []          # learnable parameters
        self.loss_weights: List[float] = []  # per top blob, set by Net
        self.propagate_down_params: List[bool] = []

    # -- contract ----------------------------------------------------------
    def setup(self, bottom: List[Blob], top: List[Blob]) -> None:
        self.check_blob_counts(bottom, top)
        self.layer_setup(bottom, top)
        self.reshape(bottom, top)

    def layer_setup(self, bottom: List[Blob], top: List[Blob]) -> None:
        pass

    def reshape(self, bottom: List[Blob], top: List[Blob]) -> None:
        raise NotImplementedError

    def forward(self, bottom: List[Blob], top: List[Blob]) -> None:
        raise NotImplementedError

    def backward(self, top: List[Blob], propagate_down: List[bool],
                 bottom: List[Blob]) -> None:
        raise NotImplementedError

    # -- blob-count contract (layer.hpp ExactNum/Min/Max Bottom/TopBlobs) --
    exact_num_bottom: Optional[int] = None
    min_bottom: Optional[int] = None
    max_bottom: Optional[int] = None
    exact_num_top: Optional[int] = None
    min_top: Optional[int] = None
    max_top: Optional[int] = None
    auto_top_blobs: bool = False

    def check_blob_counts(self, bottom: List[Blob], top: List[Blob]) -> None:
        n, t = len(bottom), len(top)
        if self.exact_num_bottom is not None and n != self.exact_num_bottom:
            raise ValueError(f"{self.name}: expects {self.exact_num_bottom} bottoms, got {n}")
        if self.min_bottom is not None and n < self.min_bottom:
            raise ValueError(f"{self.name}: expects >= {self.min_bottom} bottoms, got {n}")
        if self.max_bottom is not None and n > self.max_bottom:
            raise ValueError(f"{self.name}: expects <= {self.max_bottom} bottoms, got {n}")
        if self.exact_num_top is not None and t != self.exact_num_top:
            raise ValueError(f"{self.name}: expects {self.exact_num_top} tops, got {t}")
        if self.min_top is not None and t < self.min_top:
            raise ValueError(f"{self.name}: expects >= {self.min_top} tops, got {t}")
        if self.max_top is not None and t > self.max_top:
            raise ValueError(f"{self.name}: expects <= {self.max_top} tops, got {t}")

    # -- loss-layer protocol ------------------------------------------------
    is_loss: bool = False

    def auto_loss_weight(self, top_index: int) -> float:
        return 1.0 if (self.is_loss and top_index == 0) else 0.0

    # -- misc ---------------------------------------------------------------
    @property
    def type_name(self) -> str:
        return self.param.enum_name("type")

    def blobs_lr(self, i: int) -> float:
        lst = list(self.param.blobs_lr)
        if not lst:
            return 1.0
        return lst[i] if i < len(lst) else lst[-1]

    def weight_decay_mult(self, i: int) -> float:
        lst = list(self.param.weight_decay)
        if not lst:
            return 1.0
        return lst[i] if i < len(lst) else lst[-1]

    def __repr__(self) -> str:
        return f"<{type(self).__name__} {self.name!r}>"
