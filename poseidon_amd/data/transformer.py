"""DataTransformer: mean-subtract, crop, mirror, scale.

Parity: /root/reference/src/caffe/data_transformer.cpp:10-120. Operates on
numpy CHW float32 arrays on the host prefetch path.
"""

from __future__ import annotations

from typing import Optional

import numpy as np

from ..proto import Message, read_proto_binary


class DataTransformer:
    def __init__(self, param: Optional[Message], phase: int, rng: np.random.Generator):
        self.phase = phase
        self.rng = rng
        self.scale = 1.0
        self.mirror = False
        self.crop = 0
        self.mean: Optional[np.ndarray] = None
        self.mean_values: Optional[np.ndarray] = None
        if param is not None:
            self.scale = float(param.scale)
            self.mirror = bool(param.mirror)
            self.crop = int(param.crop_size)
            if param.has("mean_file"):
                proto = read_proto_binary(param.mean_file, "BlobProto")
                arr = np.asarray(proto.data, dtype=np.float32)
                self.mean = arr.reshape(proto.channels, proto.height, proto.width)
            elif len(param.mean_value):
                self.mean_values = np.asarray(param.mean_value, dtype=np.float32)

    def __call__(self, arr: np.ndarray) -> np.ndarray:
        c, h, w = arr.shape
        if self.mean is not None:
            arr = arr - self.mean
        elif self.mean_values is not None:
            mv = self.mean_values
            if mv.size == 1:
                arr = arr - mv[0]
            else:
                arr = arr - mv.reshape(c, 1, 1)
        if self.crop:
            cs = self.crop
            if self.phase == 0:  # TRAIN: random crop
                h_off = int(self.rng.integers(0, h - cs + 1))
                w_off = int(self.rng.integers(0, w - cs + 1))
            else:  # TEST: center crop
                h_off = (h - cs) // 2
                w_off = (w - cs) // 2
            arr = arr[:, h_off:h_off + cs, w_off:w_off + cs]
        if self.mirror and self.phase == 0 and self.rng.integers(0, 2):
            arr = arr[:, :, ::-1]
        if self.scale != 1.0:
            arr = arr * self.scale
        return np.ascontiguousarray(arr, dtype=np.float32)
