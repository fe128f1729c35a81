"""Neuron (elementwise) layers: ReLU, Sigmoid, TanH, BNLL, Dropout, Power,
AbsVal, Threshold.

Parity: /root/reference/src/caffe/layers/{relu,sigmoid,tanh,bnll,dropout,
power,absval,threshold}_layer.{cpp,cu}.
"""

from __future__ import annotations

import torch
import zlib

from ..core.layer import Layer, register_layer
from ..core.context import ctx
from ..ops import functional as ops


class NeuronLayer(Layer):
    exact_num_bottom = 1
    exact_num_top = 1

    def reshape(self, bottom, top) -> None:
        top[0].reshape(bottom[0].shape)


@register_layer("RELU")
class ReLULayer(NeuronLayer):
    def layer_setup(self, bottom, top) -> None:
        rp = self.param.relu_param
        self.slope = float(rp.negative_slope) if rp is not None else 0.0
        self.fused = False  # producer GEMM applied the clamp in its epilogue

    def forward(self, bottom, top) -> None:
        if self.fused:  # in-place: bottom[0] already holds relu(y)
            top[0].data = bottom[0].data
            return
        top[0].data = ops.relu_forward(bottom[0].data, self.slope)

    def backward(self, top, propagate_down, bottom) -> None:
        if getattr(self, "bwd_fused_into_consumer", False):
            return  # consumer CONCAT's split already applied the mask
        if propagate_down[0]:
            # in-place safe: sign(bottom.data)==sign(x) after overwrite.
            # The GPU masks dy IN ITS OWN BUFFER (single-consumer DAG via
            # insert_splits): the diff tensor identity survives, so
            # downstream convs can batch their bias colsums.
            bottom[0].diff = ops.relu_backward(bottom[0].data, top[0].diff,
                                               self.slope, in_place=True)


@register_layer("SIGMOID")
class SigmoidLayer(NeuronLayer):
    def forward(self, bottom, top) -> None:
        self._y = ops.sigmoid_forward(bottom[0].data)
        top[0].data = self._y

    def backward(self, top, propagate_down, bottom) -> None:
        if propagate_down[0]:
            bottom[0].diff = ops.sigmoid_backward(self._y, top[0].diff)


@register_layer("TANH")
class TanHLayer(NeuronLayer):
    def forward(self, bottom, top) -> None:
        self._y = ops.tanh_forward(bottom[0].data)
        top[0].data = self._y

    def backward(self, top, propagate_down, bottom) -> None:
        if propagate_down[0]:
            bottom[0].diff = ops.tanh_backward(self._y, top[0].diff)


@register_layer("BNLL")
class BNLLLayer(NeuronLayer):
    def forward(self, bottom, top) -> None:
        self._x = bottom[0].data
        top[0].data = ops.bnll_forward(bottom[0].data)

    def backward(self, top, propagate_down, bottom) -> None:
        if propagate_down[0]:
            bottom[0].diff = ops.bnll_backward(self._x, top[0].diff)


@register_layer("DROPOUT")
class DropoutLayer(NeuronLayer):
    def layer_setup(self, bottom, top) -> None:
        dp = self.param.dropout_param
        self.ratio = float(dp.dropout_ratio) if dp is not None else 0.5
        assert 0.0 <= self.ratio < 1.0
        self.seed = ctx().seed * 7919 + (zlib.crc32(self.name.encode()) & 0xFFFF)
        self.offset = 0
        self._offset_dev = None
        self._mask = None

    def forward(self, bottom, top) -> None:
        if self.phase == 0:  # TRAIN
            x = bottom[0].data
            if x.is_cuda:
                if self._offset_dev is None:
                    self._offset_dev = torch.zeros(1, dtype=torch.int64,
                                                   device=x.device)
                y, mask = ops.dropout_forward(x, self.ratio, self.seed, 0,
                                              offset_dev=self._offset_dev)
            else:
                y, mask = ops.dropout_forward(x, self.ratio, self.seed,
                                              self.offset)
                self.offset += 1
            self._mask = mask
            top[0].data = y
        else:
            top[0].data = bottom[0].data

    def backward(self, top, propagate_down, bottom) -> None:
        if propagate_down[0]:
            if self.phase == 0:
                bottom[0].diff = ops.dropout_backward(top[0].diff, self._mask,
                                                      self.ratio)
            else:
                bottom[0].diff = top[0].diff


@register_layer("POWER")
class PowerLayer(NeuronLayer):
    """y = (shift + scale*x)^power (power_layer.cpp)."""

    def layer_setup(self, bottom, top) -> None:
        pp = self.param.power_param
        self.power = float(pp.power) if pp is not None else 1.0
        self.scale = float(pp.scale) if pp is not None else 1.0
        self.shift = float(pp.shift) if pp is not None else 0.0

    def forward(self, bottom, top) -> None:
        x = bottom[0].data
        inner = self.shift + self.scale * x
        self._inner = inner
        if self.power == 1.0:
            top[0].data = inner
        else:
            top[0].data = inner.pow(self.power)

    def backward(self, top, propagate_down, bottom) -> None:
        if not propagate_down[0]:
            return
        dy = top[0].diff
        if self.power == 1.0:
            bottom[0].diff = dy * self.scale
        else:
            bottom[0].diff = dy * (self.power * self.scale
                                   * self._inner.pow(self.power - 1.0))


@register_layer("ABSVAL")
class AbsValLayer(NeuronLayer):
    def forward(self, bottom, top) -> None:
        self._x = bottom[0].data
        top[0].data = bottom[0].data.abs()

    def backward(self, top, propagate_down, bottom) -> None:
        if propagate_down[0]:
            bottom[0].diff = top[0].diff * torch.sign(self._x)


@register_layer("THRESHOLD")
class ThresholdLayer(NeuronLayer):
    def layer_setup(self, bottom, top) -> None:
        tp = self.param.threshold_param
        self.threshold = float(tp.threshold) if tp is not None else 0.0

    def forward(self, bottom, top) -> None:
        top[0].data = ops.threshold_forward(bottom[0].data, self.threshold)

    def backward(self, top, propagate_down, bottom) -> None:
        raise NotImplementedError("THRESHOLD has no backward")
