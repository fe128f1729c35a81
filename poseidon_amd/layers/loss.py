"""Loss + metric layers: SoftmaxWithLoss, Softmax, EuclideanLoss, HingeLoss,
InfogainLoss, MultinomialLogisticLoss, SigmoidCrossEntropyLoss,
ContrastiveLoss, Accuracy.

Parity: /root/reference/src/caffe/layers/*_loss_layer.{cpp,cu},
softmax_layer.cu, accuracy_layer.cpp. Losses are normalized by batch size
(num), matching Caffe. Under data parallelism the gradients are SUMMED
across ranks (parallel/comm.GradReducer all-reduces with SUM -- the same
semantics as N workers each BatchInc-ing the PS table); only display/test
metrics are averaged. Effective lr therefore scales with world size exactly
as it does for the reference's multi-worker runs.
"""

from __future__ import annotations

import numpy as np
import torch

from ..core.layer import Layer, register_layer
from ..ops import functional as ops
from ..proto import read_proto_binary


class LossLayer(Layer):
    is_loss = True
    exact_num_bottom = 2
    max_top = 1

    def reshape(self, bottom, top) -> None:
        if top:
            top[0].reshape(())  # scalar loss

    def _top_loss_weight(self, top) -> float:
        """Scalar the backward scales gradients by (the reference reads
        top cpu_diff()[0], which Net seeded with the loss weight). Reading
        the device diff would be a D2H sync every iteration -- it
        serializes the backward launch queue and aborts hipGraph capture --
        so use the host-known seeded value (Net.backward seeds loss-top
        diffs from _loss_tops each iteration) and only fall back to the
        device read for exotic graphs where the loss top feeds consumers
        that accumulate extra gradient into it."""
        if not top:
            return 1.0
        w = getattr(self, "seeded_loss_weight", None)
        if w is not None:
            return float(w)
        return float(top[0].diff.reshape(-1)[0].item())


@register_layer("SOFTMAX")
class SoftmaxLayer(Layer):
    is_loss = False
    exact_num_bottom = 1
    exact_num_top = 1

    def reshape(self, bottom, top) -> None:
        top[0].reshape(bottom[0].shape)

    def forward(self, bottom, top) -> None:
        self._y = ops.softmax_forward(bottom[0].data)
        top[0].data = self._y

    def backward(self, top, propagate_down, bottom) -> None:
        if propagate_down[0]:
            bottom[0].diff = ops.softmax_backward(self._y, top[0].diff)


@register_layer("SOFTMAX_LOSS")
class SoftmaxWithLossLayer(LossLayer):
    def forward(self, bottom, top) -> None:
        logits = bottom[0].data.reshape(bottom[0].num, -1)
        labels = bottom[1].data.reshape(-1)
        loss, prob = ops.softmax_loss_forward(logits, labels)
        self._prob = prob
        self._labels = labels
        if top:
            top[0].data = loss.reshape(())

    def backward(self, top, propagate_down, bottom) -> None:
        if len(propagate_down) > 1 and propagate_down[1]:
            raise ValueError("cannot backprop to labels")
        if propagate_down[0]:
            w = self._top_loss_weight(top)
            dx = ops.softmax_loss_backward(self._prob, self._labels, w)
            bottom[0].diff = dx.reshape(bottom[0].shape)


@register_layer("MULTINOMIAL_LOGISTIC_LOSS")
class MultinomialLogisticLossLayer(LossLayer):
    def forward(self, bottom, top) -> None:
        prob = bottom[0].data.reshape(bottom[0].num, -1)
        labels = bottom[1].data.reshape(-1).long()
        n = prob.shape[0]
        picked = prob[torch.arange(n), labels].clamp(min=1e-20)
        self._cache = (prob, labels, picked)
        if top:
            top[0].data = (-picked.log().sum() / n).reshape(())

    def backward(self, top, propagate_down, bottom) -> None:
        if propagate_down[0]:
            prob, labels, picked = self._cache
            n = prob.shape[0]
            w = self._top_loss_weight(top)
            dx = torch.zeros_like(prob)
            dx[torch.arange(n), labels] = -w / (picked * n)
            bottom[0].diff = dx.reshape(bottom[0].shape)


@register_layer("EUCLIDEAN_LOSS")
class EuclideanLossLayer(LossLayer):
    def reshape(self, bottom, top) -> None:
        assert bottom[0].count == bottom[1].count
        super().reshape(bottom, top)

    def forward(self, bottom, top) -> None:
        n = bottom[0].num
        diff = bottom[0].data.reshape(n, -1) - bottom[1].data.reshape(n, -1)
        self._diff = diff
        if top:
            top[0].data = ((diff * diff).sum() / (2.0 * n)).reshape(())

    def backward(self, top, propagate_down, bottom) -> None:
        n = bottom[0].num
        w = self._top_loss_weight(top)
        for i, sign in ((0, 1.0), (1, -1.0)):
            if propagate_down[i]:
                bottom[i].diff = (self._diff * (sign * w / n)).reshape(bottom[i].shape)


@register_layer("HINGE_LOSS")
class HingeLossLayer(LossLayer):
    def layer_setup(self, bottom, top) -> None:
        hp = self.param.hinge_loss_param
        self.norm = hp.enum_name("norm") if hp is not None else "L1"

    def forward(self, bottom, top) -> None:
        x = bottom[0].data.reshape(bottom[0].num, -1)
        labels = bottom[1].data.reshape(-1).long()
        n, k = x.shape
        margin = x.clone()
        rows = torch.arange(n)
        margin[rows, labels] *= -1
        margin = (1.0 + margin).clamp(min=0)
        margin[rows, labels] = margin[rows, labels]  # hinge incl. own class
        self._cache = (margin, labels)
        if top:
            if self.norm == "L1":
                top[0].data = (margin.sum() / n).reshape(())
            else:
                top[0].data = ((margin * margin).sum() / n).reshape(())

    def backward(self, top, propagate_down, bottom) -> None:
        if not propagate_down[0]:
            return
        margin, labels = self._cache
        n = margin.shape[0]
        w = self._top_loss_weight(top)
        rows = torch.arange(n)
        if self.norm == "L1":
            g = (margin > 0).to(margin.dtype)
        else:
            g = 2.0 * margin
        g[rows, labels] *= -1
        bottom[0].diff = (g * (w / n)).reshape(bottom[0].shape)


@register_layer("SIGMOID_CROSS_ENTROPY_LOSS")
class SigmoidCrossEntropyLossLayer(LossLayer):
    def forward(self, bottom, top) -> None:
        x = bottom[0].data
        t = bottom[1].data
        n = bottom[0].num
        # stable: sum x*(t - (x>=0)) - log(1+exp(x - 2x(x>=0)))
        pos = (x >= 0).to(x.dtype)
        loss = -(x * (t - pos) - torch.log1p(torch.exp(x - 2 * x * pos))).sum() / n
        self._cache = (x, t)
        if top:
            top[0].data = loss.reshape(())

    def backward(self, top, propagate_down, bottom) -> None:
        if len(propagate_down) > 1 and propagate_down[1]:
            raise ValueError("cannot backprop to targets")
        if propagate_down[0]:
            x, t = self._cache
            n = bottom[0].num
            w = self._top_loss_weight(top)
            bottom[0].diff = (torch.sigmoid(x) - t) * (w / n)


@register_layer("INFOGAIN_LOSS")
class InfogainLossLayer(LossLayer):
    min_bottom = 2
    max_bottom = 3

    def layer_setup(self, bottom, top) -> None:
        ip = self.param.infogain_loss_param
        self.H = None
        if len(bottom) == 2:
            if ip is None or not ip.has("source"):
                raise ValueError("INFOGAIN_LOSS needs an H matrix")
            proto = read_proto_binary(ip.source, "BlobProto")
            arr = np.asarray(proto.data, dtype=np.float32)
            dim = int(np.sqrt(arr.size))
            self.H = torch.from_numpy(arr.reshape(dim, dim).copy())

    def forward(self, bottom, top) -> None:
        prob = bottom[0].data.reshape(bottom[0].num, -1)
        labels = bottom[1].data.reshape(-1).long()
        H = self.H.to(prob.device, prob.dtype) if self.H is not None \
            else bottom[2].data.view(prob.shape[1], prob.shape[1])
        n = prob.shape[0]
        logp = prob.clamp(min=1e-20).log()
        self._cache = (prob, labels, H)
        if top:
            top[0].data = (-(H[labels] * logp).sum() / n).reshape(())

    def backward(self, top, propagate_down, bottom) -> None:
        if propagate_down[0]:
            prob, labels, H = self._cache
            n = prob.shape[0]
            w = self._top_loss_weight(top)
            bottom[0].diff = (-(H[labels] / prob.clamp(min=1e-20)) * (w / n)) \
                .view(bottom[0].shape)


@register_layer("CONTRASTIVE_LOSS")
class ContrastiveLossLayer(LossLayer):
    exact_num_bottom = 3

    def layer_setup(self, bottom, top) -> None:
        cp = self.param.contrastive_loss_param
        self.margin = float(cp.margin) if cp is not None else 1.0

    def forward(self, bottom, top) -> None:
        a = bottom[0].data.reshape(bottom[0].num, -1)
        b = bottom[1].data.view(bottom[1].num, -1)
        y = bottom[2].data.view(-1)
        diff = a - b
        d2 = (diff * diff).sum(dim=1)
        n = a.shape[0]
        self._cache = (diff, d2, y)
        if top:
            # per-pair terms via the CLLForward kernel on GPU
            # (contrastive_loss_layer.cu:49, legacy margin-d^2 form)
            terms = ops.contrastive_terms(d2, y, self.margin, legacy=True)
            loss = (terms.to(d2.dtype)).sum() / (2 * n)
            top[0].data = loss.reshape(())

    def backward(self, top, propagate_down, bottom) -> None:
        diff, d2, y = self._cache
        n = diff.shape[0]
        w = self._top_loss_weight(top)
        active = ((self.margin - d2) > 0).to(diff.dtype)
        scale = (y - (1 - y) * active).view(-1, 1) * (w / n)
        for i, sign in ((0, 1.0), (1, -1.0)):
            if propagate_down[i]:
                bottom[i].diff = (sign * scale * diff).view(bottom[i].shape)


@register_layer("ACCURACY")
class AccuracyLayer(Layer):
    exact_num_bottom = 2
    exact_num_top = 1

    def layer_setup(self, bottom, top) -> None:
        ap = self.param.accuracy_param
        self.top_k = int(ap.top_k) if ap is not None else 1

    def reshape(self, bottom, top) -> None:
        top[0].reshape(())

    def forward(self, bottom, top) -> None:
        top[0].data = ops.accuracy(bottom[0].data, bottom[1].data,
                                   self.top_k).reshape(())

    def backward(self, top, propagate_down, bottom) -> None:
        raise NotImplementedError("ACCURACY has no backward")
