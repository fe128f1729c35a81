"""Common layers: InnerProduct, Concat, Slice, Split, Flatten, Eltwise, MVN,
Silence, ArgMax.

Parity: /root/reference/src/caffe/layers/{inner_product,concat,slice,split,
flatten,eltwise,mvn,silence,argmax}_layer.{cpp,cu}. The InnerProduct layer
carries the Sufficient-Factor-Broadcast hook: with SFB on (multi-rank), the
local ∂W GEMM is skipped and the (top_diff, bottom_data) factor pair is
exposed for the solver's all-gather + MFMA outer-product reconstruction
(reference: inner_product_layer.cu:31-64, solver.cpp:477-531).
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

from ..core.blob import Blob
from ..core.layer import Layer, register_layer
from ..core import filler
from ..ops import functional as ops


@register_layer("INNER_PRODUCT")
class InnerProductLayer(Layer):
    exact_num_bottom = 1
    exact_num_top = 1

    def layer_setup(self, bottom, top) -> None:
        ip = self.param.ensure("inner_product_param")
        self.N = int(ip.num_output)
        self.bias_term = bool(ip.bias_term)
        self.K = bottom[0].count // bottom[0].num
        dtype = torch.float64 if bottom[0].dtype == torch.float64 else torch.float32
        # Caffe IP weight blob is literally (1,1,N,K), bias (1,1,1,N)
        # (inner_product_layer.cpp:83-99) -- keep those shapes for
        # .caffemodel byte-compat; compute views them as (N,K)/(N,).
        w = Blob((1, 1, self.N, self.K), dtype=dtype, name=f"{self.name}.weight")
        filler.fill(w, ip.weight_filler if ip.has("weight_filler") else None)
        self.blobs = [w]
        if self.bias_term:
            b = Blob((1, 1, 1, self.N), dtype=dtype, name=f"{self.name}.bias")
            filler.fill(b, ip.bias_filler if ip.has("bias_filler") else None)
            self.blobs.append(b)
        self.fuse_relu = False  # set by Net's IP+ReLU fusion pass
        self._wk_cache = None   # bf16 [N,K] shadow, refreshed each step by
                                # the net-level repack table
        # SFB: when set by the distributed solver, backward defers the dW GEMM
        self.sfb_active = False
        self.sfb_factors: Optional[Tuple[torch.Tensor, torch.Tensor]] = None

    def reshape(self, bottom, top) -> None:
        self.M = bottom[0].num
        assert bottom[0].count // self.M == self.K, \
            f"{self.name}: fan-in changed ({bottom[0].count // self.M} vs {self.K})"
        top[0].reshape(self.M, self.N)

    def forward(self, bottom, top) -> None:
        x = bottom[0].data.reshape(self.M, self.K)
        b = self.blobs[1].data.view(-1) if self.bias_term else None
        top[0].data = ops.linear_forward(
            x, self.blobs[0].data.view(self.N, self.K), b,
            fuse_relu=self.fuse_relu, w_shadow=self._wk_cache)

    def backward(self, top, propagate_down, bottom) -> None:
        x = bottom[0].data.reshape(self.M, self.K)
        dy = top[0].diff.reshape(self.M, self.N)
        need_dw = not self.sfb_active
        # beta=1 in-GEMM grad accumulation measured slightly SLOWER than
        # the separate add_ on AlexNet fc shapes (same-box A/B); opt in
        # with PS_DW_ACC=1
        dw_acc = None
        import os as _os
        if (need_dw and dy.is_cuda
                and self.blobs[0].diff.dtype == torch.float32
                and _os.environ.get("PS_DW_ACC", "0") == "1"):
            dw_acc = self.blobs[0].diff.view(self.N, self.K)
        dx, dw, db = ops.linear_backward(
            x, self.blobs[0].data.view(self.N, self.K), dy,
            need_dx=propagate_down[0], need_dw=need_dw,
            has_bias=self.bias_term, w_shadow=self._wk_cache,
            dw_acc=dw_acc)
        if self.sfb_active:
            # Sufficient factors a=dy [M,N], b=x [M,K]; ∇W = aᵀ·b is
            # reconstructed after the all-gather (solver/sfb.py).
            self.sfb_factors = (dy, x)
        elif dw_acc is None:
            self.blobs[0].diff.view(self.N, self.K).add_(dw)
        if self.bias_term:
            self.blobs[1].diff.view(-1).add_(db)
        if propagate_down[0]:
            bottom[0].diff = dx.reshape(bottom[0].shape)


@register_layer("CONCAT")
class ConcatLayer(Layer):
    min_bottom = 1
    exact_num_top = 1

    def layer_setup(self, bottom, top) -> None:
        cp = self.param.concat_param
        self.dim = int(cp.concat_dim) if cp is not None else 1

    def reshape(self, bottom, top) -> None:
        shape = list(bottom[0].shape)
        shape[self.dim] = sum(b.shape[self.dim] for b in bottom)
        top[0].reshape(shape)

    def forward(self, bottom, top) -> None:
        if self.dim == 1 and len(bottom[0].shape) == 4:
            top[0].data = ops.concat_channels([b.data for b in bottom])
        else:
            top[0].data = torch.cat([b.data for b in bottom], dim=self.dim)

    def backward(self, top, propagate_down, bottom) -> None:
        chan4 = self.dim == 1 and len(bottom[0].shape) == 4
        if chan4 and all(propagate_down):
            cache = getattr(self, "_split_cache", None)
            # fused consumer-ReLU backward: bottoms flagged by the net pass
            # get their gradient zeroed where the (post-relu) activation is
            # zero, right in the scatter -- the ReLU layer skips backward
            mask_idx = getattr(self, "_mask_bottoms", None)
            masks = None
            if mask_idx:
                masks = [bottom[j].data if j in mask_idx else None
                         for j in range(len(bottom))]
            parts = ops.split_channels(top[0].diff,
                                       [b.shape[1] for b in bottom],
                                       outs_cache=cache, relu_masks=masks)
            if parts and parts[0].is_cuda:
                if (cache is None or len(cache) != len(parts) or any(
                        c.data_ptr() != p.data_ptr()
                        for c, p in zip(cache, parts))):
                    self._split_cache = parts
                parts = self._split_cache
            for b, d in zip(bottom, parts):
                b.diff = d
            return
        offset = 0
        for i, b in enumerate(bottom):
            n = b.shape[self.dim]
            if propagate_down[i]:
                if chan4:
                    b.diff = ops.slice_channels(top[0].diff, offset, n)
                else:
                    b.diff = top[0].diff.narrow(self.dim, offset, n).contiguous()
            offset += n


@register_layer("SLICE")
class SliceLayer(Layer):
    exact_num_bottom = 1
    min_top = 1

    def layer_setup(self, bottom, top) -> None:
        sp = self.param.slice_param
        self.dim = int(sp.slice_dim) if sp is not None else 1
        self.points = [int(x) for x in (sp.slice_point if sp is not None else [])]

    def reshape(self, bottom, top) -> None:
        total = bottom[0].shape[self.dim]
        if self.points:
            bounds = [0] + self.points + [total]
        else:
            assert total % len(top) == 0
            step = total // len(top)
            bounds = list(range(0, total + 1, step))
        self.sizes = [bounds[i + 1] - bounds[i] for i in range(len(top))]
        for t, sz in zip(top, self.sizes):
            shape = list(bottom[0].shape)
            shape[self.dim] = sz
            t.reshape(shape)

    def forward(self, bottom, top) -> None:
        chan4 = self.dim == 1 and len(bottom[0].shape) == 4
        if chan4:
            for t, d in zip(top, ops.split_channels(bottom[0].data,
                                                    self.sizes)):
                t.data = d
            return
        offset = 0
        for t, sz in zip(top, self.sizes):
            t.data = bottom[0].data.narrow(self.dim, offset, sz).contiguous()
            offset += sz

    def backward(self, top, propagate_down, bottom) -> None:
        if not propagate_down[0]:
            return
        if self.dim == 1 and len(bottom[0].shape) == 4:
            bottom[0].diff = ops.concat_channels([t.diff for t in top])
        else:
            bottom[0].diff = torch.cat([t.diff for t in top], dim=self.dim)


@register_layer("SPLIT")
class SplitLayer(Layer):
    exact_num_bottom = 1
    min_top = 1

    def reshape(self, bottom, top) -> None:
        for t in top:
            t.reshape(bottom[0].shape)

    def forward(self, bottom, top) -> None:
        for t in top:
            t.share_data(bottom[0])

    def backward(self, top, propagate_down, bottom) -> None:
        if not propagate_down[0]:
            return
        acc = top[0].diff.clone()
        for t in top[1:]:
            acc.add_(t.diff)
        bottom[0].diff = acc


@register_layer("FLATTEN")
class FlattenLayer(Layer):
    exact_num_bottom = 1
    exact_num_top = 1

    def reshape(self, bottom, top) -> None:
        top[0].reshape(bottom[0].num, bottom[0].count // bottom[0].num)

    def forward(self, bottom, top) -> None:
        top[0].data = bottom[0].data.reshape(bottom[0].num, -1)

    def backward(self, top, propagate_down, bottom) -> None:
        if propagate_down[0]:
            bottom[0].diff = top[0].diff.reshape(bottom[0].shape)


@register_layer("ELTWISE")
class EltwiseLayer(Layer):
    min_bottom = 2
    exact_num_top = 1

    def layer_setup(self, bottom, top) -> None:
        ep = self.param.eltwise_param
        self.op = ep.enum_name("operation") if ep is not None else "SUM"
        coeffs = list(ep.coeff) if ep is not None else []
        if coeffs and len(coeffs) != len(bottom):
            raise ValueError("eltwise coeff count must match bottom count")
        self.coeffs = coeffs or [1.0] * len(bottom)
        self.stable_prod = bool(ep.stable_prod_grad) if ep is not None else True

    def reshape(self, bottom, top) -> None:
        top[0].reshape(bottom[0].shape)

    def forward(self, bottom, top) -> None:
        if self.op == "SUM":
            y = self.coeffs[0] * bottom[0].data
            for c, b in zip(self.coeffs[1:], bottom[1:]):
                y = y + c * b.data
        elif self.op == "PROD":
            y = bottom[0].data.clone()
            for b in bottom[1:]:
                y.mul_(b.data)
        else:  # MAX with argmax mask (MaxForward, eltwise_layer.cu:11)
            y, self._argmax = ops.eltwise_max([b.data for b in bottom])
        top[0].data = y

    def backward(self, top, propagate_down, bottom) -> None:
        dy = top[0].diff
        for i, b in enumerate(bottom):
            if not propagate_down[i]:
                continue
            if self.op == "SUM":
                b.diff = dy * self.coeffs[i]
            elif self.op == "PROD":
                if self.stable_prod:
                    prod = None
                    for j, other in enumerate(bottom):
                        if j == i:
                            continue
                        prod = other.data.clone() if prod is None else prod * other.data
                    b.diff = dy * prod
                else:
                    b.diff = dy * top[0].data / b.data
            else:  # MAX
                b.diff = ops.eltwise_max_backward(dy, self._argmax, i)


@register_layer("MVN")
class MVNLayer(Layer):
    exact_num_bottom = 1
    exact_num_top = 1

    def layer_setup(self, bottom, top) -> None:
        mp = self.param.mvn_param
        self.norm_var = bool(mp.normalize_variance) if mp is not None else True
        self.across = bool(mp.across_channels) if mp is not None else False
        self.eps = 1e-10

    def reshape(self, bottom, top) -> None:
        top[0].reshape(bottom[0].shape)

    def _dims(self, x):
        return (1, 2, 3) if self.across else (2, 3)

    def forward(self, bottom, top) -> None:
        x = bottom[0].data
        dims = self._dims(x)
        mean = x.mean(dim=dims, keepdim=True)
        if self.norm_var:
            var = (x * x).mean(dim=dims, keepdim=True) - mean * mean
            std = (var + self.eps).sqrt()
            self._std = std
            top[0].data = (x - mean) / std
        else:
            top[0].data = x - mean

    def backward(self, top, propagate_down, bottom) -> None:
        if not propagate_down[0]:
            return
        x, dy = bottom[0].data, top[0].diff
        dims = self._dims(x)
        m = 1.0
        for d in dims:
            m *= x.shape[d]
        if self.norm_var:
            y = top[0].data
            # Caffe mvn_layer.cpp backward: dx = (dy - mean(dy) - y*mean(dy*y))/std
            mean_dy = dy.mean(dim=dims, keepdim=True)
            mean_dyy = (dy * y).mean(dim=dims, keepdim=True)
            bottom[0].diff = (dy - mean_dy - y * mean_dyy) / self._std
        else:
            bottom[0].diff = dy - dy.mean(dim=dims, keepdim=True)


@register_layer("SILENCE")
class SilenceLayer(Layer):
    def reshape(self, bottom, top) -> None:
        pass

    def forward(self, bottom, top) -> None:
        pass

    def backward(self, top, propagate_down, bottom) -> None:
        for i, b in enumerate(bottom):
            if propagate_down[i]:
                b.diff.zero_()


@register_layer("ARGMAX")
class ArgMaxLayer(Layer):
    exact_num_bottom = 1
    exact_num_top = 1

    def layer_setup(self, bottom, top) -> None:
        ap = self.param.argmax_param
        self.out_max_val = bool(ap.out_max_val) if ap is not None else False
        self.top_k = int(ap.top_k) if ap is not None else 1

    def reshape(self, bottom, top) -> None:
        n = bottom[0].num
        top[0].reshape(n, 2 if self.out_max_val else 1, self.top_k, 1)

    def forward(self, bottom, top) -> None:
        x = bottom[0].data.reshape(bottom[0].num, -1)
        vals, idx = x.topk(self.top_k, dim=1)
        if self.out_max_val:
            top[0].data = torch.stack(
                [idx.to(x.dtype), vals], dim=1).view(top[0].shape)
        else:
            top[0].data = idx.to(x.dtype).view(top[0].shape)

    def backward(self, top, propagate_down, bottom) -> None:
        raise NotImplementedError("ARGMAX has no backward")
