"""Vision layers: Convolution, Pooling, LRN, Im2col.

Parity: /root/reference/src/caffe/layers/{conv_layer,pooling_layer,lrn_layer,
im2col_layer}.{cpp,cu}. All GPU math routes through ops.functional into
hand-written CDNA4 kernels (implicit-GEMM MFMA conv, fused pool/LRN kernels).
"""

from __future__ import annotations

from typing import List

import torch
import torch.nn.functional as F

from ..core.blob import Blob
from ..core.layer import Layer, register_layer
from ..core import filler
from ..ops import functional as ops


def _pair(param, base: str, generic: str):
    """Resolve Caffe's (kernel_size | kernel_h/kernel_w)-style params."""
    h = getattr(param, f"{base}_h")
    w = getattr(param, f"{base}_w")
    if param.has(f"{base}_h") or param.has(f"{base}_w"):
        return int(h), int(w)
    g = getattr(param, generic)
    if g is None:
        raise ValueError(f"missing {generic} / {base}_h/w")
    return int(g), int(g)


@register_layer("CONVOLUTION")
class ConvolutionLayer(Layer):
    exact_num_top = None  # conv supports multiple bottom/top pairs

    def layer_setup(self, bottom: List[Blob], top: List[Blob]) -> None:
        cp = self.param.ensure("convolution_param")
        self.kernel = _pair(cp, "kernel", "kernel_size")
        self.stride = _pair(cp, "stride", "stride")
        ph, pw = cp.pad_h, cp.pad_w
        self.pad = (int(ph), int(pw)) if (cp.has("pad_h") or cp.has("pad_w")) \
            else (int(cp.pad), int(cp.pad))
        self.group = int(cp.group)
        self.num_output = int(cp.num_output)
        self.bias_term = bool(cp.bias_term)
        self.fuse_relu = False  # set by Net's conv+ReLU fusion pass
        self._wk_cache = None   # (wk, wkT) persistent buffers filled by the
                                # net-level multi-tensor repack each step
        self._dwk_cache = None  # persistent khwc wgrad scratch (zeroed per
                                # iter by the net zero table)
        self._unpack_pending = False  # deferred wgrad unpack flag
        self._dx_cache = {}     # persistent dgrad outputs (stable dy
                                # identities for downstream batching)
        self._pending_colsum = None  # (dy, db) for the net-level batch
        self._last_dy_ptr = None     # stability tracker for the batch
        channels = bottom[0].channels
        assert channels % self.group == 0 and self.num_output % self.group == 0

        # params are fp32 masters even when activations are bf16
        # (f64 preserved for CPU finite-difference checks)
        dtype = torch.float64 if bottom[0].dtype == torch.float64 else torch.float32
        w = Blob((self.num_output, channels // self.group,
                  self.kernel[0], self.kernel[1]), dtype=dtype,
                 name=f"{self.name}.weight")
        filler.fill(w, cp.weight_filler if cp.has("weight_filler") else None)
        self.blobs = [w]
        if self.bias_term:
            # Caffe conv bias blob is (1,1,1,Cout) (conv_layer.cpp:96-99)
            b = Blob((1, 1, 1, self.num_output), dtype=dtype,
                     name=f"{self.name}.bias")
            filler.fill(b, cp.bias_filler if cp.has("bias_filler") else None)
            self.blobs.append(b)

    def reshape(self, bottom: List[Blob], top: List[Blob]) -> None:
        n, c, h, w = bottom[0].shape
        ho = ops.conv_out_size(h, self.kernel[0], self.pad[0], self.stride[0])
        wo = ops.conv_out_size(w, self.kernel[1], self.pad[1], self.stride[1])
        for t in top:
            t.reshape(n, self.num_output, ho, wo)

    def forward(self, bottom: List[Blob], top: List[Blob]) -> None:
        w = self.blobs[0].data
        b = self.blobs[1].data.view(-1) if self.bias_term else None
        self._colT = []
        for bo, t in zip(bottom, top):
            y, colT = ops.conv2d_forward_ex(bo.data, w, b, self.stride,
                                            self.pad, self.group,
                                            fuse_relu=self.fuse_relu,
                                            cached=self._wk_cache)
            t.data = y
            self._colT.append(colT)

    def backward(self, top: List[Blob], propagate_down: List[bool],
                 bottom: List[Blob]) -> None:
        w = self.blobs[0].data
        single = len(bottom) == 1  # dwk reuse assumes one wgrad write/iter
        # Deferred unpack (single-GPU): leave the gradient in the khwc
        # scratch; Net.backward runs ONE unpack_mt kernel for all convs at
        # the end (DWBP multi-rank mode needs per-layer grads final, so the
        # solver only sets defer_unpack when there is no reducer).
        defer = (single and getattr(self, "defer_unpack", False)
                 and self._dwk_cache is not None)
        for i, (bo, t) in enumerate(zip(bottom, top)):
            dy = t.diff
            # Defer the bias colsum into the net-level batch ONLY when this
            # layer's dy identity repeated from last iteration (producers
            # with persistent grad buffers): an unstable dy would force a
            # table rebuild (H2D) every step. Unstable ones keep the
            # in-call colsum, which also overlaps on the backward streams.
            vecw = 8 if dy.dtype == torch.bfloat16 else 4
            ptr = (dy.data_ptr() if dy.is_cuda and dy.dim() == 4
                   and dy.shape[1] % vecw == 0
                   and dy.is_contiguous(memory_format=torch.channels_last)
                   else None)
            defer_db = (defer and self.bias_term and ptr is not None
                        and ptr == self._last_dy_ptr)
            self._last_dy_ptr = ptr
            db = self.blobs[1].diff.view(-1) if self.bias_term else None
            cache = self._colT[i]
            colT, wkT = cache if isinstance(cache, tuple) else (cache, None)
            dwk = ops.conv2d_backward_weight_acc(
                bo.data, colT, dy, self.blobs[0].diff, db,
                self.stride, self.pad, self.group,
                dwk_buf=self._dwk_cache if single else None,
                skip_unpack=defer, skip_db=defer_db)
            if defer_db:
                # bias grad joins the ONE net-level colsum_mt launch
                self._pending_colsum = (dy, db)
            if defer:
                if dwk.data_ptr() != self._dwk_cache.data_ptr():
                    raise RuntimeError(
                        f"{self.name}: conv geometry changed under deferred "
                        "wgrad unpack (dwk buffer was reallocated)")
                self._unpack_pending = True
            if single and dwk is not None:
                # persistent khwc wgrad scratch: the Net's zero table zeroes
                # it each iteration, letting the atomic split-K GEMM skip
                # its per-launch memset. Keep the SAME python object when
                # the storage is unchanged -- the binding returns a fresh
                # wrapper around the buffer we passed in, and rebinding it
                # would churn the zero-table identity key every iteration
                # (a rebuild + H2D copy per iter, and a capture abort).
                cur = self._dwk_cache
                if (cur is None or cur.data_ptr() != dwk.data_ptr()
                        or cur.numel() != dwk.numel()):
                    self._dwk_cache = dwk
            if propagate_down[i]:
                dx = ops.conv2d_backward_input(
                    w, dy, bo.shape, self.stride, self.pad, self.group,
                    wkT_cache=wkT, dx_out=self._dx_cache.get(i))
                cur = self._dx_cache.get(i)
                if dx.is_cuda and (cur is None
                                   or cur.data_ptr() != dx.data_ptr()):
                    self._dx_cache[i] = dx
                bo.diff = self._dx_cache.get(i, dx) if dx.is_cuda else dx
        self._colT = []

    def extra_zero_buffers(self) -> List[torch.Tensor]:
        return [self._dwk_cache] if self._dwk_cache is not None else []


@register_layer("POOLING")
class PoolingLayer(Layer):
    exact_num_bottom = 1

    def layer_setup(self, bottom, top) -> None:
        pp = self.param.ensure("pooling_param")
        self.kernel = _pair(pp, "kernel", "kernel_size")
        self.stride = _pair(pp, "stride", "stride")
        self.pad = (int(pp.pad_h), int(pp.pad_w)) \
            if (pp.has("pad_h") or pp.has("pad_w")) else (int(pp.pad), int(pp.pad))
        self.method = pp.enum_name("pool")
        if self.method != "AVE":
            assert self.pad == (0, 0) or self.method == "MAX"
        self._mask = None

    def reshape(self, bottom, top) -> None:
        n, c, h, w = bottom[0].shape
        ho = ops.pool_out_size(h, self.kernel[0], self.pad[0], self.stride[0])
        wo = ops.pool_out_size(w, self.kernel[1], self.pad[1], self.stride[1])
        top[0].reshape(n, c, ho, wo)
        if len(top) > 1:
            top[1].reshape(n, c, ho, wo)

    def forward(self, bottom, top) -> None:
        x = bottom[0].data
        if self.method == "MAX":
            y, mask = ops.pool_max_forward(x, self.kernel, self.stride, self.pad)
            self._mask = mask
            if len(top) > 1:
                top[1].data = self._mask_to_spatial(mask, x).to(x.dtype)
        elif self.method == "AVE":
            y = ops.pool_ave_forward(x, self.kernel, self.stride, self.pad)
        else:  # STOCHASTIC
            if self.phase == 0:
                y, mask = ops.pool_stoch_forward_train(
                    x, self.kernel, self.stride, self.pad)
                self._mask = mask
            else:
                y = ops.pool_stoch_forward_test(
                    x, self.kernel, self.stride, self.pad)
        top[0].data = y

    def _mask_to_spatial(self, mask, x):
        """GPU masks are u8 window-local argmax indices; Caffe's optional
        mask top wants the bottom spatial index h*W + w."""
        if not mask.is_cuda:
            return mask
        N, C, Ho, Wo = mask.shape
        W = x.shape[3]
        dev = mask.device
        oh = torch.arange(Ho, device=dev).view(1, 1, Ho, 1)
        ow = torch.arange(Wo, device=dev).view(1, 1, 1, Wo)
        kh = mask.long() // self.kernel[1]
        kw = mask.long() % self.kernel[1]
        h = oh * self.stride[0] - self.pad[0] + kh
        w = ow * self.stride[1] - self.pad[1] + kw
        return h * W + w

    def backward(self, top, propagate_down, bottom) -> None:
        if not propagate_down[0]:
            return
        dy = top[0].diff
        if self.method == "AVE":
            bottom[0].diff = ops.pool_ave_backward(
                dy, bottom[0].shape, self.kernel, self.stride, self.pad)
        else:  # MAX and STOCHASTIC both scatter via stored mask
            dx = ops.pool_max_backward(
                dy, self._mask, bottom[0].shape, self.kernel, self.stride,
                self.pad, dx_out=getattr(self, "_dx_cache", None))
            if dx.is_cuda:
                cur = getattr(self, "_dx_cache", None)
                if cur is None or cur.data_ptr() != dx.data_ptr():
                    self._dx_cache = dx
                bottom[0].diff = self._dx_cache
            else:
                bottom[0].diff = dx


@register_layer("LRN")
class LRNLayer(Layer):
    exact_num_bottom = 1
    exact_num_top = 1

    def layer_setup(self, bottom, top) -> None:
        lp = self.param.ensure("lrn_param")
        self.size = int(lp.local_size)
        assert self.size % 2 == 1, "LRN local_size must be odd"
        self.alpha = float(lp.alpha)
        self.beta = float(lp.beta)
        self.region = lp.enum_name("norm_region")
        self._scale = None
        self._y = None
        if self.region == "WITHIN_CHANNEL":
            # composite: x^2 -> ave-pool(size) -> scale -> x * scale^-beta
            self.pre_pad = (self.size - 1) // 2

    def reshape(self, bottom, top) -> None:
        top[0].reshape(bottom[0].shape)

    def forward(self, bottom, top) -> None:
        x = bottom[0].data
        if self.region == "ACROSS_CHANNELS":
            y, scale = ops.lrn_forward(x, self.size, self.alpha, self.beta)
        else:
            k = (self.size, self.size)
            sq = x * x
            avg = ops.pool_ave_forward(sq, k, (1, 1), (self.pre_pad, self.pre_pad))
            scale = 1.0 + self.alpha * avg
            y = x * scale.pow(-self.beta)
        self._scale = scale
        self._y = y
        top[0].data = y

    def backward(self, top, propagate_down, bottom) -> None:
        if not propagate_down[0]:
            return
        x, dy = bottom[0].data, top[0].diff
        if self.region == "ACROSS_CHANNELS":
            bottom[0].diff = ops.lrn_backward(x, self._y, self._scale, dy,
                                              self.size, self.alpha, self.beta)
        else:
            scale = self._scale
            k = (self.size, self.size)
            # dx = dy*scale^-b - 2*a*b/size^2-less: within-channel window avg
            ratio = dy * self._y / scale
            spread = ops.pool_ave_backward(ratio, bottom[0].shape, k, (1, 1),
                                           (self.pre_pad, self.pre_pad))
            bottom[0].diff = dy * scale.pow(-self.beta) \
                - 2.0 * self.alpha * self.beta * x * spread


@register_layer("IM2COL")
class Im2colLayer(Layer):
    exact_num_bottom = 1
    exact_num_top = 1

    def layer_setup(self, bottom, top) -> None:
        cp = self.param.ensure("convolution_param")
        self.kernel = _pair(cp, "kernel", "kernel_size")
        self.stride = _pair(cp, "stride", "stride")
        self.pad = (int(cp.pad), int(cp.pad))

    def reshape(self, bottom, top) -> None:
        n, c, h, w = bottom[0].shape
        ho = ops.conv_out_size(h, self.kernel[0], self.pad[0], self.stride[0])
        wo = ops.conv_out_size(w, self.kernel[1], self.pad[1], self.stride[1])
        top[0].reshape(n, c * self.kernel[0] * self.kernel[1], ho * wo)

    def forward(self, bottom, top) -> None:
        top[0].data = F.unfold(bottom[0].data, self.kernel, padding=self.pad,
                               stride=self.stride)

    def backward(self, top, propagate_down, bottom) -> None:
        if not propagate_down[0]:
            return
        n, c, h, w = bottom[0].shape
        bottom[0].diff = F.fold(top[0].diff, (h, w), self.kernel,
                                padding=self.pad, stride=self.stride)
