"""Op dispatch: hand-written HIP/CDNA4 kernels on GPU, plain fp32 torch on CPU.

The CPU path is the numerics reference for every GPU kernel (tests compare
the two). The GPU path REQUIRES the compiled in-tree extension
(poseidon_amd/ops/_hip) -- it raises rather than silently falling back to
eager torch, so a GPU run that passes is running our CDNA4 kernels.

Semantics mirror the reference layer math:
  conv      /root/reference/src/caffe/layers/conv_layer.{cpp,cu}
  pooling   /root/reference/src/caffe/layers/pooling_layer.cu (Caffe ceil
            geometry + far-edge-clipped AVE pool_size)
  lrn       /root/reference/src/caffe/layers/lrn_layer.cu
  softmax   /root/reference/src/caffe/layers/softmax_layer.cu
  sgd       /root/reference/src/caffe/solver.cpp:815-892 (fused here)
"""

from __future__ import annotations

import math
from typing import Optional, Tuple

import torch
import torch.nn.functional as F

_EXT = None
_EXT_ERR: Optional[str] = None


def _ext():
    global _EXT, _EXT_ERR
    if _EXT is None and _EXT_ERR is None:
        try:
            from . import _backend
            _EXT = _backend.load()
        except Exception as e:  # noqa: BLE001
            _EXT_ERR = str(e)
    if _EXT is None:
        raise RuntimeError(
            "poseidon_amd HIP extension is not built/loadable but a GPU op was "
            f"requested. Build it with `python setup.py build_ext --inplace` "
            f"(PYTORCH_ROCM_ARCH=gfx950). Original error: {_EXT_ERR}")
    return _EXT


def set_implicit_gemm(on: bool) -> None:
    """Toggle implicit-GEMM convolution (im2col gathered inside the MFMA
    GEMM staging; no column matrix). Default off: the materialized pipeline
    measures faster on current shapes."""
    _ext().set_implicit_gemm(bool(on))


def set_implicit_threshold(bytes_: int) -> None:
    """colT size above which a conv auto-selects implicit GEMM per layer."""
    if ext_available():
        _ext().set_implicit_threshold(int(bytes_))


def ext_available() -> bool:
    try:
        _ext()
        return True
    except RuntimeError:
        return False


# ---------------------------------------------------------------------------
# Geometry helpers (Caffe pooling/conv shape rules)
# ---------------------------------------------------------------------------

def conv_out_size(h: int, k: int, p: int, s: int) -> int:
    return (h + 2 * p - k) // s + 1


def pool_out_size(h: int, k: int, p: int, s: int) -> Tuple[int, bool]:
    """Caffe pooled size: ceil((h+2p-k)/s)+1, minus 1 when the last window
    would start in the padding (pooling_layer.cpp:64-77)."""
    out = int(math.ceil((h + 2 * p - k) / s)) + 1
    if p > 0 and (out - 1) * s >= h + p:
        out -= 1
    return out


# ---------------------------------------------------------------------------
# Convolution
# ---------------------------------------------------------------------------

def conv2d_forward_ex(x: torch.Tensor, w: torch.Tensor, b: Optional[torch.Tensor],
                      stride: Tuple[int, int], pad: Tuple[int, int],
                      groups: int, fuse_relu: bool = False, cached=None):
    """Returns (y, colT_cache). colT is the im2col matrix on GPU (reused by
    the backward GEMMs); None on CPU. fuse_relu clamps the output in the
    GEMM epilogue (used by the Net-level conv+ReLU fusion pass)."""
    if x.is_cuda:
        wk_c, wkT_c = cached if cached is not None else (None, None)
        y, colT, wkT = _ext().conv2d_forward_ex(x, w, b, stride[0], stride[1],
                                                pad[0], pad[1], groups,
                                                fuse_relu, wk_c, wkT_c)
        return y, (colT, wkT)
    y = F.conv2d(x, w, b, stride=stride, padding=pad, groups=groups)
    if fuse_relu:
        y = F.relu(y)
    return y, None


def conv2d_backward_input(w: torch.Tensor, dy: torch.Tensor,
                          x_shape, stride, pad, groups: int,
                          wkT_cache=None, dx_out=None) -> torch.Tensor:
    if dy.is_cuda:
        return _ext().conv2d_backward_input(w, dy, list(x_shape), stride[0],
                                            stride[1], pad[0], pad[1], groups,
                                            wkT_cache, dx_out)
    return torch.nn.grad.conv2d_input(list(x_shape), w, dy, stride=stride,
                                      padding=pad, groups=groups)


def conv2d_backward_weight_acc(x: torch.Tensor, colT, dy: torch.Tensor,
                               dw: torch.Tensor, db: Optional[torch.Tensor],
                               stride, pad, groups: int,
                               dwk_buf: Optional[torch.Tensor] = None,
                               skip_unpack: bool = False,
                               skip_db: bool = False
                               ) -> Optional[torch.Tensor]:
    """Accumulates into dw (NCHW) and db. Returns the khwc dwk scratch used
    on GPU: hand it back as dwk_buf on later iterations (keeping it zeroed
    each iteration via the net-level zero table) to skip the per-GEMM
    split-K memset."""
    if dy.is_cuda:
        return _ext().conv2d_backward_weight_acc(
            x, colT, dy, dw, db, stride[0], stride[1], pad[0], pad[1],
            groups, dwk_buf, skip_unpack, skip_db)
    dw.add_(torch.nn.grad.conv2d_weight(x, list(dw.shape), dy, stride=stride,
                                        padding=pad, groups=groups))
    if db is not None:
        db.add_(dy.sum(dim=(0, 2, 3)))
    return None


# ---------------------------------------------------------------------------
# Inner product (FC): y[M,N] = x[M,K] @ w[N,K]^T + b
# ---------------------------------------------------------------------------

def linear_forward(x: torch.Tensor, w: torch.Tensor,
                   b: Optional[torch.Tensor],
                   fuse_relu: bool = False,
                   w_shadow: Optional[torch.Tensor] = None) -> torch.Tensor:
    if x.is_cuda:
        return _ext().linear_forward(x, w, b, fuse_relu, w_shadow)
    y = x.matmul(w.t())
    if b is not None:
        y = y + b
    if fuse_relu:
        y = F.relu(y)
    return y


def linear_backward(x: torch.Tensor, w: torch.Tensor, dy: torch.Tensor,
                    need_dx: bool, need_dw: bool, has_bias: bool,
                    w_shadow: Optional[torch.Tensor] = None,
                    dw_acc: Optional[torch.Tensor] = None):
    """dw_acc: fp32 [N,K] buffer the weight grad is ACCUMULATED into
    (beta=1 GEMM); when set, the returned dw is None."""
    if dy.is_cuda:
        return _ext().linear_backward(x, w, dy, need_dx, need_dw, has_bias,
                                      w_shadow, dw_acc)
    dx = dy.matmul(w) if need_dx else None
    dw = dy.t().matmul(x) if need_dw else None
    db = dy.sum(dim=0) if has_bias else None
    return dx, dw, db


def gemm_at_b(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """a[M,N]^T-free outer-product GEMM used by SFB reconstruction:
    returns a^T @ b for a[B,N], b[B,K] -> [N,K]."""
    if a.is_cuda:
        return _ext().gemm_at_b(a, b)
    return a.t().matmul(b)


# ---------------------------------------------------------------------------
# Pooling
# ---------------------------------------------------------------------------

def _unfold_pad(x, k, s, p, pad_value: float, Ho: int, Wo: int):
    """Pad so that all Ho/Wo Caffe ceil-mode windows exist (the last window
    may extend past H+p), then unfold."""
    H, W = x.shape[2], x.shape[3]
    far_h = max(p[0], (Ho - 1) * s[0] + k[0] - H - p[0])
    far_w = max(p[1], (Wo - 1) * s[1] + k[1] - W - p[1])
    xp = F.pad(x, (p[1], far_w, p[0], far_h), value=pad_value)
    return xp.unfold(2, k[0], s[0]).unfold(3, k[1], s[1])


def pool_max_forward(x, k, s, p):
    if x.is_cuda:
        return _ext().pool_max_forward(x, k[0], k[1], s[0], s[1], p[0], p[1])
    N, C, H, W = x.shape
    Ho = pool_out_size(H, k[0], p[0], s[0])
    Wo = pool_out_size(W, k[1], p[1], s[1])
    win = _unfold_pad(x, k, s, p, float("-inf"), Ho, Wo)[:, :, :Ho, :Wo]
    flat = win.contiguous().view(N, C, Ho, Wo, -1)
    y, idx = flat.max(dim=-1)
    # map window-local argmax to bottom (unpadded) flat index h*W + w
    kh = idx // k[1]
    kw = idx % k[1]
    oh = torch.arange(Ho, device=x.device).view(1, 1, Ho, 1)
    ow = torch.arange(Wo, device=x.device).view(1, 1, 1, Wo)
    h = oh * s[0] - p[0] + kh
    wdx = ow * s[1] - p[1] + kw
    mask = (h * W + wdx).to(torch.int32)
    return y, mask


def pool_max_backward(dy, mask, x_shape, k=None, s_=None, p=None,
                      dx_out=None):
    if dy.is_cuda:
        return _ext().pool_max_backward(dy, mask, list(x_shape), k[0], k[1],
                                        s_[0], s_[1], p[0], p[1], dx_out)
    N, C, H, W = x_shape
    dx = torch.zeros(N, C, H * W, dtype=dy.dtype, device=dy.device)
    dx.scatter_add_(2, mask.view(N, C, -1).long(), dy.view(N, C, -1))
    return dx.view(N, C, H, W)


def _ave_pool_sizes(H, W, k, s, p, Ho, Wo, device):
    oh = torch.arange(Ho, device=device)
    ow = torch.arange(Wo, device=device)
    hstart = oh * s[0] - p[0]
    wstart = ow * s[1] - p[1]
    hend = torch.clamp(hstart + k[0], max=H + p[0])
    wend = torch.clamp(wstart + k[1], max=W + p[1])
    return ((hend - hstart).view(Ho, 1) * (wend - wstart).view(1, Wo)).float()


def pool_ave_forward(x, k, s, p):
    if x.is_cuda:
        return _ext().pool_ave_forward(x, k[0], k[1], s[0], s[1], p[0], p[1])
    N, C, H, W = x.shape
    Ho = pool_out_size(H, k[0], p[0], s[0])
    Wo = pool_out_size(W, k[1], p[1], s[1])
    win = _unfold_pad(x, k, s, p, 0.0, Ho, Wo)[:, :, :Ho, :Wo]
    ssum = win.contiguous().view(N, C, Ho, Wo, -1).sum(dim=-1)
    sizes = _ave_pool_sizes(H, W, k, s, p, Ho, Wo, x.device).to(x.dtype)
    return ssum / sizes


def pool_ave_backward(dy, x_shape, k, s, p):
    if dy.is_cuda:
        return _ext().pool_ave_backward(dy, list(x_shape), k[0], k[1],
                                        s[0], s[1], p[0], p[1])
    N, C, H, W = x_shape
    Ho, Wo = dy.shape[2], dy.shape[3]
    sizes = _ave_pool_sizes(H, W, k, s, p, Ho, Wo, dy.device).to(dy.dtype)
    g = (dy / sizes).view(N * C, 1, Ho, Wo)
    # distribute each output's gradient over its (real-region) window = fold
    Hp = max(H + 2 * p[0], (Ho - 1) * s[0] + k[0])
    Wp = max(W + 2 * p[1], (Wo - 1) * s[1] + k[1])
    cols = g.expand(N * C, k[0] * k[1], Ho, Wo).reshape(N * C, k[0] * k[1], Ho * Wo)
    dxp = F.fold(cols, (Hp, Wp), (k[0], k[1]), stride=(s[0], s[1]))
    dx = dxp[:, :, p[0]:p[0] + H, p[1]:p[1] + W]
    return dx.reshape(N, C, H, W)


def pool_stoch_forward_train(x, k, s, p, rand=None):
    """Stochastic pooling (train): pick an element of each window with
    probability proportional to its value (pooling_layer.cu:81-120)."""
    if x.is_cuda:
        return _ext().pool_stoch_forward_train(x, k[0], k[1], s[0], s[1],
                                               p[0], p[1],
                                               int(torch.randint(0, 2**31 - 1, ()).item()))
    N, C, H, W = x.shape
    Ho = pool_out_size(H, k[0], p[0], s[0])
    Wo = pool_out_size(W, k[1], p[1], s[1])
    win = _unfold_pad(x, k, s, p, 0.0, Ho, Wo)[:, :, :Ho, :Wo]
    flat = win.contiguous().view(N, C, Ho, Wo, -1)
    csum = flat.cumsum(dim=-1)
    total = csum[..., -1:]
    if rand is None:
        rand = torch.rand(N, C, Ho, Wo, 1, device=x.device, dtype=x.dtype)
    thresh = rand * total
    idx = (csum < thresh).sum(dim=-1).clamp(max=k[0] * k[1] - 1)
    y = flat.gather(-1, idx.unsqueeze(-1)).squeeze(-1)
    kh = idx // k[1]
    kw = idx % k[1]
    oh = torch.arange(Ho, device=x.device).view(1, 1, Ho, 1)
    ow = torch.arange(Wo, device=x.device).view(1, 1, 1, Wo)
    h = (oh * s[0] - p[0] + kh).clamp(0, H - 1)
    wdx = (ow * s[1] - p[1] + kw).clamp(0, W - 1)
    mask = (h * W + wdx).to(torch.int32)
    return y, mask


def pool_stoch_forward_test(x, k, s, p):
    if x.is_cuda:
        return _ext().pool_stoch_forward_test(x, k[0], k[1], s[0], s[1], p[0], p[1])
    N, C, H, W = x.shape
    Ho = pool_out_size(H, k[0], p[0], s[0])
    Wo = pool_out_size(W, k[1], p[1], s[1])
    win = _unfold_pad(x, k, s, p, 0.0, Ho, Wo)[:, :, :Ho, :Wo]
    flat = win.contiguous().view(N, C, Ho, Wo, -1)
    num = (flat * flat).sum(-1)
    den = flat.sum(-1)
    return num / (den + torch.finfo(x.dtype).tiny)


# ---------------------------------------------------------------------------
# LRN (across channels)
# ---------------------------------------------------------------------------

def lrn_forward(x, size: int, alpha: float, beta: float):
    if x.is_cuda:
        return _ext().lrn_forward(x, size, alpha, beta)
    N, C, H, W = x.shape
    sq = (x * x).view(N, 1, C, H * W)
    pad = (size - 1) // 2
    # sliding sum over channel dim: conv with ones kernel
    kernel = torch.ones(1, 1, size, 1, dtype=x.dtype, device=x.device)
    ssum = F.conv2d(sq, kernel, padding=(size - 1 - pad, 0))
    ssum = ssum[:, :, :C, :].view(N, C, H, W)
    scale = 1.0 + (alpha / size) * ssum
    y = x * scale.pow(-beta)
    return y, scale


def lrn_backward(x, y, scale, dy, size: int, alpha: float, beta: float):
    if x.is_cuda:
        return _ext().lrn_backward(x, y, scale, dy, size, alpha, beta)
    N, C, H, W = x.shape
    pad = (size - 1) // 2
    ratio = (dy * y / scale).view(N, 1, C, H * W)
    kernel = torch.ones(1, 1, size, 1, dtype=x.dtype, device=x.device)
    acc = F.conv2d(ratio, kernel, padding=(pad, 0))[:, :, :C, :].view(N, C, H, W)
    return dy * scale.pow(-beta) - (2.0 * alpha * beta / size) * x * acc


# ---------------------------------------------------------------------------
# Softmax + losses
# ---------------------------------------------------------------------------

def softmax_forward(x: torch.Tensor) -> torch.Tensor:
    """Channel softmax per spatial position (softmax_layer.cu)."""
    if x.is_cuda:
        return _ext().softmax_forward(x)
    return F.softmax(x, dim=1)


def softmax_backward(y: torch.Tensor, dy: torch.Tensor) -> torch.Tensor:
    if y.is_cuda:
        return _ext().softmax_backward(y, dy)
    dot = (dy * y).sum(dim=1, keepdim=True)
    return (dy - dot) * y


def softmax_loss_forward(logits: torch.Tensor, labels: torch.Tensor):
    """Returns (mean NLL over batch as 0-d tensor, probs)."""
    if logits.is_cuda:
        return _ext().softmax_loss_forward(logits, labels)
    prob = F.softmax(logits, dim=1)
    n = logits.shape[0]
    picked = prob[torch.arange(n), labels.long().view(-1)]
    loss = -torch.log(torch.clamp(picked, min=torch.finfo(prob.dtype).tiny)).sum() / n
    return loss, prob


def softmax_loss_backward(prob: torch.Tensor, labels: torch.Tensor,
                          loss_weight: float) -> torch.Tensor:
    if prob.is_cuda:
        return _ext().softmax_loss_backward(prob, labels, loss_weight)
    n = prob.shape[0]
    dx = prob.clone()
    dx[torch.arange(n), labels.long().view(-1)] -= 1.0
    return dx * (loss_weight / n)


# ---------------------------------------------------------------------------
# Elementwise / neuron ops
# ---------------------------------------------------------------------------

def relu_forward(x, negative_slope: float = 0.0):
    if x.is_cuda:
        return _ext().relu_forward(x, negative_slope)
    return F.leaky_relu(x, negative_slope)


def relu_backward(x, dy, negative_slope: float = 0.0,
                  in_place: bool = False):
    """in_place: write dx into dy's buffer (elementwise same-index safe).
    Keeps the grad identity stable for net-level batching."""
    if x.is_cuda:
        return _ext().relu_backward(x, dy, negative_slope, in_place)
    return torch.where(x > 0, dy, dy * negative_slope)


def sigmoid_forward(x):
    if x.is_cuda:
        return _ext().sigmoid_forward(x)
    return torch.sigmoid(x)


def sigmoid_backward(y, dy):
    if y.is_cuda:
        return _ext().sigmoid_backward(y, dy)
    return dy * y * (1 - y)


def tanh_forward(x):
    if x.is_cuda:
        return _ext().tanh_forward(x)
    return torch.tanh(x)


def tanh_backward(y, dy):
    if y.is_cuda:
        return _ext().tanh_backward(y, dy)
    return dy * (1 - y * y)


def bnll_forward(x):
    """log(1 + exp(x)), computed stably (bnll_layer.cu)."""
    if x.is_cuda:
        return _ext().bnll_forward(x)
    return F.softplus(x)


def bnll_backward(x, dy):
    if x.is_cuda:
        return _ext().bnll_backward(x, dy)
    return dy * torch.sigmoid(x)


def dropout_forward(x, ratio: float, seed: int, offset, offset_dev=None):
    """Train-mode dropout. Returns (y, mask) where mask is uint8 keep-mask;
    y = x * mask * 1/(1-ratio). On GPU, offset_dev (int64 device scalar)
    makes the op hipGraph-replayable (fresh mask per replay)."""
    if x.is_cuda:
        if offset_dev is not None:
            return _ext().dropout_forward_offdev(x, ratio, seed, offset_dev)
        return _ext().dropout_forward(x, ratio, seed, offset)
    scale = 1.0 / (1.0 - ratio)
    g = torch.Generator(device="cpu").manual_seed(seed + offset)
    mask = (torch.rand(x.shape, generator=g) >= ratio)
    return x * mask.to(x.dtype) * scale, mask


def dropout_backward(dy, mask, ratio: float):
    if dy.is_cuda:
        return _ext().dropout_backward(dy, mask, ratio)
    return dy * mask.to(dy.dtype) * (1.0 / (1.0 - ratio))


# ---------------------------------------------------------------------------
# Fused SGD update (solver.cpp:858-892 collapsed to one kernel)
#   hist = momentum*hist + local_rate*(grad + decay*w)
#   w   -= hist
# ---------------------------------------------------------------------------

def threshold_forward(x: torch.Tensor, thr: float) -> torch.Tensor:
    if x.is_cuda:
        return _ext().threshold_forward(x, float(thr))
    return (x > thr).to(x.dtype)


def eltwise_max(blobs) -> tuple:
    """Running pairwise max with a u8 argmax mask (ELTWISE MAX,
    eltwise_layer.cu:11 MaxForward semantics)."""
    if blobs[0].is_cuda:
        y = blobs[0].contiguous().clone()
        mask = torch.zeros(y.numel(), dtype=torch.uint8, device=y.device)
        for i, b in enumerate(blobs[1:], start=1):
            _ext().eltwise_max_step(y.view(-1), b.reshape(-1), mask, i)
        return y, mask.view(y.shape)
    stacked = torch.stack(list(blobs))
    y, idx = stacked.max(dim=0)
    return y, idx.to(torch.uint8)


def eltwise_max_backward(dy: torch.Tensor, mask: torch.Tensor,
                         idx: int) -> torch.Tensor:
    if dy.is_cuda:
        return _ext().eltwise_max_backward(dy.reshape(-1),
                                           mask.reshape(-1),
                                           idx).view(dy.shape)
    return dy * (mask == idx).to(dy.dtype)


def contrastive_terms(dist_sq: torch.Tensor, sim: torch.Tensor,
                      margin: float, legacy: bool) -> torch.Tensor:
    """Per-pair contrastive loss terms (contrastive_loss_layer.cu:49)."""
    if dist_sq.is_cuda:
        return _ext().contrastive_forward(dist_sq, sim, float(margin),
                                          bool(legacy))
    m = torch.clamp(margin - (dist_sq if legacy else dist_sq.sqrt()), min=0)
    return torch.where(sim != 0, dist_sq, m if legacy else m * m)


def sgd_update(w: torch.Tensor, grad: torch.Tensor, hist: torch.Tensor,
               local_rate: float, momentum: float, decay: float,
               lr_dev=None) -> None:
    if w.is_cuda:
        if lr_dev is not None:
            _ext().sgd_update_lrdev(w, grad, hist, local_rate, momentum,
                                    decay, lr_dev)
        else:
            _ext().sgd_update(w, grad, hist, local_rate, momentum, decay)
        return
    gw = grad if decay == 0.0 else grad + decay * w
    hist.mul_(momentum).add_(gw, alpha=local_rate)
    w.sub_(hist)


# -- multi-tensor apply: one launch for ALL params (GPU only) ---------------

def sgd_mt_prepare(ws, gs, hs, lr_mults, wds):
    """Build device-resident descriptor/chunk tables for sgd_mt_run. Shapes
    and tensor identities must stay fixed afterwards (they do: param blobs
    are allocated once at net build)."""
    d, c, n = _ext().sgd_mt_prepare(list(ws), list(gs), list(hs),
                                    [float(v) for v in lr_mults],
                                    [float(v) for v in wds])
    return d, c, int(n.item())


def sgd_mt_run(mt, lr: float, momentum: float, lr_dev=None) -> None:
    desc, chunk, nchunks = mt
    _ext().sgd_mt_run(desc, chunk, nchunks, float(lr), float(momentum),
                      lr_dev)


def conv_colT_ld(G: int, C: int, kh: int, kw: int, vec: int) -> int:
    return int(_ext().colT_ld(G, C, kh, kw, vec))


def repack_mt_prepare(masters, wks, wkTs, Gs):
    d, c, n = _ext().repack_mt_prepare(list(masters), list(wks), list(wkTs),
                                       [int(g) for g in Gs])
    return d, c, int(n.item())


def repack_mt_run(mt) -> None:
    d, c, n = mt
    _ext().repack_mt_run(d, c, n)


def unpack_mt_prepare(dwks, dws, Cigs, khs, kws):
    d, c, n = _ext().unpack_mt_prepare(list(dwks), list(dws), list(Cigs),
                                       list(khs), list(kws))
    return d, c, int(n)


def unpack_mt_run(mt) -> None:
    d, c, n = mt
    _ext().unpack_mt_run(d, c, n)


def colsum_mt_prepare(dys, dbs):
    d, c, n, mc = _ext().colsum_mt_prepare(list(dys), list(dbs))
    return d, c, int(n), int(mc)


def colsum_mt_run(mt, bf16: bool) -> None:
    d, c, n, mc = mt
    _ext().colsum_mt_run(d, c, n, bf16, mc)


def colsum_acc(dy, db) -> None:
    _ext().colsum_acc(dy, db)


def zero_mt_prepare(tensors):
    d, c, n = _ext().zero_mt_prepare(list(tensors))
    return d, c, int(n.item())


def zero_mt_run(mt) -> None:
    desc, chunk, nchunks = mt
    _ext().zero_mt_run(desc, chunk, nchunks)


def nesterov_update(w, grad, hist, local_rate: float, momentum: float,
                    decay: float) -> None:
    """update = (1+mu)*h_new - mu*h_old (solver.cpp:1013-1120)."""
    if w.is_cuda:
        _ext().nesterov_update(w, grad, hist, local_rate, momentum, decay)
        return
    gw = grad if decay == 0.0 else grad + decay * w
    h_old = hist.clone()
    hist.mul_(momentum).add_(gw, alpha=local_rate)
    w.sub_((1 + momentum) * hist - momentum * h_old)


def adagrad_update(w, grad, hist, local_rate: float, delta: float,
                   decay: float) -> None:
    """hist += g^2 ; w -= lr * g / (sqrt(hist)+delta) (solver.cpp:1240-1364)."""
    if w.is_cuda:
        _ext().adagrad_update(w, grad, hist, local_rate, delta, decay)
        return
    gw = grad if decay == 0.0 else grad + decay * w
    hist.add_(gw * gw)
    w.sub_(local_rate * gw / (hist.sqrt() + delta))


# ---------------------------------------------------------------------------
# NHWC channel concat / slice (CONCAT + SLICE layers)
# ---------------------------------------------------------------------------

def concat_channels(tensors):
    if tensors[0].is_cuda and tensors[0].dim() == 4:
        return _ext().concat_channels(list(tensors))
    return torch.cat(tensors, dim=1)


def slice_channels(x, c_off: int, c_len: int):
    if x.is_cuda and x.dim() == 4:
        return _ext().slice_channels(x, c_off, c_len)
    return x.narrow(1, c_off, c_len).contiguous()


def split_channels(x, sizes, outs_cache=None, relu_masks=None):
    """All channel ranges in one go: up to 4 ranges per kernel launch
    (concat backward / slice forward over inception joins). outs_cache:
    persistent output tensors (keeps downstream grad identities stable
    for net-level batching). relu_masks: per-output post-relu activation
    tensors -- outputs are zeroed where the activation is zero, fusing
    the consumer ReLU's backward into the scatter."""
    if x.is_cuda and x.dim() == 4:
        return _ext().split_channels(x, [int(s) for s in sizes], outs_cache,
                                     relu_masks)
    out, off = [], 0
    for s in sizes:
        out.append(x.narrow(1, off, int(s)).contiguous())
        off += int(s)
    return out


# ---------------------------------------------------------------------------
# Metrics
# ---------------------------------------------------------------------------

def accuracy(pred: torch.Tensor, labels: torch.Tensor, top_k: int = 1) -> torch.Tensor:
    n = pred.shape[0]
    flat = pred.reshape(n, -1)
    if top_k == 1:
        correct = (flat.argmax(dim=1) == labels.long().view(-1))
    else:
        topk = flat.topk(top_k, dim=1).indices
        correct = (topk == labels.long().view(-1, 1)).any(dim=1)
    return correct.to(torch.float32).sum() / n
