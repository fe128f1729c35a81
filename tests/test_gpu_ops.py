"""GPU kernel numerics: every HIP/CDNA4 kernel vs the plain fp32 torch CPU
reference in ops/functional.py. Asymmetric inputs + odd shapes catch
transposed fragments and edge-guard bugs (guide §5.4 rule 16)."""

import pytest
import torch

import poseidon_amd as pa
from poseidon_amd.ops import functional as ops

pytestmark = pytest.mark.gpu

DEV = "cuda"


def setup_module():
    assert ops.ext_available(), "HIP extension must be loadable on GPU hosts"


def rnd(*shape, seed=0, scale=1.0):
    g = torch.Generator().manual_seed(seed)
    return torch.randn(*shape, generator=g) * scale


def close(a, b, rtol=2e-4, atol=2e-4, what=""):
    a = a.detach().cpu().float()
    b = b.detach().cpu().float()
    assert a.shape == b.shape, f"{what}: shape {a.shape} vs {b.shape}"
    err = (a - b).abs()
    denom = b.abs().clamp(min=1.0)
    rel = (err / denom).max().item()
    assert rel <= rtol or err.max().item() <= atol, \
        f"{what}: max abs {err.max():.3e} rel {rel:.3e}"


# ---------------------------------------------------------------------------
# GEMM
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("ak,bk", [(True, True), (True, False),
                                   (False, True), (False, False)])
@pytest.mark.parametrize("M,N,K", [(128, 128, 64), (190, 70, 85), (64, 257, 300),
                                   (64, 576, 25088)])  # wgrad shape: atomic split-K
def test_gemm_layouts(ak, bk, M, N, K):
    from poseidon_amd.ops._backend import load
    ext = load()
    opA = rnd(M, K, seed=1)
    opB = rnd(K, N, seed=2)
    ref = opA @ opB
    A = opA.contiguous() if ak else opA.t().contiguous()
    B = opB.t().contiguous() if bk else opB.contiguous()
    out = ext.gemm(A.to(DEV), B.to(DEV), M, N, K, ak, bk)
    # f32 rounding grows ~sqrt(K) with accumulation depth (split-K partial
    # order differs from torch's); loosen for the deep-K wgrad shape
    tol = 2e-4 if K < 4096 else 1e-3
    close(out, ref, rtol=tol, atol=tol, what=f"gemm ak={ak} bk={bk} {M}x{N}x{K}")


def test_linear_fwd_bwd():
    M, K, N = 37, 130, 75
    x, w, b = rnd(M, K, seed=3), rnd(N, K, seed=4, scale=0.2), rnd(N, seed=5)
    y_ref = ops.linear_forward(x, w, b)
    y = ops.linear_forward(x.to(DEV), w.to(DEV), b.to(DEV))
    close(y, y_ref, what="linear fwd")

    dy = rnd(M, N, seed=6)
    dx_r, dw_r, db_r = ops.linear_backward(x, w, dy, True, True, True)
    dx, dw, db = ops.linear_backward(x.to(DEV), w.to(DEV), dy.to(DEV),
                                     True, True, True)
    close(dx, dx_r, what="linear dx")
    close(dw, dw_r, what="linear dw")
    close(db, db_r, what="linear db")


def test_gemm_at_b():
    a, b = rnd(50, 33, seed=7), rnd(50, 44, seed=8)
    ref = ops.gemm_at_b(a, b)
    out = ops.gemm_at_b(a.to(DEV), b.to(DEV))
    close(out, ref, what="gemm_at_b")


# ---------------------------------------------------------------------------
# Convolution
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("cfg", [
    dict(N=2, C=3, H=19, W=19, Co=8, k=3, s=1, p=1, g=1),
    dict(N=2, C=3, H=35, W=35, Co=16, k=11, s=4, p=0, g=1),   # AlexNet conv1
    dict(N=2, C=8, H=13, W=13, Co=12, k=5, s=1, p=2, g=2),    # grouped
    dict(N=3, C=16, H=9, W=9, Co=24, k=1, s=1, p=0, g=1),     # 1x1 fast path
    dict(N=1, C=4, H=8, W=8, Co=6, k=3, s=2, p=1, g=1),
    dict(N=2, C=8, H=12, W=12, Co=10, k=3, s=1, p=1, g=1),   # implicit (f32 V=4)
    dict(N=2, C=64, H=10, W=10, Co=32, k=3, s=1, p=1, g=1),  # implicit + glds
    dict(N=2, C=16, H=9, W=9, Co=8, k=5, s=2, p=2, g=2),     # implicit grouped
])
@pytest.mark.parametrize("implicit", [False, True])
def test_conv_forward_backward(cfg, implicit):
    ops.set_implicit_gemm(implicit)
    try:
        _run_conv_case(cfg)
    finally:
        ops.set_implicit_gemm(False)


def _run_conv_case(cfg):
    N, C, H, W = cfg["N"], cfg["C"], cfg["H"], cfg["W"]
    Co, k, s, p, g = cfg["Co"], cfg["k"], cfg["s"], cfg["p"], cfg["g"]
    x = rnd(N, C, H, W, seed=11)
    w = rnd(Co, C // g, k, k, seed=12, scale=0.2)
    b = rnd(Co, seed=13)
    stride, pad = (s, s), (p, p)

    y_ref, _ = ops.conv2d_forward_ex(x, w, b, stride, pad, g)
    y, cache = ops.conv2d_forward_ex(x.to(DEV), w.to(DEV), b.to(DEV), stride, pad, g)
    colT, _wkT = cache
    close(y, y_ref, what="conv fwd")

    dy = rnd(*y_ref.shape, seed=14)
    dx_ref = ops.conv2d_backward_input(w, dy, x.shape, stride, pad, g)
    dx = ops.conv2d_backward_input(w.to(DEV), dy.to(DEV), x.shape, stride, pad, g)
    close(dx, dx_ref, what="conv dgrad")

    dw_ref = torch.zeros_like(w)
    db_ref = torch.zeros(Co)
    ops.conv2d_backward_weight_acc(x, None, dy, dw_ref, db_ref, stride, pad, g)
    dw = torch.zeros_like(w).to(DEV)
    db = torch.zeros(Co).to(DEV)
    ops.conv2d_backward_weight_acc(x.to(DEV), colT, dy.to(DEV), dw, db,
                                   stride, pad, g)
    close(dw, dw_ref, rtol=5e-4, atol=5e-4, what="conv wgrad")
    close(db, db_ref, rtol=5e-4, atol=5e-4, what="conv bgrad")


# ---------------------------------------------------------------------------
# Pooling / LRN
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("k,s,p", [(3, 2, 0), (3, 2, 1), (2, 2, 0)])
def test_maxpool(k, s, p):
    x = torch.randperm(2 * 4 * 13 * 13, generator=torch.Generator().manual_seed(2)
                       ).float().reshape(2, 4, 13, 13)
    y_ref, _ = ops.pool_max_forward(x, (k, k), (s, s), (p, p))
    y, mask = ops.pool_max_forward(x.to(DEV), (k, k), (s, s), (p, p))
    close(y, y_ref, what="maxpool fwd")
    dy = rnd(*y_ref.shape, seed=21)
    dx_ref = ops.pool_max_backward(dy, *_cpu_mask(x, k, s, p), x.shape,
                                   (k, k), (s, s), (p, p))
    dx = ops.pool_max_backward(dy.to(DEV), mask, x.shape, (k, k), (s, s), (p, p))
    close(dx, dx_ref, what="maxpool bwd")


def _cpu_mask(x, k, s, p):
    _, mask = ops.pool_max_forward(x, (k, k), (s, s), (p, p))
    return (mask,)


@pytest.mark.parametrize("k,s,p", [(3, 2, 1), (7, 1, 0), (3, 1, 1)])
def test_avepool(k, s, p):
    x = rnd(2, 5, 14, 14, seed=23)
    y_ref = ops.pool_ave_forward(x, (k, k), (s, s), (p, p))
    y = ops.pool_ave_forward(x.to(DEV), (k, k), (s, s), (p, p))
    close(y, y_ref, what="avepool fwd")
    dy = rnd(*y_ref.shape, seed=24)
    dx_ref = ops.pool_ave_backward(dy, x.shape, (k, k), (s, s), (p, p))
    dx = ops.pool_ave_backward(dy.to(DEV), x.shape, (k, k), (s, s), (p, p))
    close(dx, dx_ref, what="avepool bwd")


@pytest.mark.parametrize("C", [16,   # C%8==0: halo-register v8 path
                               13])  # odd C: LDS row-block fallback
def test_lrn(C):
    x = rnd(2, C, 7, 7, seed=25)
    y_ref, sc_ref = ops.lrn_forward(x, 5, 1e-4, 0.75)
    y, sc = ops.lrn_forward(x.to(DEV), 5, 1e-4, 0.75)
    close(y, y_ref, what="lrn fwd")
    # scale is an internal fwd->bwd cache; the v8 path doesn't store one
    if sc.numel():
        close(sc, sc_ref, what="lrn scale")
    dy = rnd(2, C, 7, 7, seed=26)
    dx_ref = ops.lrn_backward(x, y_ref, sc_ref, dy, 5, 1e-4, 0.75)
    dx = ops.lrn_backward(x.to(DEV), y, sc, dy.to(DEV), 5, 1e-4, 0.75)
    close(dx, dx_ref, what="lrn bwd")


# ---------------------------------------------------------------------------
# Softmax / losses / neurons
# ---------------------------------------------------------------------------

def test_softmax():
    for shape in [(6, 11), (2, 9, 5, 5)]:
        x = rnd(*shape, seed=27, scale=3.0)
        y_ref = ops.softmax_forward(x)
        y = ops.softmax_forward(x.to(DEV))
        close(y, y_ref, what=f"softmax {shape}")
        dy = rnd(*shape, seed=28)
        dx_ref = ops.softmax_backward(y_ref, dy)
        dx = ops.softmax_backward(y, dy.to(DEV))
        close(dx, dx_ref, what=f"softmax bwd {shape}")


def test_softmax_loss():
    x = rnd(33, 17, seed=29, scale=2.0)
    labels = torch.randint(0, 17, (33,)).float()
    loss_ref, prob_ref = ops.softmax_loss_forward(x, labels)
    loss, prob = ops.softmax_loss_forward(x.to(DEV), labels.to(DEV))
    close(loss.reshape(1), loss_ref.reshape(1), what="smloss")
    close(prob, prob_ref, what="smloss prob")
    dx_ref = ops.softmax_loss_backward(prob_ref, labels, 2.5)
    dx = ops.softmax_loss_backward(prob, labels.to(DEV), 2.5)
    close(dx, dx_ref, what="smloss bwd")


def test_neurons():
    x = rnd(3, 7, 9, 9, seed=31)
    dy = rnd(3, 7, 9, 9, seed=32)
    xg, dyg = x.to(DEV), dy.to(DEV)
    close(ops.relu_forward(xg, 0.1), ops.relu_forward(x, 0.1), what="relu")
    close(ops.relu_backward(xg, dyg, 0.1), ops.relu_backward(x, dy, 0.1),
          what="relu bwd")
    y = ops.sigmoid_forward(x)
    close(ops.sigmoid_forward(xg), y, what="sigmoid")
    close(ops.sigmoid_backward(y.to(DEV), dyg), ops.sigmoid_backward(y, dy),
          what="sigmoid bwd")
    yt = ops.tanh_forward(x)
    close(ops.tanh_forward(xg), yt, what="tanh")
    close(ops.tanh_backward(yt.to(DEV), dyg), ops.tanh_backward(yt, dy),
          what="tanh bwd")
    close(ops.bnll_forward(xg), ops.bnll_forward(x), what="bnll")
    close(ops.bnll_backward(xg, dyg), ops.bnll_backward(x, dy), what="bnll bwd")


def test_dropout():
    x = torch.ones(100000).to(DEV)
    y, mask = ops.dropout_forward(x, 0.4, seed=42, offset=0)
    keep = mask.float().mean().item()
    assert abs(keep - 0.6) < 0.02, keep
    scale = 1.0 / 0.6
    close(y, (mask.float() * scale).cpu(), what="dropout fwd")
    dy = rnd(100000, seed=33).to(DEV)
    dx = ops.dropout_backward(dy, mask, 0.4)
    close(dx, (dy * mask.float() * scale).cpu(), what="dropout bwd")
    # reproducible
    y2, mask2 = ops.dropout_forward(x, 0.4, seed=42, offset=0)
    assert torch.equal(mask, mask2)
    _, mask3 = ops.dropout_forward(x, 0.4, seed=42, offset=1)
    assert not torch.equal(mask, mask3)


def test_optimizer_updates():
    for name, fn, args in [
        ("sgd", ops.sgd_update, (0.1, 0.9, 0.001)),
        ("nesterov", ops.nesterov_update, (0.1, 0.9, 0.001)),
        ("adagrad", ops.adagrad_update, (0.1, 1e-8, 0.001)),
    ]:
        w0 = rnd(1000, seed=41)
        g = rnd(1000, seed=42)
        h0 = rnd(1000, seed=43).abs()
        w_ref, h_ref = w0.clone(), h0.clone()
        fn(w_ref, g, h_ref, *args)
        w_gpu, h_gpu = w0.clone().to(DEV), h0.clone().to(DEV)
        fn(w_gpu, g.to(DEV), h_gpu, *args)
        close(w_gpu, w_ref, what=f"{name} w")
        close(h_gpu, h_ref, what=f"{name} h")


# ---------------------------------------------------------------------------
# end-to-end on GPU
# ---------------------------------------------------------------------------

def test_cifar_quick_step_gpu():
    from poseidon_amd.core.net import Net, TRAIN
    from poseidon_amd.models import zoo
    pa.init(device="cuda", seed=3)
    try:
        net = Net(zoo.cifar10_quick(batch=32), phase=TRAIN)
        loss0 = net.forward()
        assert torch.isfinite(torch.tensor(loss0)), loss0
        net.zero_param_diffs()
        net.backward()
        for ps_ in net.learnable_params:
            assert torch.isfinite(ps_.blob.diff).all(), ps_.blob.name
            assert float(ps_.blob.diff.abs().sum()) > 0, ps_.blob.name
    finally:
        pa.init(device="cpu")


def test_alexnet_converges_gpu():
    """Tiny AlexNet-style training on separable synthetic data: loss drops."""
    from poseidon_amd.solver.solver import SGDSolver
    from poseidon_amd.proto import Message
    from poseidon_amd.models import zoo
    pa.init(device="cuda", seed=5)
    try:
        sp = Message("SolverParameter", base_lr=0.01, lr_policy="fixed",
                     momentum=0.9, weight_decay=0.0005, max_iter=30)
        sp.net_param = zoo.cifar10_quick(batch=64, num_classes=4)
        solver = SGDSolver(sp, verbose=False)
        # make data separable: class-prototype images
        data_layer = solver.net.layers[0]
        g = torch.Generator().manual_seed(9)
        protos = torch.randn(4, 3, 32, 32, generator=g)
        labels = torch.randint(0, 4, (64,), generator=g)
        imgs = protos[labels] + 0.2 * torch.randn(64, 3, 32, 32, generator=g)
        dev = pa.ctx().torch_device
        solver.net.blobs["data"].data = imgs.to(dev)
        solver.net.blobs["label"].data = labels.float().to(dev)
        data_layer._filled = True
        data_layer.refill = [False, False]
        first = float(solver.net.forward())
        solver.step(150)
        last = float(solver.net.forward_async().item())
        # cifar-quick's 1e-4 conv1 init learns slowly at first; a clear,
        # monotonic-ish decrease proves the GPU kernels train end-to-end
        # (convergence quality itself is covered by the CPU tests)
        assert last < first * 0.75, (first, last)
        assert all(torch.isfinite(p.blob.data).all()
                   for p in solver.net.learnable_params)
    finally:
        pa.init(device="cpu")


def test_solver_checkpoint_gpu(tmp_path):
    """Snapshot/restore with device tensors + test-net eval on GPU."""
    from poseidon_amd.solver.solver import SGDSolver
    from poseidon_amd.proto import Message
    from poseidon_amd.models import zoo
    pa.init(device="cuda", seed=11)
    try:
        sp = Message("SolverParameter", base_lr=0.01, lr_policy="fixed",
                     momentum=0.9, max_iter=100,
                     snapshot_prefix=str(tmp_path / "gpuck"))
        sp.net_param = zoo.cifar10_quick(batch=16, num_classes=4)
        solver = SGDSolver(sp, verbose=False)
        solver.step(3)
        path = solver.snapshot()
        w0 = next(l for l in solver.net.layers if l.name == "conv1").blobs[0]
        ref = w0.data.clone()

        solver2 = SGDSolver(sp, verbose=False)
        solver2.restore(str(tmp_path / "gpuck") + "_iter_3.solverstate.0.0")
        w1 = next(l for l in solver2.net.layers if l.name == "conv1").blobs[0]
        assert solver2.iter == 3
        assert torch.allclose(w1.data, ref)
    finally:
        pa.init(device="cpu")


def test_googlenet_step_gpu():
    from poseidon_amd.core.net import Net, TRAIN
    from poseidon_amd.models import zoo
    pa.init(device="cuda", seed=2, compute_dtype=torch.bfloat16)
    try:
        net = Net(zoo.googlenet(batch=4, num_classes=100), phase=TRAIN)
        loss = net.forward()
        assert torch.isfinite(torch.tensor(loss))
        net.zero_param_diffs()
        net.backward()
        bad = [p.blob.name for p in net.learnable_params
               if not torch.isfinite(p.blob.diff).all()]
        assert not bad, bad
    finally:
        pa.init(device="cpu", compute_dtype=torch.float32)


def test_full_solver_protocol_gpu(tmp_path):
    """The complete Solve() protocol on GPU in one run (solver.cpp:246-402):
    test_initialization eval, periodic TestAll + display, snapshot at
    snapshot interval and at max_iter, then restore + continue."""
    from poseidon_amd.solver.solver import SGDSolver
    from poseidon_amd.proto import Message
    from poseidon_amd.models import zoo
    pa.init(device="cuda", seed=23, compute_dtype=torch.bfloat16)
    try:
        sp = Message("SolverParameter", base_lr=0.005, lr_policy="step",
                     gamma=0.5, stepsize=8, momentum=0.9,
                     weight_decay=0.0005, max_iter=20, display=5,
                     test_interval=10, test_initialization=True,
                     snapshot=10, snapshot_prefix=str(tmp_path / "proto"))
        sp.test_iter.append(2)
        sp.net_param = zoo.cifar10_quick(batch=32, num_classes=4)
        solver = SGDSolver(sp, verbose=False)
        solver.solve()
        assert solver.iter == 20
        # snapshot at iter 10 (interval) and 20 (snapshot_after_train)
        for it in (10, 20):
            assert (tmp_path / f"proto_iter_{it}.caffemodel").exists()
            assert (tmp_path / f"proto_iter_{it}.solverstate.0.0").exists()
        # display path recorded scalar net outputs
        assert solver._net_outputs_rows, "display recorded no outputs"
        # resume from the mid snapshot and continue to max_iter
        solver2 = SGDSolver(sp, verbose=False)
        solver2.solve(resume_file=str(tmp_path / "proto_iter_10.solverstate.0.0"))
        assert solver2.iter == 20
        w0 = solver.net.learnable_params[0].blob.data
        assert torch.isfinite(w0).all()
    finally:
        pa.init(device="cpu")


def test_threshold_eltwise_contrastive_kernels():
    """Cold-layer CDNA4 kernels (reference eltwise_layer.cu:11,72,
    threshold_layer.cu:10, contrastive_loss_layer.cu:49)."""
    x = rnd(4, 8, 5, 5, seed=31)
    y = ops.threshold_forward(x.to(DEV), 0.1)
    close(y, (x > 0.1).float(), what="threshold")

    blobs = [rnd(3, 6, 4, 4, seed=s) for s in (32, 33, 34)]
    yg, mg = ops.eltwise_max([b.to(DEV) for b in blobs])
    yr, mr = ops.eltwise_max(blobs)
    close(yg, yr, what="eltwise max fwd")
    dy = rnd(3, 6, 4, 4, seed=35)
    for i in range(3):
        dg = ops.eltwise_max_backward(dy.to(DEV), mg, i)
        dr = ops.eltwise_max_backward(dy, mr, i)
        close(dg, dr, what=f"eltwise max bwd {i}")

    d2 = rnd(16, seed=36).abs()
    sim = (rnd(16, seed=37) > 0).float()
    tg = ops.contrastive_terms(d2.to(DEV), sim.to(DEV), 1.0, True)
    tr2 = ops.contrastive_terms(d2, sim, 1.0, True)
    close(tg, tr2, what="contrastive terms")


def test_concat_split_channels_fused():
    """4-way fused NHWC concat/split kernels (chan_concat4_k) vs torch.cat
    -- aligned (one launch per <=4 branches), >4 inputs (two slabs), and
    unaligned widths (per-input fallback)."""
    for widths in ([64, 128, 32, 32],           # inception-style, 1 launch
                   [16, 8, 24, 32, 40, 8],      # 6 inputs -> 2 slabs
                   [10, 6, 12]):                # unaligned -> fallback
        xs = [rnd(2, c, 5, 7, seed=100 + i) for i, c in enumerate(widths)]
        yg = ops.concat_channels([x.to(DEV) for x in xs])
        close(yg, torch.cat(xs, dim=1), what=f"concat {widths}")
        wide = rnd(2, sum(widths), 5, 7, seed=99)
        parts = ops.split_channels(wide.to(DEV), widths)
        off = 0
        for w, p in zip(widths, parts):
            close(p, wide.narrow(1, off, w), what=f"split {widths}@{off}")
            off += w


def test_deferred_wgrad_unpack_matches_eager():
    """Single-GPU deferred conv-wgrad unpack (one unpack_mt kernel at end
    of backward) must produce the same trained weights as the per-layer
    weight_from_khwc path."""
    from poseidon_amd.proto import Message, parse_text
    from poseidon_amd.solver.solver import SGDSolver

    def net_param():
        return parse_text("NetParameter", """
            name: "dconv"
            layers { name: "data" type: DUMMY_DATA top: "data" top: "label"
                     dummy_data_param { num: 8 channels: 8 height: 10
                         width: 10 num: 8 channels: 1 height: 1 width: 1
                         data_filler { type: "gaussian" std: 1.0 }
                         data_filler { type: "constant" } } }
            layers { name: "c1" type: CONVOLUTION bottom: "data" top: "c1"
                     convolution_param { num_output: 16 kernel_size: 3 pad: 1
                         weight_filler { type: "xavier" }
                         bias_filler { type: "constant" } } }
            layers { name: "r1" type: RELU bottom: "c1" top: "c1" }
            layers { name: "c2" type: CONVOLUTION bottom: "c1" top: "c2"
                     convolution_param { num_output: 8 kernel_size: 3 pad: 1
                         weight_filler { type: "xavier" } } }
            layers { name: "ip" type: INNER_PRODUCT bottom: "c2" top: "ip"
                     inner_product_param { num_output: 4
                         weight_filler { type: "xavier" } } }
            layers { name: "loss" type: SOFTMAX_LOSS bottom: "ip"
                     bottom: "label" top: "loss" }
        """)

    results = []
    for defer in (False, True):
        pa.init(device="cuda", seed=77)
        sp = Message("SolverParameter", base_lr=0.05, lr_policy="fixed",
                     momentum=0.9, weight_decay=0.001, max_iter=10,
                     display=0, snapshot=0)
        sp.net_param = net_param()
        s = SGDSolver(sp, verbose=False)
        for l in s.net.layers:
            if l.type_name == "CONVOLUTION":
                l.defer_unpack = defer
        s.step(7)  # >=3 settling iters + several batched-colsum iters
        torch.cuda.synchronize()
        results.append({i: ps.blob.data.clone().cpu()
                        for i, ps in enumerate(s.net.params)
                        if ps.owner == i})
    for i in results[0]:
        assert torch.allclose(results[0][i], results[1][i],
                              rtol=1e-5, atol=1e-6), f"param {i}"


def test_concat_fused_relu_bwd_matches_eager():
    """Inception-style branches (conv+relu -> concat): the ReLU backward
    fused into the concat scatter must train identically to the eager
    per-layer relu_bwd (PS_NO_CONCAT_MASK=1 baseline)."""
    import os
    from poseidon_amd.proto import Message, parse_text
    from poseidon_amd.solver.solver import SGDSolver

    def net_param():
        return parse_text("NetParameter", """
            name: "mini_inception"
            layers { name: "data" type: DUMMY_DATA top: "data" top: "label"
                     dummy_data_param { num: 8 channels: 16 height: 8
                         width: 8 num: 8 channels: 1 height: 1 width: 1
                         data_filler { type: "gaussian" std: 1.0 }
                         data_filler { type: "constant" } } }
            layers { name: "b1" type: CONVOLUTION bottom: "data" top: "b1"
                     convolution_param { num_output: 16 kernel_size: 1
                         weight_filler { type: "xavier" } } }
            layers { name: "rb1" type: RELU bottom: "b1" top: "b1" }
            layers { name: "b2" type: CONVOLUTION bottom: "data" top: "b2"
                     convolution_param { num_output: 24 kernel_size: 3 pad: 1
                         weight_filler { type: "xavier" } } }
            layers { name: "rb2" type: RELU bottom: "b2" top: "b2" }
            layers { name: "cc" type: CONCAT bottom: "b1" bottom: "b2"
                     top: "cc" }
            layers { name: "ip" type: INNER_PRODUCT bottom: "cc" top: "ip"
                     inner_product_param { num_output: 4
                         weight_filler { type: "xavier" } } }
            layers { name: "loss" type: SOFTMAX_LOSS bottom: "ip"
                     bottom: "label" top: "loss" }
        """)

    results = []
    for disable in ("1", ""):
        if disable:
            os.environ["PS_NO_CONCAT_MASK"] = disable
        else:
            os.environ.pop("PS_NO_CONCAT_MASK", None)
        pa.init(device="cuda", seed=99)
        sp = Message("SolverParameter", base_lr=0.05, lr_policy="fixed",
                     momentum=0.9, weight_decay=0.001, max_iter=10,
                     display=0, snapshot=0)
        sp.net_param = net_param()
        s = SGDSolver(sp, verbose=False)
        cc = next(l for l in s.net.layers if l.name == "cc")
        if not disable:
            assert getattr(cc, "_mask_bottoms", None) == {0, 1}
        s.step(5)
        torch.cuda.synchronize()
        results.append({i: ps.blob.data.clone().cpu()
                        for i, ps in enumerate(s.net.params)
                        if ps.owner == i})
    os.environ.pop("PS_NO_CONCAT_MASK", None)
    for i in results[0]:
        assert torch.allclose(results[0][i], results[1][i],
                              rtol=1e-5, atol=1e-6), f"param {i}"
