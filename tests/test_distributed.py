"""Multi-process data parallelism over gloo (CPU stand-in for RCCL; the
collective pattern is identical): DWBP all-reduce and SFB factor all-gather
must reproduce the single-process summed-gradient result exactly."""

import os

import pytest
import torch
import torch.multiprocessing as mp

WORLD = 2
PORT = "29771"


def _net_param(batch, dim=12, classes=4):
    from poseidon_amd.proto import parse_text
    return parse_text("NetParameter", f"""
        name: "toy"
        layers {{ name: "data" type: MEMORY_DATA top: "data" top: "label"
                 memory_data_param {{ batch_size: {batch} channels: {dim}
                                      height: 1 width: 1 }} }}
        layers {{ name: "ip1" type: INNER_PRODUCT bottom: "data" top: "ip1"
                 inner_product_param {{ num_output: 16
                     weight_filler {{ type: "xavier" }} }} }}
        layers {{ name: "relu1" type: RELU bottom: "ip1" top: "ip1" }}
        layers {{ name: "ip2" type: INNER_PRODUCT bottom: "ip1" top: "ip2"
                 inner_product_param {{ num_output: {classes}
                     weight_filler {{ type: "xavier" }} }} }}
        layers {{ name: "loss" type: SOFTMAX_LOSS bottom: "ip2" bottom: "label"
                 top: "loss" }}
    """)


def _dataset(n=64, dim=12, classes=4, seed=5):
    g = torch.Generator().manual_seed(seed)
    protos = torch.randn(classes, dim, generator=g) * 2.0
    labels = torch.randint(0, classes, (n,), generator=g)
    data = protos[labels] + 0.2 * torch.randn(n, dim, generator=g)
    return data.view(n, dim, 1, 1), labels.float()


def _solver_param():
    from poseidon_amd.proto import Message
    sp = Message("SolverParameter")
    sp.base_lr = 0.05
    sp.lr_policy = "fixed"
    sp.momentum = 0.9
    sp.weight_decay = 0.001
    sp.max_iter = 10
    return sp


def _reference_run(iters, batch, use_sfb_shapes=False):
    """Single process emulating 2 ranks: per-rank grads summed, wd doubled."""
    import poseidon_amd as pa
    from poseidon_amd.solver.solver import SGDSolver
    pa.init(device="cpu", rank=0, world_size=1, seed=42)
    sp = _solver_param()
    sp.net_param = _net_param(batch)
    solver = SGDSolver(sp, verbose=False)
    data, labels = _dataset()
    shards = [(data[r::WORLD], labels[r::WORLD]) for r in range(WORLD)]
    net = solver.net
    for it in range(iters):
        grads = None
        loss = 0.0
        for r in range(WORLD):
            d, l = shards[r]
            idx = torch.arange(it * batch, (it + 1) * batch) % d.shape[0]
            net.layers[0].add_data(d, l)
            net.layers[0]._pos = int(it * batch % d.shape[0])
            net.zero_param_diffs()
            net.forward()
            net.backward()
            g = [ps.blob.diff.clone() for i, ps in enumerate(net.params)
                 if ps.owner == i]
            grads = g if grads is None else [a + b for a, b in zip(grads, g)]
        own = [i for i, ps in enumerate(net.params) if ps.owner == i]
        for i, g in zip(own, grads):
            net.params[i].blob.diff.copy_(g)
        solver.distributed = True  # apply the W*decay convention
        pa.ctx().world_size = WORLD
        for i in own:
            solver._apply_update(i, net.params[i], solver.get_learning_rate())
        pa.ctx().world_size = 1
        solver.distributed = False
    return {i: net.params[i].blob.data.clone() for i in own}


def _worker(rank, iters, batch, use_sfb, out_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = PORT
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(WORLD)
    import poseidon_amd as pa
    from poseidon_amd.solver.solver import SGDSolver
    from poseidon_amd.solver import sfb as sfb_mod
    pa.init(device="cpu", rank=rank, world_size=WORLD, seed=42)
    sp = _solver_param()
    sp.net_param = _net_param(batch)
    if use_sfb:
        # force factors even though the volume test says "not worth it" at
        # toy sizes -- we are testing correctness, not the heuristic
        sfb_mod.sfb_worthwhile = lambda *a, **k: True
    solver = SGDSolver(sp, use_sfb=use_sfb, verbose=False)
    data, labels = _dataset()
    solver.net.layers[0].add_data(data[rank::WORLD], labels[rank::WORLD])
    solver.step(iters)
    own = [i for i, ps in enumerate(solver.net.params) if ps.owner == i]
    torch.save({i: solver.net.params[i].blob.data for i in own},
               os.path.join(out_dir, f"rank{rank}.pt"))
    import torch.distributed as dist
    dist.destroy_process_group()


@pytest.mark.parametrize("use_sfb", [False, True])
def test_two_rank_training_matches_reference(tmp_path, use_sfb):
    global PORT
    PORT = str(29770 + (7 if use_sfb else 0))
    iters, batch = 6, 8
    ref = _reference_run(iters, batch)
    mp.start_processes(_worker, args=(iters, batch, use_sfb, str(tmp_path)),
                       nprocs=WORLD, join=True, start_method="spawn")
    r0 = torch.load(tmp_path / "rank0.pt")
    r1 = torch.load(tmp_path / "rank1.pt")
    for i in ref:
        assert torch.allclose(r0[i], r1[i], atol=1e-6), f"ranks diverged p{i}"
        assert torch.allclose(r0[i], ref[i], atol=1e-5), \
            f"param {i}: distributed != reference (max err " \
            f"{(r0[i]-ref[i]).abs().max():.3g})"


def _ckpt_worker(rank, out_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = PORT
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(WORLD)
    import poseidon_amd as pa
    from poseidon_amd.solver.solver import SGDSolver
    pa.init(device="cpu", rank=rank, world_size=WORLD, seed=42)
    sp = _solver_param()
    sp.snapshot_prefix = os.path.join(out_dir, "ck")
    sp.net_param = _net_param(8)
    solver = SGDSolver(sp, use_sfb=False, verbose=False)
    data, labels = _dataset()
    solver.net.layers[0].add_data(data[rank::WORLD], labels[rank::WORLD])
    solver.step(3)
    if rank == 0:
        solver.snapshot()          # rank 0 writes caffemodel + its state
    import torch.distributed as dist
    dist.barrier()
    # BOTH ranks resume; rank 1 has no own solverstate file and must fall
    # back to rank 0's (solver.cpp:670-696 thread-0 fallback semantics)
    solver2 = SGDSolver(sp, use_sfb=False, verbose=False)
    solver2.net.layers[0].add_data(data[rank::WORLD], labels[rank::WORLD])
    solver2.restore(os.path.join(out_dir, "ck_iter_3.solverstate.0.0"))
    assert solver2.iter == 3
    own = [i for i, ps in enumerate(solver2.net.params) if ps.owner == i]
    ref = {i: solver.net.params[i].blob.data for i in own}
    for i in own:
        assert torch.allclose(solver2.net.params[i].blob.data, ref[i],
                              atol=1e-6), f"restore mismatch p{i} rank{rank}"
    solver2.step(2)  # resumes training without error
    dist.destroy_process_group()


def test_two_rank_snapshot_restore(tmp_path):
    global PORT
    PORT = str(29791)
    mp.start_processes(_ckpt_worker, args=(str(tmp_path),),
                       nprocs=WORLD, join=True, start_method="spawn")


# ---------------------------------------------------------------------------
# 8-rank bucketed DWBP + SFB interleave: a deeper net
# (conv stack + two SFB'd IP layers), tiny buckets so several all-reduces
# interleave with the SFB all-gathers, world_size 8.
# ---------------------------------------------------------------------------

W8 = 8


def _deep_net_param(batch):
    from poseidon_amd.proto import parse_text
    return parse_text("NetParameter", f"""
        name: "deep"
        layers {{ name: "data" type: MEMORY_DATA top: "data" top: "label"
                 memory_data_param {{ batch_size: {batch} channels: 3
                                      height: 12 width: 12 }} }}
        layers {{ name: "conv1" type: CONVOLUTION bottom: "data" top: "conv1"
                 convolution_param {{ num_output: 8 kernel_size: 3 pad: 1
                     weight_filler {{ type: "xavier" }} }} }}
        layers {{ name: "relu1" type: RELU bottom: "conv1" top: "conv1" }}
        layers {{ name: "pool1" type: POOLING bottom: "conv1" top: "pool1"
                 pooling_param {{ pool: MAX kernel_size: 2 stride: 2 }} }}
        layers {{ name: "conv2" type: CONVOLUTION bottom: "pool1" top: "conv2"
                 convolution_param {{ num_output: 8 kernel_size: 3 pad: 1
                     weight_filler {{ type: "xavier" }} }} }}
        layers {{ name: "relu2" type: RELU bottom: "conv2" top: "conv2" }}
        layers {{ name: "ip1" type: INNER_PRODUCT bottom: "conv2" top: "ip1"
                 inner_product_param {{ num_output: 24
                     weight_filler {{ type: "xavier" }} }} }}
        layers {{ name: "relu3" type: RELU bottom: "ip1" top: "ip1" }}
        layers {{ name: "ip2" type: INNER_PRODUCT bottom: "ip1" top: "ip2"
                 inner_product_param {{ num_output: 4
                     weight_filler {{ type: "xavier" }} }} }}
        layers {{ name: "loss" type: SOFTMAX_LOSS bottom: "ip2" bottom: "label"
                 top: "loss" }}
    """)


def _deep_dataset(n=128, seed=9):
    g = torch.Generator().manual_seed(seed)
    labels = torch.randint(0, 4, (n,), generator=g)
    protos = torch.randn(4, 3, 12, 12, generator=g)
    data = protos[labels] + 0.2 * torch.randn(n, 3, 12, 12, generator=g)
    return data, labels.float()


def _deep_reference_run(iters, batch, world):
    """Single process emulating `world` ranks: per-rank grads summed."""
    import poseidon_amd as pa
    from poseidon_amd.solver.solver import SGDSolver
    pa.init(device="cpu", rank=0, world_size=1, seed=42)
    sp = _solver_param()
    sp.net_param = _deep_net_param(batch)
    solver = SGDSolver(sp, verbose=False)
    data, labels = _deep_dataset()
    shards = [(data[r::world], labels[r::world]) for r in range(world)]
    net = solver.net
    own = [i for i, ps in enumerate(net.params) if ps.owner == i]
    for it in range(iters):
        grads = None
        for r in range(world):
            d, l = shards[r]
            net.layers[0].add_data(d, l)
            net.layers[0]._pos = int(it * batch % d.shape[0])
            net.zero_param_diffs()
            net.forward()
            net.backward()
            g = [net.params[i].blob.diff.clone() for i in own]
            grads = g if grads is None else [a + b for a, b in zip(grads, g)]
        for i, g in zip(own, grads):
            net.params[i].blob.diff.copy_(g)
        solver.distributed = True
        pa.ctx().world_size = world
        for i in own:
            solver._apply_update(i, net.params[i], solver.get_learning_rate())
        pa.ctx().world_size = 1
        solver.distributed = False
    return {i: net.params[i].blob.data.clone() for i in own}


def _deep_worker(rank, iters, batch, out_dir, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = port
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(W8)
    os.environ["PS_BUCKET_BYTES"] = "1024"  # force several buckets/iter
    import poseidon_amd as pa
    from poseidon_amd.solver.solver import SGDSolver
    from poseidon_amd.solver import sfb as sfb_mod
    pa.init(device="cpu", rank=rank, world_size=W8, seed=42)
    sp = _solver_param()
    sp.net_param = _deep_net_param(batch)
    sfb_mod.sfb_worthwhile = lambda *a, **k: True  # force factors on both IPs
    solver = SGDSolver(sp, use_sfb=True, verbose=False)
    assert solver.sfb is not None and len(solver.sfb.layers) == 2
    data, labels = _deep_dataset()
    solver.net.layers[0].add_data(data[rank::W8], labels[rank::W8])
    solver.step(iters)
    if rank == 0:
        own = [i for i, ps in enumerate(solver.net.params) if ps.owner == i]
        torch.save({i: solver.net.params[i].blob.data for i in own},
                   os.path.join(out_dir, "rank0.pt"))
    import torch.distributed as dist
    dist.barrier()
    dist.destroy_process_group()


def test_eight_rank_bucketed_dwbp_sfb(tmp_path):
    iters, batch = 4, 4
    ref = _deep_reference_run(iters, batch, W8)
    mp.start_processes(_deep_worker,
                       args=(iters, batch, str(tmp_path), "29815"),
                       nprocs=W8, join=True, start_method="spawn")
    r0 = torch.load(tmp_path / "rank0.pt")
    for i in ref:
        assert torch.allclose(r0[i], ref[i], atol=1e-5), \
            f"param {i}: 8-rank distributed != reference (max err " \
            f"{(r0[i]-ref[i]).abs().max():.3g})"
