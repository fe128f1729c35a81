#!/usr/bin/env python3
"""Flagship benchmark: whole-node images/sec for AlexNet (default) /
GoogLeNet / VGG-16 / CIFAR-quick on synthetic data with random-init weights
(BASELINE.json metric). One rank per GPU over RCCL; launch N>1 via
torch.distributed.run with --master-addr 127.0.0.1.

    python bench.py --gpus 1 --steps 20 --warmup 5
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 bench.py --gpus 8 --steps 20 --warmup 5
"""

import argparse
import json
import os
import time

import torch


DEFAULT_BATCH = {  # per-GPU batch, matching the reference's training configs
    "alexnet": 256,   # models/bvlc_alexnet/train_val.prototxt:10
    "caffenet": 256,  # models/bvlc_reference_caffenet/train_val.prototxt
    "googlenet": 32,  # models/bvlc_googlenet/train_test.prototxt:9
    "vgg16": 32,
    "cifar10_quick": 100,
    "lenet": 64,
}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=50)
    ap.add_argument("--warmup", type=int, default=10)
    ap.add_argument("--model", default="alexnet",
                    choices=list(DEFAULT_BATCH))
    ap.add_argument("--batch", type=int, default=0, help="per-GPU batch")
    ap.add_argument("--no-sfb", action="store_true")
    ap.add_argument("--implicit", action="store_true",
                    help="implicit-GEMM convolutions (gather in GEMM staging"
                         " instead of materialized im2col)")
    ap.add_argument("--no-graph", action="store_true",
                    help="disable hipGraph capture of the iteration")
    ap.add_argument("--dtype", default="bf16", choices=["bf16", "fp32"],
                    help="compute dtype (activations + GEMM inputs); "
                         "master weights and gradient accumulation stay fp32")
    args = ap.parse_args()

    import poseidon_amd as pa
    from poseidon_amd.models import zoo
    from poseidon_amd.proto import Message
    from poseidon_amd.solver.solver import SGDSolver
    from poseidon_amd.parallel import comm

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    assert world == args.gpus or args.gpus == 1 or world == 1, \
        f"WORLD_SIZE={world} vs --gpus={args.gpus}"
    n_gpus = max(world, 1)

    device = "cuda" if torch.cuda.is_available() else "cpu"
    cd = torch.bfloat16 if (args.dtype == "bf16" and device == "cuda") \
        else torch.float32
    pa.init(device=device, seed=1234, compute_dtype=cd)
    if args.implicit and device == "cuda":
        from poseidon_amd.ops import functional as F
        F.set_implicit_gemm(True)
    batch = args.batch or DEFAULT_BATCH[args.model]

    sp = Message("SolverParameter", base_lr=0.01, lr_policy="fixed",
                 momentum=0.9, weight_decay=0.0005, max_iter=1 << 30,
                 display=0, snapshot=0)
    sp.net_param = zoo.build_net(args.model, batch=batch)
    solver = SGDSolver(sp, use_sfb=not args.no_sfb, verbose=False)
    graphed = False
    if not args.no_graph and device == "cuda":
        # RCCL collectives are hipGraph-capturable: the multi-rank graph
        # replays DWBP all-reduces + SFB all-gathers in-graph; capture
        # success is agreed across ranks and falls back to eager everywhere
        # if any rank fails (solver._capture_graph). PS_GRAPH=0 disables.
        if os.environ.get("PS_GRAPH", "1") != "0":
            graphed = solver.enable_graph()

    def sync():
        if device == "cuda":
            torch.cuda.synchronize()
        comm.barrier()

    solver.step(args.warmup)
    # report the truth: capture may have fallen back to eager during warmup
    graphed = bool(solver._use_graph and solver._graph is not None)
    sync()
    t0 = time.perf_counter()
    solver.step(args.steps)
    sync()
    elapsed = time.perf_counter() - t0

    # MAX elapsed over ranks (staged through the device for nccl/RCCL)
    t = torch.tensor([elapsed], dtype=torch.float64)
    if solver.distributed:
        comm.allreduce_max(t)
    elapsed = float(t[0])

    if rank == 0:
        images = batch * n_gpus * args.steps
        value = images / elapsed
        out = {
            "metric": "images/sec (whole node)",
            "value": round(value, 2),
            "unit": "images/sec",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1e3, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": args.dtype if device == "cuda" else "fp32",
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": batch * n_gpus,
                "input": {"alexnet": "3x227x227", "caffenet": "3x227x227",
                          "googlenet": "3x224x224",
                          "vgg16": "3x224x224", "cifar10_quick": "3x32x32",
                          "lenet": "1x28x28"}[args.model],
                "parallelism": f"dp{n_gpus}",
                "sfb": not args.no_sfb and n_gpus > 1,
                "hipgraph": graphed,
            },
        }
        print(json.dumps(out), flush=True)


if __name__ == "__main__":
    main()
