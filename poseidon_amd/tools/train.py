"""`caffe_main train` equivalent (reference tools/caffe_main.cpp:100-186).

Single GPU:
    python -m poseidon_amd.tools.train --solver solver.prototxt

All 8 GPUs of a node (one rank per GPU over RCCL; DWBP overlap + SFB on):
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \\
        --master-addr 127.0.0.1 -m poseidon_amd.tools.train \\
        --solver solver.prototxt

Flags mirror the reference CLI: --snapshot (resume), --weights (finetune),
--net_outputs (metrics CSV), --svb (sufficient-factor broadcast). The
reference's --gpu id list and PS flags (hostfile/num_clients/staleness...)
collapse into torchrun's rank/world env: one process drives one GPU and
consistency is BSP (staleness=0).
"""

from __future__ import annotations

import argparse
import os

import torch


def main(argv=None):
    ap = argparse.ArgumentParser(description="poseidon_amd trainer")
    ap.add_argument("--solver", required=True, help="SolverParameter prototxt")
    ap.add_argument("--snapshot", default="", help="resume from .solverstate")
    ap.add_argument("--weights", default="", help="finetune from .caffemodel")
    ap.add_argument("--net_outputs", default="", help="metrics CSV path prefix")
    ap.add_argument("--svb", action="store_true", default=True,
                    help="sufficient-factor broadcast for FC layers (default on)")
    ap.add_argument("--no-svb", dest="svb", action="store_false")
    ap.add_argument("--dtype", default="bf16", choices=["bf16", "fp32"])
    ap.add_argument("--stats", default="",
                    help="dump per-layer fwd/bwd timing YAML here (the "
                         "Bösen caffe_stats.yaml analogue)")
    ap.add_argument("--cpu", action="store_true", help="force CPU mode")
    args = ap.parse_args(argv)

    import poseidon_amd as pa
    from poseidon_amd.proto import read_proto_text
    from poseidon_amd.solver.solver import get_solver

    solver_param = read_proto_text(args.solver, "SolverParameter")
    use_gpu = torch.cuda.is_available() and not args.cpu \
        and solver_param.enum_name("solver_mode") == "GPU"
    cd = torch.bfloat16 if (args.dtype == "bf16" and use_gpu) else torch.float32
    seed = int(solver_param.random_seed)
    pa.init(device="cuda" if use_gpu else "cpu",
            seed=seed if seed >= 0 else 1,
            compute_dtype=cd)

    solver = get_solver(solver_param, use_sfb=args.svb)
    if args.weights:
        solver.load_weights(args.weights)
    stats = None
    if args.stats:
        from poseidon_amd.utils.stats import LayerStats
        stats = LayerStats(solver.net, roctx=False)
        ctx_mgr = stats.timed()
        ctx_mgr.__enter__()
    solver.solve(resume_file=args.snapshot or None)
    if stats is not None:
        ctx_mgr.__exit__(None, None, None)
        if pa.ctx().is_root():
            stats.dump(args.stats)
            print(stats.report())
    if args.net_outputs:
        solver.write_net_outputs(args.net_outputs)
    if pa.ctx().is_root():
        print("[poseidon] optimization done", flush=True)


if __name__ == "__main__":
    main()
