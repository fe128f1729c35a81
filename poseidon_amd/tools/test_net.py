"""`caffe_main test` equivalent: score a trained model on its TEST-phase
net (reference tools/caffe_main.cpp registered train/test/time/device_query
brew functions; test = load weights, run N forward batches, average the
output blobs).

    python -m poseidon_amd.tools.test_net --solver solver.prototxt \
        --weights snap_iter_1000.caffemodel --iters 50
"""

from __future__ import annotations

import argparse

import torch


def main(argv=None):
    ap = argparse.ArgumentParser(description="poseidon_amd model scorer")
    ap.add_argument("--solver", required=True, help="SolverParameter prototxt")
    ap.add_argument("--weights", required=True, help=".caffemodel to score")
    ap.add_argument("--iters", type=int, default=0,
                    help="override test_iter (0 = use solver's)")
    ap.add_argument("--dtype", default="bf16", choices=["bf16", "fp32"])
    ap.add_argument("--cpu", action="store_true")
    args = ap.parse_args(argv)

    import poseidon_amd as pa
    from poseidon_amd.proto import read_proto_text
    from poseidon_amd.solver.solver import get_solver

    solver_param = read_proto_text(args.solver, "SolverParameter")
    use_gpu = torch.cuda.is_available() and not args.cpu
    cd = torch.bfloat16 if (args.dtype == "bf16" and use_gpu) else torch.float32
    pa.init(device="cuda" if use_gpu else "cpu", seed=1, compute_dtype=cd)

    if not list(solver_param.test_iter):
        solver_param.test_iter.append(args.iters or 50)
    elif args.iters:
        solver_param.test_iter[0] = args.iters

    solver = get_solver(solver_param, use_sfb=False)
    solver.load_weights(args.weights)
    for ti in range(len(solver.test_nets)):
        res = solver.test(ti)
        line = "  ".join(f"{k}={v:.6f}" for k, v in sorted(res.items()))
        print(f"[poseidon] test net #{ti}: {line}", flush=True)


if __name__ == "__main__":
    main()
