"""GEMM kernel microbench (within-probe repeats, random data -- guide §5.4
rules 24/25). Usage:

    python -m poseidon_amd.tools.gemm_bench [--size 4096] [--dtype bf16]
        [--layout nt|nn|tn] [--reps 20]
"""

import argparse
import time

import torch


def main(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("--size", type=int, default=4096)
    ap.add_argument("--mnk", default="", help="M,N,K override")
    ap.add_argument("--dtype", default="bf16", choices=["bf16", "fp32"])
    ap.add_argument("--layout", default="nt", choices=["nt", "nn", "tn"])
    ap.add_argument("--reps", type=int, default=20)
    args = ap.parse_args(argv)

    from poseidon_amd.ops._backend import load
    ext = load()
    if args.mnk:
        M, N, K = (int(x) for x in args.mnk.split(","))
    else:
        M = N = K = args.size
    dt = torch.bfloat16 if args.dtype == "bf16" else torch.float32
    ak, bk = {"nt": (True, True), "nn": (True, False),
              "tn": (False, False)}[args.layout]

    g = torch.Generator().manual_seed(0)
    opA = torch.randn(M, K, generator=g)
    opB = torch.randn(K, N, generator=g)
    A = (opA if ak else opA.t()).contiguous().to("cuda", dt)
    B = (opB.t() if bk else opB).contiguous().to("cuda", dt)

    # correctness spot-check FIRST: the CPU reference matmul takes seconds,
    # and any GPU-idle gap right before the timed region lets DVFS drop the
    # clocks for the whole measurement (guide rule 24; measured 50x once)
    out = ext.gemm(A, B, M, N, K, ak, bk)
    ref = (opA.to(dt).float() @ opB.to(dt).float())
    err = (out.cpu() - ref).abs().max().item()
    # clock ramp: keep the GPU busy ~100 ms right before timing
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    while time.perf_counter() - t0 < 0.1:
        ext.gemm(A, B, M, N, K, ak, bk)
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.reps):
        ext.gemm(A, B, M, N, K, ak, bk)
    torch.cuda.synchronize()
    dt_s = (time.perf_counter() - t0) / args.reps
    tf = 2.0 * M * N * K / dt_s / 1e12
    print(f"{args.layout} {args.dtype} {M}x{N}x{K}: {tf:.1f} TF/s "
          f"({dt_s * 1e3:.3f} ms), max_err={err:.3g}")


if __name__ == "__main__":
    main()
