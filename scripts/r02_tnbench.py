import os, sys; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import time, torch
import poseidon_amd as pa
from poseidon_amd.ops import functional as F
ext = F._ext()
pa.init(device="cuda", seed=1)

def bench(M, N, K, iters=20):
    A = torch.randn(K, M, device="cuda").bfloat16().contiguous()  # K-major A [K][M]
    B = torch.randn(K, N, device="cuda").bfloat16().contiguous()  # K-major B [K][N]
    # C[M,N] = A^T @ B  -> a_klast=False, b_klast=False
    out = ext.gemm(A, B, M, N, K, False, False)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        out = ext.gemm(A, B, M, N, K, False, False)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    tf = 2.0 * M * N * K / dt / 1e12
    # correctness spot check
    ref = (A.float().t() @ B.float())
    err = (out - ref).abs().max().item() / (ref.abs().max().item() + 1e-9)
    print(f"TN {M}x{N}x{K}: {dt*1e3:.3f} ms  {tf:.0f} TF/s  relerr {err:.2e}",
          flush=True)

bench(64, 576, 1605632)     # VGG conv1_2 wgrad
bench(128, 1152, 401408)    # VGG conv2_2
bench(256, 2304, 100352)    # VGG conv3_x
bench(512, 4608, 25088)     # VGG conv5_x / r01 table shape
bench(128, 832, 25088)      # r01 table shape
bench(4096, 4096, 4096)
