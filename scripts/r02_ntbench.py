import os, sys; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import time, torch
import poseidon_amd as pa
from poseidon_amd.ops import functional as F
ext = F._ext()
pa.init(device="cuda", seed=1)
def bench(M, N, K, iters=20):
    A = torch.randn(M, K, device="cuda").bfloat16().contiguous()
    B = torch.randn(N, K, device="cuda").bfloat16().contiguous()
    out = ext.gemm(A, B, M, N, K, True, True)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        out = ext.gemm(A, B, M, N, K, True, True)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    ref = A.float() @ B.float().t()
    err = (out - ref).abs().max().item() / (ref.abs().max().item() + 1e-9)
    print(f"NT {M}x{N}x{K}: {dt*1e3:.3f} ms {2*M*N*K/dt/1e12:.0f} TF/s relerr {err:.2e}", flush=True)
bench(4096, 4096, 4096)
bench(8192, 8192, 8192, 10)
