import os, sys; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import time, torch
import poseidon_amd as pa
from poseidon_amd.ops import functional as F
ext = F._ext()
pa.init(device="cuda", seed=1)
def bench(M, N, K, iters=30):
    A = torch.randn(K, M, device="cuda").bfloat16().contiguous()
    B = torch.randn(K, N, device="cuda").bfloat16().contiguous()
    out = ext.gemm(A, B, M, N, K, False, False)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        out = ext.gemm(A, B, M, N, K, False, False)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    ref = A.float().t() @ B.float()
    err = (out - ref).abs().max().item() / (ref.abs().max().item() + 1e-9)
    print(f"TN {M}x{N}x{K}: {dt*1e3:.3f} ms {2*M*N*K/dt/1e12:.0f} TF/s relerr {err:.2e}", flush=True)
bench(4096, 9216, 256)   # AlexNet fc6 wgrad
bench(4096, 4096, 256)   # fc7
bench(4096, 25088, 32)   # VGG fc6 wgrad (b32)
bench(4096, 4096, 4096)  # sanity: big-K unaffected
