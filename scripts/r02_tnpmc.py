import os, sys; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import poseidon_amd as pa
from poseidon_amd.ops import functional as F
ext = F._ext()
pa.init(device="cuda", seed=1)
M, N, K = 64, 576, 1605632
A = torch.randn(K, M, device="cuda").bfloat16().contiguous()
B = torch.randn(K, N, device="cuda").bfloat16().contiguous()
for _ in range(10):
    out = ext.gemm(A, B, M, N, K, False, False)
torch.cuda.synchronize()
