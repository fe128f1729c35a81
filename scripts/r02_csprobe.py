import os, sys; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import poseidon_amd as pa
from poseidon_amd.models import zoo
from poseidon_amd.proto import Message
from poseidon_amd.solver.solver import SGDSolver

pa.init(device="cuda", seed=1, compute_dtype=torch.bfloat16)
sp = Message("SolverParameter", base_lr=0.01, lr_policy="fixed", momentum=0.9,
             weight_decay=0.0005, max_iter=1 << 30, display=0, snapshot=0)
sp.net_param = zoo.build_net("googlenet", batch=32)
s = SGDSolver(sp, use_sfb=False, verbose=False)
s.step(6)
torch.cuda.synchronize()
convs = [l for l in s.net.layers if l.type_name == "CONVOLUTION"]
stable = sum(1 for l in convs if l._last_dy_ptr is not None)
mt = s.net._colsum_mt
print(f"convs={len(convs)} with_cl_dy={stable} "
      f"batched={len(mt[1]) if mt else 0}")
