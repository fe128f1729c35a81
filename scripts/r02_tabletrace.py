import os, sys; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import poseidon_amd as pa
from poseidon_amd.models import zoo
from poseidon_amd.proto import Message
from poseidon_amd.solver.solver import SGDSolver
from poseidon_amd.ops import functional as ops

# trace every host->device table upload
orig_zp = ops.zero_mt_prepare
orig_sp = ops.sgd_mt_prepare
orig_rp = ops.repack_mt_prepare
it = [0]
ops.zero_mt_prepare = lambda *a, **k: (print(f"iter{it[0]}: zero_mt_prepare", flush=True), orig_zp(*a, **k))[1]
ops.sgd_mt_prepare = lambda *a, **k: (print(f"iter{it[0]}: sgd_mt_prepare", flush=True), orig_sp(*a, **k))[1]
ops.repack_mt_prepare = lambda *a, **k: (print(f"iter{it[0]}: repack_mt_prepare", flush=True), orig_rp(*a, **k))[1]
# also patch the module-level references solver/net captured
import poseidon_amd.core.net as netmod
import poseidon_amd.solver.solver as solmod

pa.init(device="cuda", seed=1, compute_dtype=torch.bfloat16)
sp = Message("SolverParameter", base_lr=0.01, lr_policy="fixed", momentum=0.9,
             weight_decay=0.0005, max_iter=1 << 30, display=0, snapshot=0)
sp.net_param = zoo.build_net("googlenet", batch=32)
s = SGDSolver(sp, use_sfb=False, verbose=False)
assert s.enable_graph()
s._lr_dev = torch.zeros(1, dtype=torch.float32, device="cuda")
for i in range(5):
    it[0] = i
    print(f"--- iter {i}", flush=True)
    s._graph_body()
torch.cuda.synchronize()
