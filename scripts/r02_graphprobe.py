import os, sys; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import sys, traceback, torch
import poseidon_amd as pa
from poseidon_amd.models import zoo
from poseidon_amd.proto import Message
from poseidon_amd.solver.solver import SGDSolver

model = sys.argv[1] if len(sys.argv) > 1 else "googlenet"
pa.init(device="cuda", seed=1, compute_dtype=torch.bfloat16)
sp = Message("SolverParameter", base_lr=0.01, lr_policy="fixed", momentum=0.9,
             weight_decay=0.0005, max_iter=1 << 30, display=0, snapshot=0)
sp.net_param = zoo.build_net(model, batch=32)
s = SGDSolver(sp, use_sfb=False, verbose=True)
print("enable_graph ->", s.enable_graph())
try:
    s._capture_graph()
    print("capture OK")
    import time
    s._graph.replay(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(30): s._graph.replay()
    torch.cuda.synchronize()
    print("replay ms/step:", (time.perf_counter()-t0)/30*1e3)
except Exception:
    traceback.print_exc()
