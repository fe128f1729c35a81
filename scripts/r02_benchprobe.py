import os, sys; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import time, torch
import poseidon_amd as pa
from poseidon_amd.models import zoo
from poseidon_amd.proto import Message
from poseidon_amd.solver.solver import SGDSolver

pa.init(device="cuda", seed=1234, compute_dtype=torch.bfloat16)
sp = Message("SolverParameter", base_lr=0.01, lr_policy="fixed", momentum=0.9,
             weight_decay=0.0005, max_iter=1 << 30, display=0, snapshot=0)
sp.net_param = zoo.build_net("alexnet", batch=256)
s = SGDSolver(sp, use_sfb=False, verbose=False)
print("enable:", s.enable_graph())
s.step(10)  # warmup incl. capture
print("graph:", s._graph is not None)
torch.cuda.synchronize()

t0 = time.perf_counter(); s.step(50); torch.cuda.synchronize()
print("step(50) ms:", (time.perf_counter()-t0)/50*1e3)

t0 = time.perf_counter()
for _ in range(50): s._graph.replay()
torch.cuda.synchronize()
print("raw replay ms:", (time.perf_counter()-t0)/50*1e3)

t0 = time.perf_counter()
for _ in range(50):
    s._lr_dev.fill_(s.get_learning_rate())
    s._graph.replay()
torch.cuda.synchronize()
print("fill+replay ms:", (time.perf_counter()-t0)/50*1e3)

t0 = time.perf_counter(); s._step_graphed(50); torch.cuda.synchronize()
print("_step_graphed ms:", (time.perf_counter()-t0)/50*1e3)
