import os, sys; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import traceback, torch
import poseidon_amd as pa
from poseidon_amd.models import zoo
from poseidon_amd.proto import Message
from poseidon_amd.solver.solver import SGDSolver

model = sys.argv[1] if len(sys.argv) > 1 else "googlenet"
pa.init(device="cuda", seed=1, compute_dtype=torch.bfloat16)
sp = Message("SolverParameter", base_lr=0.01, lr_policy="fixed", momentum=0.9,
             weight_decay=0.0005, max_iter=1 << 30, display=0, snapshot=0)
sp.net_param = zoo.build_net(model, batch=32)
s = SGDSolver(sp, use_sfb=False, verbose=False)
assert s.enable_graph()
net = s.net
import torch as T
s._lr_dev = T.zeros(1, dtype=T.float32, device="cuda")
for _ in range(3):
    s._graph_body()
torch.cuda.synchronize()

def try_capture(tag, fn):
    torch.cuda.synchronize()
    try:
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            fn()
        print(f"OK   {tag}", flush=True)
    except Exception as e:
        torch.cuda.synchronize()
        print(f"FAIL {tag}: {str(e).splitlines()[0]}", flush=True)

try_capture("zero_param_diffs", net.zero_param_diffs)
try_capture("forward_async", net.forward_async)
def bwd():
    net.backward()
try_capture("backward", bwd)
try_capture("mt_update", lambda: s._mt_update(0.0, lr_dev=s._lr_dev))
try_capture("body", s._graph_body)
