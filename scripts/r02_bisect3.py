import os, sys; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import poseidon_amd as pa
from poseidon_amd.models import zoo
from poseidon_amd.proto import Message
from poseidon_amd.solver.solver import SGDSolver

def make():
    pa.init(device="cuda", seed=1, compute_dtype=torch.bfloat16)
    sp = Message("SolverParameter", base_lr=0.01, lr_policy="fixed",
                 momentum=0.9, weight_decay=0.0005, max_iter=1 << 30,
                 display=0, snapshot=0)
    sp.net_param = zoo.build_net("googlenet", batch=32)
    s = SGDSolver(sp, use_sfb=False, verbose=False)
    assert s.enable_graph()
    s._lr_dev = torch.zeros(1, dtype=torch.float32, device="cuda")
    return s

def attempt(tag, nwarm, side_stream):
    s = make()
    if side_stream:
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(nwarm):
                s._graph_body()
        torch.cuda.current_stream().wait_stream(side)
    else:
        for _ in range(nwarm):
            s._graph_body()
    torch.cuda.synchronize()
    try:
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            s._graph_body()
        print(f"OK   {tag}", flush=True)
    except Exception as e:
        torch.cuda.synchronize()
        print(f"FAIL {tag}: {str(e).splitlines()[0]}", flush=True)

attempt("2 warm, side stream", 2, True)
attempt("3 warm, side stream", 3, True)
attempt("2 warm, default stream", 2, False)
attempt("3 warm, default stream", 3, False)
