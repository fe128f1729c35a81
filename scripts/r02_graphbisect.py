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
# warmup eagerly twice
for _ in range(2):
    s._graph_body()
torch.cuda.synchronize()

def try_capture(tag, fn):
    side = torch.cuda.Stream()
    side.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(side):
        fn()
    torch.cuda.current_stream().wait_stream(side)
    torch.cuda.synchronize()
    try:
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            fn()
        print(f"OK   {tag}", flush=True)
        return True
    except Exception as e:
        torch.cuda.synchronize()
        print(f"FAIL {tag}: {str(e).splitlines()[0]}", flush=True)
        return False

# per-layer forward
bad = []
for i, layer in enumerate(net.layers):
    ok = try_capture(f"fwd {i} {layer.name} ({layer.type_name})",
                     lambda i=i: net.layers[i].forward(net.bottoms[i], net.tops[i]))
    if not ok:
        bad.append(("fwd", i))
# seed diffs then per-layer backward
for (li, ti, w) in net._loss_tops:
    net.tops[li][ti].diff.fill_(w)
for i in range(len(net.layers) - 1, -1, -1):
    if not net.layer_need_bwd[i]:
        continue
    ok = try_capture(f"bwd {i} {net.layers[i].name}",
                     lambda i=i: net.layers[i].backward(net.tops[i], net.bottom_need_bwd[i], net.bottoms[i]))
    if not ok:
        bad.append(("bwd", i))
ok = try_capture("repack", net._maybe_mt_repack)
ok = try_capture("zero", net.zero_param_diffs)
ok = try_capture("mt_update", lambda: s._mt_update(0.0, lr_dev=s._lr_dev))
print("BAD:", bad)
