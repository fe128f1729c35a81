"""`caffe_main time` equivalent: per-layer forward/backward timing
(the reference shipped this disabled under #if 0, tools/caffe_main.cpp:
188-329; here it is a first-class tool built on LayerStats / hipEvents).

    python -m poseidon_amd.tools.time_net --model alexnet --iters 10
    python -m poseidon_amd.tools.time_net --net my_net.prototxt --iters 10
"""

from __future__ import annotations

import argparse

import torch


def main(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="", help="model zoo name")
    ap.add_argument("--net", default="", help="NetParameter prototxt path")
    ap.add_argument("--batch", type=int, default=0)
    ap.add_argument("--iters", type=int, default=10)
    ap.add_argument("--dtype", default="bf16", choices=["bf16", "fp32"])
    ap.add_argument("--stats", default="", help="also dump YAML here")
    args = ap.parse_args(argv)

    import poseidon_amd as pa
    from poseidon_amd.core.net import Net, TRAIN
    from poseidon_amd.models import zoo
    from poseidon_amd.proto import read_proto_text
    from poseidon_amd.utils.stats import LayerStats

    use_gpu = torch.cuda.is_available()
    cd = torch.bfloat16 if (args.dtype == "bf16" and use_gpu) else torch.float32
    pa.init(device="cuda" if use_gpu else "cpu", seed=1, compute_dtype=cd)

    if args.model:
        kw = {"batch": args.batch} if args.batch else {}
        net_param = zoo.build_net(args.model, **kw)
    elif args.net:
        net_param = read_proto_text(args.net, "NetParameter")
    else:
        ap.error("--model or --net required")

    net = Net(net_param, phase=TRAIN)
    net.forward()  # warmup + allocations
    net.zero_param_diffs()
    net.backward()
    if use_gpu:
        torch.cuda.synchronize()

    stats = LayerStats(net)
    with stats.timed():
        for _ in range(args.iters):
            net.forward()
            net.zero_param_diffs()
            net.backward()
    print(f"per-layer time over {args.iters} iterations "
          f"({'GPU events' if use_gpu else 'host clock'}):")
    print(stats.report(top=40))
    tf = sum(stats.fwd_ms.values()) / args.iters
    tb = sum(stats.bwd_ms.values()) / args.iters
    print(f"\naverage forward: {tf:.3f} ms  backward: {tb:.3f} ms  "
          f"total: {tf + tb:.3f} ms/iter")
    if args.stats:
        stats.dump(args.stats)


if __name__ == "__main__":
    main()
