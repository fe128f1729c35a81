"""End-to-end training convergence evidence on synthetic separable data.

Builds CIFAR-10-quick (bf16 on GPU) with class-prototype images, trains
with the full SGD protocol (momentum + weight decay + step lr), and prints
the loss/accuracy trajectory. The committed output lives in
profiles/r01_convergence.md.

    python scripts/convergence_check.py [--iters 400] [--model cifar10_quick]
"""

import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=600)
    ap.add_argument("--batch", type=int, default=128)
    ap.add_argument("--classes", type=int, default=10)
    args = ap.parse_args()

    import poseidon_amd as pa
    from poseidon_amd.models import zoo
    from poseidon_amd.proto import Message
    from poseidon_amd.solver.solver import SGDSolver

    use_gpu = torch.cuda.is_available()
    pa.init(device="cuda" if use_gpu else "cpu", seed=7,
            compute_dtype=torch.bfloat16 if use_gpu else torch.float32)

    sp = Message("SolverParameter", base_lr=0.01, lr_policy="step",
                 gamma=0.5, stepsize=400, momentum=0.9, weight_decay=0.004,
                 max_iter=args.iters)
    sp.net_param = zoo.cifar10_quick(batch=args.batch,
                                     num_classes=args.classes)
    solver = SGDSolver(sp, verbose=False)

    # separable synthetic dataset: class prototypes + noise
    g = torch.Generator().manual_seed(11)
    protos = torch.randn(args.classes, 3, 32, 32, generator=g)
    labels = torch.randint(0, args.classes, (args.batch,), generator=g)
    imgs = protos[labels] + 0.25 * torch.randn(args.batch, 3, 32, 32,
                                               generator=g)
    dev = pa.ctx().torch_device
    net = solver.net
    net.blobs["data"].data = imgs.to(dev)
    net.blobs["label"].data = labels.float().to(dev)
    data_layer = net.layers[0]
    data_layer._filled = True
    data_layer.refill = [False, False]

    print(f"device={'cuda' if use_gpu else 'cpu'} "
          f"dtype={'bf16' if use_gpu else 'fp32'} batch={args.batch} "
          f"classes={args.classes}")
    for step in range(0, args.iters, 50):
        loss = float(net.forward())
        logits = net.blobs["ip2"].data.float()
        acc = (logits.argmax(1).cpu() == labels).float().mean().item()
        print(f"iter {solver.iter:4d}  loss {loss:.4f}  train_acc {acc:.3f}",
              flush=True)
        solver.step(50)
    loss = float(net.forward())
    logits = net.blobs["ip2"].data.float()
    acc = (logits.argmax(1).cpu() == labels).float().mean().item()
    print(f"iter {solver.iter:4d}  loss {loss:.4f}  train_acc {acc:.3f}")
    assert acc > 0.95, "did not converge"
    print("CONVERGED")


if __name__ == "__main__":
    main()
