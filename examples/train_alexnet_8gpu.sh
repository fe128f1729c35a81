#!/bin/bash
# AlexNet on all 8 MI355X of one node: one rank per GPU over RCCL,
# DWBP-overlapped gradient all-reduce + SFB on fc6/fc7.
# (replaces the reference's SSH fan-out + hostfile + PS flag assembly,
#  examples/imagenet/train_imagenet.sh)
exec python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
    --master-addr 127.0.0.1 \
    -m poseidon_amd.tools.train --solver examples/alexnet_solver.prototxt "$@"
