#!/bin/bash
# Config 1: LeNet/MNIST, repetition r=3, 1 PS + 3 workers, CPU/gloo, no adversary.
python -m torch.distributed.run --nnodes=1 --nproc-per-node 4 --master-addr 127.0.0.1 \
    -m draco_amd.train -- --topology ps --approach maj_vote --group-size 3 \
    --worker-fail 0 --err-mode none --network LeNet --dataset MNIST --device cpu \
    --batch-size 128 --max-steps "${STEPS:-50}" --eval-freq 25 --train-dir output/cfg1
