#!/bin/bash
# Config 2: ResNet-18 CIFAR-10, repetition r=3, s=1 sign-flip adversary, 8x MI355X.
N=${1:-8}
python -m torch.distributed.run --nnodes=1 --nproc-per-node "$N" --master-addr 127.0.0.1 \
    bench.py --gpus "$N" --steps "${STEPS:-40}" --warmup "${WARMUP:-10}" \
    --approach maj_vote --group-size 3 --worker-fail 1 --err-mode rev_grad \
    --network ResNet18 --dataset Cifar10 --batch-size 128
