#!/bin/bash
# Config 4: VGG-11 CIFAR-10 — geometric-median baseline vs Draco cyclic r=5 (s=2), 8x MI355X.
N=${1:-8}
echo "== geometric-median baseline =="
python -m torch.distributed.run --nnodes=1 --nproc-per-node "$N" --master-addr 127.0.0.1 \
    bench.py --gpus "$N" --steps "${STEPS:-30}" --warmup "${WARMUP:-8}" \
    --approach baseline --mode geometric_median --worker-fail 2 --err-mode rev_grad \
    --network VGG11 --dataset Cifar10 --batch-size 128
echo "== Draco cyclic r=5 s=2 =="
python -m torch.distributed.run --nnodes=1 --nproc-per-node "$N" --master-addr 127.0.0.1 \
    bench.py --gpus "$N" --steps "${STEPS:-30}" --warmup "${WARMUP:-8}" \
    --approach cyclic --mode cyclic --worker-fail 2 --err-mode rev_grad \
    --network VGG11 --dataset Cifar10 --batch-size 128
