#!/bin/bash
# Config 3: ResNet-18 CIFAR-10, cyclic code r=3 (s=1), 8x MI355X (cyclic-decode HIP path).
N=${1:-8}
python -m torch.distributed.run --nnodes=1 --nproc-per-node "$N" --master-addr 127.0.0.1 \
    bench.py --gpus "$N" --steps "${STEPS:-40}" --warmup "${WARMUP:-10}" \
    --approach cyclic --mode cyclic --worker-fail 1 --err-mode rev_grad \
    --network ResNet18 --dataset Cifar10 --batch-size 128
