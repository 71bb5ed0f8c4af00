#!/bin/bash
# Config 5: ResNet-50 synthetic 224x224 bf16, cyclic r=5 (s=2), 8x MI355X (large-grad decode).
N=${1:-8}
python -m torch.distributed.run --nnodes=1 --nproc-per-node "$N" --master-addr 127.0.0.1 \
    bench.py --gpus "$N" --steps "${STEPS:-20}" --warmup "${WARMUP:-6}" \
    --approach cyclic --mode cyclic --worker-fail 2 --err-mode rev_grad \
    --network ResNet50 --dataset ImageNetSynthetic --batch-size 32 --dtype bf16 \
    --compile false --hip-graphs false
# eager path pinned: the graph-replay path for this config intermittently produced
# non-finite decoded gradients on some boxes (guarded + reported by the bench
# nan_guard/skipped_updates telemetry; standalone repros on fresh boxes pass) —
# under investigation, see RESULTS.md
