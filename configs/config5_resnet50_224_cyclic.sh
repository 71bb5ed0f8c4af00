#!/bin/bash
# Config 5: ResNet-50 synthetic 224x224 bf16, cyclic r=5 (s=2), 8x MI355X (large-grad decode).
# hipGraph path ON: round-1's intermittent non-finite decodes were root-caused to a
# missing record_stream on the side-stream graph-input copies (trainer._run_fwd_bwd);
# verified in round 2 with 200-step graph soaks (skipped_updates=0, finite loss,
# 39.4 ms/step at N=1 — see RESULTS.md / gpurun_out/cfg5.log).
N=${1:-8}
python -m torch.distributed.run --nnodes=1 --nproc-per-node "$N" --master-addr 127.0.0.1 \
    bench.py --gpus "$N" --steps "${STEPS:-20}" --warmup "${WARMUP:-6}" \
    --approach cyclic --mode cyclic --worker-fail 2 --err-mode rev_grad \
    --network ResNet50 --dataset ImageNetSynthetic --batch-size 32 --dtype bf16 \
    --compile false
