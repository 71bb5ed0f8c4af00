#!/bin/bash
# Verification of KNOWN_ISSUES #1 (cfg5 intermittent NaN on the graph path;
# record_stream fix applied in trainer._run_fwd_bwd).  Run on ONE gpurun box:
#   gpurun --timeout 1800 -- 'bash tools/verify_cfg5.sh'
# Two independent 200-step graph-path soaks (startup dominates, so long runs are
# cheaper per step than many short ones) + one eager cross-check.
# Expect: skipped=0 and finite final_loss on every run.
set -x
for i in 1 2; do
  timeout 600 python bench.py --steps 200 --warmup 3 --approach cyclic --worker-fail 2 \
    --network ResNet50 --dataset ImageNetSynthetic --batch-size 32 --compile false --channels-last false \
    2>/dev/null | python -c "import json,sys; r=json.load(sys.stdin); print('graphs soak', r['ms_per_step'], 'ms/step loss', r['final_loss'], 'skipped', r['skipped_updates'])"
done
timeout 400 python bench.py --steps 10 --warmup 3 --approach cyclic --worker-fail 2 \
  --network ResNet50 --dataset ImageNetSynthetic --batch-size 32 --compile false --channels-last false --hip-graphs false \
  2>/dev/null | python -c "import json,sys; r=json.load(sys.stdin); print('eager run', r['ms_per_step'], 'ms/step loss', r['final_loss'], 'skipped', r['skipped_updates'])"
