#!/bin/bash
# All BASELINE configs at N=1 (the driver runs N>1), JSON lines to gpurun_out/.
# gpurun --timeout 1800 -- 'bash tools/bench_all.sh'
mkdir -p gpurun_out
run () {
  name=$1; shift
  timeout 500 python bench.py --steps 20 --warmup 5 "$@" \
    > "gpurun_out/bench_${name}.json" 2>"gpurun_out/bench_${name}.err"
  echo "== $name rc=$?"
  grep -o '{"metric".*}' "gpurun_out/bench_${name}.json" | \
    python -c "import json,sys; r=json.loads(sys.stdin.readline()); print(' ', r['ms_per_step'],'ms/step', r['value'], r['unit'], 'vs_baseline', r['vs_baseline'], 'loss', r['final_loss'], 'skipped', r['skipped_updates'])" || tail -3 "gpurun_out/bench_${name}.err"
}
run cfg2_rep_r18        # defaults: ResNet18 maj_vote r=3 s=1
run cfg3_cyc_r18        --approach cyclic --mode cyclic --worker-fail 1
run cfg4_geomed_vgg11   --approach baseline --mode geometric_median --network VGG11 --worker-fail 2
run cfg4_cyc_vgg11      --approach cyclic --mode cyclic --network VGG11 --worker-fail 2
run cfg5_cyc_r50        --approach cyclic --mode cyclic --worker-fail 2 --network ResNet50 --dataset ImageNetSynthetic --batch-size 32 --channels-last false
