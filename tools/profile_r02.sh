#!/bin/bash
# Round-2 rocprofv3 evidence: kernel stats for cfg2/cfg3 + PMC counters.
# Trace dbs are deleted after stats extraction (merge-back size limit);
# counters run SEPARATELY from trace domains per pool policy.
#   gpurun --timeout 1500 -- 'bash tools/profile_r02.sh'
set -x
REPO=$(cd "$(dirname "$0")/.." && pwd)
cd "$REPO"
export TMPDIR=/tmp
mkdir -p gpurun_out/p2
timeout 420 rocprofv3 --output-format csv --kernel-trace --stats -d gpurun_out/p2 -o steady_r02 -- \
  python bench.py --steps 40 --warmup 10 > gpurun_out/prof_a.log 2>&1
echo "P1=$?"
rm -f gpurun_out/p2/*results.db gpurun_out/p2/*.db gpurun_out/p2/*_kernel_trace.csv
timeout 420 rocprofv3 --output-format csv --kernel-trace --stats -d gpurun_out/p2 -o cyclic_r02 -- \
  python bench.py --steps 40 --warmup 10 --approach cyclic --worker-fail 1 \
  > gpurun_out/prof_b.log 2>&1
echo "P2=$?"
rm -f gpurun_out/p2/*results.db gpurun_out/p2/*.db gpurun_out/p2/*_kernel_trace.csv
timeout 420 rocprofv3 --output-format csv --pmc GRBM_GUI_ACTIVE,SQ_BUSY_CYCLES,SQ_VALU_MFMA_BUSY_CYCLES,SQ_WAVE_CYCLES \
  -d gpurun_out/p2 -o pmc_r02 -- python bench.py --steps 10 --warmup 4 \
  > gpurun_out/prof_c.log 2>&1
echo "P3=$?"
CC=$(ls gpurun_out/p2/*counter_collection.csv 2>/dev/null | head -1)
[ -n "$CC" ] && python tools/pmc_summary.py "$CC" gpurun_out/p2/pmc_kernel_summary_r02.csv
rm -f gpurun_out/p2/*counter_collection.csv gpurun_out/p2/*.db
ls -la gpurun_out/p2
du -sh gpurun_out
