#!/bin/bash
# Config 5: ResNet-50 synthetic 224x224 bf16, cyclic r=5 (s=2), 8x MI355X (large-grad decode).
# hipGraph path ON; conv layout pinned to NCHW for this model: long-horizon soaks
# isolated a bf16+channels_last+hipGraph interaction on the R50-224 shapes where a
# captured conv kernel starts producing garbage after ~50 replays (eager NHWC is
# clean, graphs+NCHW is clean, graphs+fp32-NHWC is clean; ResNet-18 NHWC graphs are
# unaffected over 1000+ steps).  See KNOWN_ISSUES.md #cfg5-nhwc and
# tools/diag_cfg5_isolate.py for the isolation matrix.
N=${1:-8}
python -m torch.distributed.run --nnodes=1 --nproc-per-node "$N" --master-addr 127.0.0.1 \
    bench.py --gpus "$N" --steps "${STEPS:-20}" --warmup "${WARMUP:-6}" \
    --approach cyclic --mode cyclic --worker-fail 2 --err-mode rev_grad \
    --network ResNet50 --dataset ImageNetSynthetic --batch-size 32 --dtype bf16 \
    --compile false --channels-last false
