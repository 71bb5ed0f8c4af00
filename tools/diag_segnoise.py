"""Measure HONEST per-segment replica noise for the segment-granular vote.

Runs r=3 replicated ResNet-18 fwd/bwd on one GPU (bf16 autocast, hipGraphs, the
bench configuration) with NO adversary and reports, per parameter segment, the max
over steps of  max|g_a - g_b|_seg / max(|g_a|_seg, |g_b|_seg)  between honest
replicas — the quantity the segment vote compares against rtol.  If the worst
segment ratio sits well under the row-level rtol (0.2), granularity="segment" is
safe at the same rtol and is the strictly tighter adversary bound.

  gpurun -- 'python tools/diag_segnoise.py > gpurun_out/segnoise.txt 2>&1'
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from draco_amd import ops
from draco_amd.config import Config
from draco_amd.parallel.trainer import Trainer


def main():
    steps = int(os.environ.get("SEG_STEPS", "30"))
    cfg = Config(network="ResNet18", dataset="Cifar10", batch_size=128,
                 approach="maj_vote", mode="maj_vote", group_size=3, worker_fail=0,
                 err_mode="none", device="cuda", dtype="bf16",
                 max_steps=steps + 10, eval_freq=0, log_dir="", train_dir="/tmp/ck")
    t = Trainer(cfg)
    t.logger.stdout_every = 0
    seg = t.space.seg_bounds.to(t.device)
    L = seg.numel() - 1
    pairs_a = torch.tensor([0, 0, 1], device=t.device)
    pairs_b = torch.tensor([1, 2, 2], device=t.device)
    worst = torch.zeros(3, L, device=t.device)
    worst_row = 0.0
    for _ in range(steps):
        t.train_step()  # weights advance; replicas (L=3 slots, same batch at world=1
        # they see different groups -> use a manual identical-batch probe instead)
        x, y = t.data.batch_for(0, t.step_num)
        for l in range(3):
            t._run_fwd_bwd(("slot", l), t.payload[l], x, y,
                           stream=t._worker_stream(l))
        if t._streams:
            for st in t._streams:
                torch.cuda.current_stream().wait_stream(st)
        recv = t.payload  # (3, d_pad) identical-batch honest replicas
        segdiff = ops.segment_pair_maxdiff(recv, pairs_a, pairs_b, seg)
        segmax = ops.segment_absmax(recv, seg)
        denom = torch.maximum(segmax[pairs_a], segmax[pairs_b]).clamp_min(1e-30)
        ratio = segdiff / denom
        worst = torch.maximum(worst, ratio)
        rowdiff = ops.pair_maxdiff(recv, pairs_a, pairs_b)
        rowmax = ops.row_absmax(recv)
        worst_row = max(worst_row, float((rowdiff / torch.maximum(
            rowmax[pairs_a], rowmax[pairs_b]).clamp_min(1e-30)).max()))
    w = worst.max(dim=0).values.cpu()
    names = [f"seg{i}" for i in range(L)]
    print(f"steps={steps} worst ROW-level honest ratio: {worst_row:.4f}")
    print(f"worst SEGMENT-level honest ratio (max over segments): {float(w.max()):.4f}")
    order = torch.argsort(w, descending=True)[:10]
    for i in order:
        print(f"  {names[int(i)]}: ratio {float(w[int(i)]):.4f} "
              f"(numel {int(t.space.numels[int(i)]) if int(i) < len(t.space.numels) else '?'})")
    print("VERDICT: segment rtol margin vs 0.2 =",
          f"{0.2 / max(float(w.max()), 1e-9):.1f}x")


if __name__ == "__main__":
    main()
