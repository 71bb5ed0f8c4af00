"""Achieved-bandwidth microbench for every draco HIP kernel at bench-realistic
shapes.  All of these are HBM-bound streaming kernels (RESULTS.md design note), so
GB/s vs the ~8 TB/s HBM3E peak is the speed-of-light scorecard.

  gpurun -- 'python tools/kernel_bw.py > gpurun_out/kernel_bw.txt 2>&1'
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from draco_amd import ops

DEV = "cuda:0"


def t_ms(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def report(name, ms, bytes_moved):
    print(f"{name:34} {ms*1e3:9.1f} us  {bytes_moved/ms/1e6:8.1f} GB/s")


def main():
    d = 11_173_952  # ResNet-18 flat space (d_pad at world=1)
    d = (d + 63) // 64 * 64
    rows = 24  # r=3 x world=8 recv rows
    x = torch.randn(rows, d, device=DEV)
    out = torch.empty(d, device=DEV)
    shard_out = torch.empty(d, device=DEV)
    z = torch.randn(d, device=DEV)
    seg = torch.linspace(0, d, 63, dtype=torch.int64).to(DEV)
    pa = torch.arange(0, 12, device=DEV, dtype=torch.int64)
    pb = torch.arange(12, 24, device=DEV, dtype=torch.int64)

    B = 4 * rows * d  # fp32 bytes in the full payload

    ms = t_ms(lambda: ops.row_absmax(x))
    report("row_absmax (24 x 11.2M)", ms, B)
    ms = t_ms(lambda: ops.pair_maxdiff(x, pa, pb))
    report("pair_maxdiff (12 pairs)", ms, B)
    ms = t_ms(lambda: ops.segment_absmax(x, seg))
    report("segment_absmax (62 segs)", ms, B)
    ms = t_ms(lambda: ops.segment_pair_maxdiff(x, pa, pb, seg))
    report("segment_pair_maxdiff", ms, B)
    ms = t_ms(lambda: ops.cyclic_project(x, z))
    report("cyclic_project (24 rows)", ms, B + 4 * d)
    ms = t_ms(lambda: ops.mean_rows(x, pa, out))
    report("mean_rows (12 rows)", ms, 4 * 12 * d + 4 * d)
    ms = t_ms(lambda: ops.sum_rows(x, out))
    report("sum_rows (24 rows)", ms, B + 4 * d)
    w = torch.randn(rows, device=DEV)
    ms = t_ms(lambda: ops.combine_rows(x, torch.arange(rows, device=DEV), w, out))
    report("combine_rows (24 rows)", ms, B + 4 * d)
    g5 = torch.randn(5, d, device=DEV)
    wre, wim = torch.randn(5, device=DEV), torch.randn(5, device=DEV)
    enc = torch.empty(2, d, device=DEV)
    ms = t_ms(lambda: ops.cyclic_encode(g5, wre, wim, enc))
    report("cyclic_encode (5 rows->2)", ms, 4 * 5 * d + 8 * d)
    p = torch.randn(d, device=DEV)
    gr = torch.randn(d, device=DEV)
    buf = torch.zeros(d, device=DEV)
    ms = t_ms(lambda: ops.fused_sgd_step(p, gr, buf, lr=0.01, momentum=0.9, dampening=0.0,
                                         weight_decay=0.0, nesterov=False, first_step=False))
    report("fused_sgd (momentum)", ms, 4 * d * 4)  # r p,g,buf + w p,buf ~ 4-5 streams
    m1 = torch.zeros(d, device=DEV)
    v1 = torch.zeros(d, device=DEV)
    ms = t_ms(lambda: ops.fused_adam_step(p, gr, m1, v1, None, step=5, lr=1e-3, beta1=0.9,
                                          beta2=0.999, eps=1e-8, weight_decay=0.0,
                                          amsgrad=False))
    report("fused_adam", ms, 4 * d * 7)
    ms = t_ms(lambda: ops.inject_(gr, "rev_grad"))
    report("inject rev_grad", ms, 4 * d * 2)
    part = t_ms(lambda: ops.segment_sqdist(x, z, seg))
    report("segment_sqdist (geomed)", part, B + 4 * d)
    wt = torch.rand(rows, 62, device=DEV)
    wt = wt / wt.sum(0, keepdim=True)
    ms = t_ms(lambda: ops.segment_weighted_mean(x, wt, seg, out))
    report("segment_weighted_mean", ms, B + 4 * d)
    ms = t_ms(lambda: ops.segment_gram(x, seg))
    report("segment_gram (krum)", ms, B)


def gram_compare():
    """Hand-written k_seg_gram vs the rocBLAS route (62 torch.mm per-segment GEMMs
    — the brief's 'plain library GEMMs belong on MFMA' case)."""
    from draco_amd.ops import fallback as fb
    d = 11_173_952
    d = (d + 63) // 64 * 64
    rows = 24
    x = torch.randn(rows, d, device=DEV)
    seg = torch.linspace(0, d, 63, dtype=torch.int64).to(DEV)
    B = 4 * rows * d
    ms = t_ms(lambda: ops.segment_gram(x, seg), iters=10, warmup=3)
    report("segment_gram HIP kernel", ms, B)
    ms = t_ms(lambda: fb.segment_gram(x, seg), iters=10, warmup=3)
    report("segment_gram rocBLAS (62 mm)", ms, B)


if __name__ == "__main__":
    if os.environ.get("GRAM_ONLY") == "1":
        gram_compare()
        raise SystemExit(0)
    main()
