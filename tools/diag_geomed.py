"""Pinpoint the GPU GeoMedian divergence: compare each op (segment_sqdist,
segment_weighted_mean) GPU vs CPU at the failing test's exact shapes, then trace
the Weiszfeld iteration divergence step by step.

  gpurun -- 'python tools/diag_geomed.py'
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn as nn

from draco_amd import ops
from draco_amd.ops import fallback as fb
from draco_amd.parallel.flat import FlatSpace


def main():
    device = torch.device("cuda:0")
    torch.manual_seed(3)
    model = nn.Sequential(nn.Linear(200, 50), nn.Linear(50, 7))
    space = FlatSpace(model, 1, torch.device("cpu"))
    P = 7
    torch.manual_seed(0)
    payload = torch.randn(P, space.d_pad)
    payload[2] *= 40.0
    seg = space.local_seg_bounds(0)
    print("d_pad", space.d_pad, "seg", seg.tolist())

    xg = payload.to(device)
    z = payload.mean(dim=0)
    zg = z.to(device)

    part_c = fb.segment_sqdist(payload, z, seg)
    part_g = ops.segment_sqdist(xg, zg, seg).cpu()
    print("sqdist maxreldiff:", float(((part_c - part_g).abs() / part_c.clamp(min=1e-9)).max()))
    print("cpu:", part_c.flatten()[:8].tolist())
    print("gpu:", part_g.flatten()[:8].tolist())

    dist = part_c.clamp_min(1e-24).sqrt()
    w = 1.0 / dist
    w = w / w.sum(dim=0, keepdim=True)
    out_c = torch.zeros(space.d_pad)
    fb.segment_weighted_mean(payload, w, seg, out_c)
    out_g = torch.zeros(space.d_pad, device=device)
    ops.segment_weighted_mean(xg, w.to(device), seg, out_g)
    d = (out_c - out_g.cpu()).abs()
    print("wmean maxdiff:", float(d.max()), "at", int(d.argmax()))
    i = int(d.argmax())
    segl = int((seg[:-1] <= i).sum()) - 1
    print("  index", i, "segment", segl, "cpu", float(out_c[i]), "gpu", float(out_g[i]))
    # manual torch-on-GPU reference: distinguishes kernel bug vs fallback bug
    man = torch.zeros(space.d_pad, device=device)
    wgpu = w.to(device)
    for l in range(len(seg) - 1):
        lo, hi = int(seg[l]), int(seg[l + 1])
        man[lo:hi] = wgpu[:, l] @ xg[:, lo:hi]
    print("  manual-gpu vs cpu:", float((man.cpu() - out_c).abs().max()),
          " manual-gpu vs kernel:", float((man - out_g).abs().max().cpu()))
    print("  mismatching elements:", int((d > 1e-3).sum()), "of", space.d_pad)
    bad = (d > 1e-3).nonzero().flatten()
    print("  first bad idxs:", bad[:10].tolist())

    # full iteration trace
    z_c, z_g = z.clone(), z.to(device).clone()
    zc_new, zg_new = torch.zeros_like(z_c), torch.zeros_like(z_g)
    for it in range(24):
        pc = fb.segment_sqdist(payload, z_c, seg)
        pg = ops.segment_sqdist(xg, z_g, seg)
        wc = 1.0 / pc.clamp_min(1e-24).sqrt()
        wc = wc / wc.sum(dim=0, keepdim=True)
        wg = 1.0 / pg.clamp_min(1e-24).sqrt()
        wg = wg / wg.sum(dim=0, keepdim=True)
        fb.segment_weighted_mean(payload, wc, seg, zc_new)
        ops.segment_weighted_mean(xg, wg, seg, zg_new)
        z_c, zc_new = zc_new, z_c
        z_g, zg_new = zg_new, z_g
        d = float((z_c - z_g.cpu()).abs().max())
        print(f"iter {it}: z maxdiff {d:.4e}")
        if d > 1.0:
            break


if __name__ == "__main__":
    main()
