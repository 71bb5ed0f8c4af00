"""Flat parameter / gradient space.

The reference transmits and aggregates gradients tensor-by-tensor (62 MPI messages per
worker per step for ResNet-18, baseline_master.py:188-200).  On MI355X with 288 GB of
HBM3E the idiomatic layout is ONE contiguous fp32 buffer for all parameters and one
(L, d) payload buffer for the L local logical workers' gradients:

  * p.data is re-pointed at a view of the flat param buffer (zero-copy updates by the
    fused SGD/Adam kernel over one contiguous range);
  * before each logical worker's backward, p.grad is re-pointed at views of that
    worker's payload row, so autograd accumulates gradients directly into the comm
    buffer — the GPU analog of the reference's interleaved per-layer isends, with no
    packing pass at all;
  * d is padded so every rank owns an aligned d/world shard (all_to_all / reduce
    scatter friendly) — shard alignment 64 floats = 256 B.

Layer segment metadata (offsets per parameter tensor) is kept so the per-layer robust
aggregation semantics of the reference (geo-median / Krum per parameter tensor,
baseline_master.py:267-296) are preserved over the flat buffer.
"""
from __future__ import annotations

import torch

ALIGN = 64  # floats; 256 B


def _pad(d: int, world: int) -> int:
    q = world * ALIGN
    return (d + q - 1) // q * q


class FlatSpace:
    """channels_last=True stores 4D (conv-weight) parameters NHWC inside the flat
    buffer and re-points p.data at a permuted view with channels_last strides, so
    MIOpen picks its NHWC kernels (+7% measured on ResNet-18 fwd+bwd) while the flat
    comm/decode space stays one contiguous fp32 range."""

    def __init__(self, model: torch.nn.Module, world: int, device: torch.device,
                 channels_last: bool = False):
        self.device = device
        self.world = world
        self.channels_last = channels_last
        params = [p for p in model.parameters() if p.requires_grad]
        self.params = params
        self.shapes = [p.shape for p in params]
        self.numels = [p.numel() for p in params]
        self.offsets = []
        off = 0
        for n in self.numels:
            self.offsets.append(off)
            off += n
        self.d = off
        self.d_pad = _pad(off, world)
        self.shard = self.d_pad // world

        self.flat_param = torch.zeros(self.d_pad, dtype=torch.float32, device=device)
        with torch.no_grad():
            for p, o, n in zip(params, self.offsets, self.numels):
                if self._is_cl(p):
                    src = p.data.to(device=device, dtype=torch.float32)
                    self.flat_param[o : o + n].copy_(src.permute(0, 2, 3, 1).reshape(-1))
                else:
                    self.flat_param[o : o + n].copy_(p.data.reshape(-1).to(device=device, dtype=torch.float32))
                p.data = self._view(self.flat_param, o, n, p.shape)

        # (L+1) segment bounds in global flat coordinates (one segment per parameter)
        self.seg_bounds = torch.tensor(self.offsets + [self.d], dtype=torch.int64)

    def _is_cl(self, p) -> bool:
        return self.channels_last and p.dim() == 4

    def _view(self, buf, o, n, shape):
        if self.channels_last and len(shape) == 4:
            O, C, H, W = shape
            return buf[o : o + n].view(O, H, W, C).permute(0, 3, 1, 2)
        return buf[o : o + n].view(shape)

    # ------------------------------------------------------------------ grads
    def alloc_payload(self, rows: int) -> torch.Tensor:
        return torch.zeros(rows, self.d_pad, dtype=torch.float32, device=self.device)

    def attach_grads(self, buf: torch.Tensor) -> None:
        """Point every parameter's .grad at views of the given flat (d_pad,) buffer."""
        assert buf.shape == (self.d_pad,)
        for p, o, n, shape in zip(self.params, self.offsets, self.numels, self.shapes):
            p.grad = self._view(buf, o, n, shape)

    def detach_grads(self) -> None:
        for p in self.params:
            p.grad = None

    # ------------------------------------------------------------------ buckets
    def build_buckets(self, bucket_mb: float):
        """Partition the parameter list into contiguous flat ranges of ~bucket_mb,
        grouped in REVERSE parameter order (the approximate autograd fire order),
        for the per-layer comm/compute overlap (post-accumulate-grad hooks launch
        each bucket's exchange while the rest of backward computes).

        Range boundaries are snapped UP to 64-float alignment (the slice kernels
        stream float4): the few head elements a snap excludes belong to the
        EARLIER-firing neighbour bucket's last param, which is final by the time
        this bucket ships, so they ride with this (later) bucket.  The first-fired
        bucket extends to d_pad — the zero padding ships with it, keeping bucketed
        and whole-row wire traffic identical.

        Returns [(lo, hi, (param_indices...)), ...] in expected fire order;
        ranges are disjoint and cover [0, d_pad) exactly.
        """
        limit = max(int(bucket_mb * 1024 * 1024 / 4), 1)
        groups, cur, cur_elems = [], [], 0
        for idx in reversed(range(len(self.params))):
            cur.append(idx)
            cur_elems += self.numels[idx]
            if cur_elems >= limit:
                groups.append(cur)
                cur, cur_elems = [], 0
        if cur:
            groups.append(cur)
        out = []
        prev_lo = self.d_pad  # first-fired bucket ends at the padded top
        for group in groups:
            lo = min(self.offsets[i] for i in group)
            lo = (lo + ALIGN - 1) // ALIGN * ALIGN if lo > 0 else 0
            lo = min(lo, prev_lo)
            out.append((lo, prev_lo, tuple(group)))
            prev_lo = lo
        if prev_lo > 0:  # snap of the last group must still cover [0, ..)
            last_lo, last_hi, idxs = out[-1]
            out[-1] = (0, last_hi, idxs)
        return out

    # ------------------------------------------------------------------ shards
    def local_seg_bounds(self, rank: int) -> torch.Tensor:
        """Segment bounds clipped to this rank's shard, in shard-local coordinates.

        Returns (L+1,) int64 monotone bounds (segments outside the shard are empty).
        """
        lo = rank * self.shard
        hi = lo + self.shard
        return (self.seg_bounds.clamp(min=lo, max=hi) - lo).contiguous()

    def shard_of(self, flat: torch.Tensor, rank: int) -> torch.Tensor:
        return flat[rank * self.shard : (rank + 1) * self.shard]

