"""Communicator: RCCL (backend "nccl" on ROCm) over xGMI on GPU, gloo on CPU,
and a zero-cost local path for world_size == 1.

Design (vs the reference's PS star over mpi4py):
  * no parameter broadcast at all — every rank applies the identical decoded gradient
    with the identical fused optimizer kernel, so weights stay bit-identical by
    construction (replaces C2/C3);
  * gradients move as ONE all_to_all_single of d/world shards (colocated topology) —
    on xGMI every GPU drives its 7 point-to-point links concurrently, instead of
    serialising through one PS GPU's links (replaces C4/C5/C6);
  * cross-shard decode decisions (vote equality bits, Weiszfeld norms, Krum Gram,
    cyclic syndrome projections) travel as tiny allreduces.
"""
from __future__ import annotations

import datetime
import os

import torch
import torch.distributed as dist


class Communicator:
    def __init__(self, rank: int, world: int, device: torch.device, backend: str | None = None,
                 group=None):
        self.rank = rank
        self.world = world
        self.device = device
        self.distributed = world > 1
        self.group = group  # None = default process group; else a dist subgroup
        if self.distributed and not dist.is_initialized():
            if backend is None:
                backend = "nccl" if device.type == "cuda" else "gloo"
            os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
            os.environ.setdefault("MASTER_PORT", "29517")
            dist.init_process_group(
                backend=backend,
                rank=rank,
                world_size=world,
                # generous watchdog: first-step skew across ranks includes MIOpen
                # find + dynamo compile + graph capture (minutes on big models)
                timeout=datetime.timedelta(seconds=1800),
            )
        self.backend = dist.get_backend() if self.distributed else "local"

    def subgroup_communicator(self, group, ranks: list) -> "Communicator":
        """A Communicator bound to a pre-created dist subgroup (survivor set after a
        rank failure — parallel/health.py).  rank/world are POSITIONAL within the
        subgroup; collectives address only the surviving ranks."""
        c = Communicator.__new__(Communicator)
        c.device = self.device
        c.group = group
        c.world = len(ranks)
        c.rank = ranks.index(dist.get_rank()) if c.world > 1 else 0
        c.distributed = c.world > 1
        c.backend = self.backend
        return c

    @classmethod
    def from_env(cls, device: torch.device, backend: str | None = None) -> "Communicator":
        rank = int(os.environ.get("RANK", 0))
        world = int(os.environ.get("WORLD_SIZE", 1))
        return cls(rank, world, device, backend)

    # ------------------------------------------------------------------ collectives
    def barrier(self):
        if self.distributed:
            if self.backend == "nccl":
                dist.barrier(device_ids=[self.device.index], group=self.group)
            else:
                dist.barrier(group=self.group)

    def all_to_all_row(self, payload_row: torch.Tensor, recv_row: torch.Tensor,
                       async_op: bool = False):
        """One payload row (d_pad,) -> recv_row (world, shard): recv_row[src] is rank
        src's shard-for-me of this row.  Zero-copy (each rank's chunk of a contiguous
        flat row IS the send buffer); async_op returns the work handle so the exchange
        overlaps the remaining backward compute.
        """
        if not self.distributed:
            recv_row.view(-1).copy_(payload_row)
            return None
        work = dist.all_to_all_single(recv_row.view(-1), payload_row, group=self.group, async_op=async_op)
        return work if async_op else None

    def all_to_all_rows(self, payload: torch.Tensor) -> torch.Tensor:
        """payload: (L, world * shard) -> received (L * world, shard).

        ROW CONVENTION: row l * world + src of the result is rank src's local row l's
        shard for this rank (l-major).  world == 1 degenerates to a reshape (no copy).
        """
        L, d_pad = payload.shape
        shard = d_pad // self.world
        if not self.distributed:
            return payload.view(L, shard)
        recv = torch.empty(L, self.world, shard, dtype=payload.dtype, device=payload.device)
        works = [self.all_to_all_row(payload[l], recv[l], async_op=True) for l in range(L)]
        for w in works:
            if w is not None:
                w.wait()
        return recv.view(L * self.world, shard)

    def reduce_scatter_row(self, payload_row: torch.Tensor, out_shard: torch.Tensor,
                           async_op: bool = False):
        """payload_row (d_pad,) -> out_shard (shard,) = global sum of this rank's
        shard of the row.  The overlap-capable form of reduce_scatter_sum: the
        trainer posts one per payload row as its backward completes."""
        if not self.distributed:
            out_shard.copy_(payload_row)
            return None
        work = dist.reduce_scatter_tensor(out_shard, payload_row, group=self.group, async_op=async_op)
        return work if async_op else None

    def reduce_scatter_sum(self, payload_sum: torch.Tensor) -> torch.Tensor:
        """payload_sum: (d_pad,) local sum -> (shard,) global sum of this rank's shard."""
        shard = payload_sum.shape[0] // self.world
        if not self.distributed:
            return payload_sum
        out = torch.empty(shard, dtype=payload_sum.dtype, device=payload_sum.device)
        dist.reduce_scatter_tensor(out, payload_sum, group=self.group)
        return out

    def all_gather_shard(self, shard: torch.Tensor, out: torch.Tensor) -> None:
        """shard: (shard,) -> out: (d_pad,) gathered from all ranks."""
        if not self.distributed:
            out.copy_(shard)
            return
        dist.all_gather_into_tensor(out, shard.contiguous(), group=self.group)

    def all_reduce(self, t: torch.Tensor, op: str = "sum", async_op: bool = False):
        if not self.distributed:
            return t if not async_op else None
        red = {"sum": dist.ReduceOp.SUM, "min": dist.ReduceOp.MIN, "max": dist.ReduceOp.MAX}[op]
        work = dist.all_reduce(t, op=red, group=self.group, async_op=async_op)
        return work if async_op else t

    def all_to_all_bucket(self, send: torch.Tensor, recv_flat: torch.Tensor,
                          in_splits: list, out_split: int):
        """Unequal-split all_to_all for a bucket [lo, hi) of a payload row: rank j
        receives every rank's intersection of the bucket with j's shard chunk.
        send: the (hi-lo,) slice; recv_flat: (world*out_split,).  Always async (the
        per-layer overlap path posts these from backward hooks)."""
        return dist.all_to_all_single(
            recv_flat, send,
            output_split_sizes=[out_split] * self.world,
            input_split_sizes=in_splits,
            group=self.group,
            async_op=True,
        )

    def broadcast(self, t: torch.Tensor, src: int = 0) -> torch.Tensor:
        if self.distributed:
            dist.broadcast(t, src=src, group=self.group)
        return t

    # ------------------------------------------------------------------ p2p (PS mode)
    def send(self, t: torch.Tensor, dst: int, tag: int = 0):
        dist.send(t, dst=dst, tag=tag)

    def recv(self, t: torch.Tensor, src: int, tag: int = 0):
        dist.recv(t, src=src, tag=tag)

    def isend(self, t: torch.Tensor, dst: int, tag: int = 0):
        return dist.isend(t, dst=dst, tag=tag)

    def irecv(self, t: torch.Tensor, src: int, tag: int = 0):
        return dist.irecv(t, src=src, tag=tag)

    def shutdown(self):
        if self.distributed and dist.is_initialized():
            dist.destroy_process_group()
