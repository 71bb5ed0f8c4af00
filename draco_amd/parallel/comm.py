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
    def __init__(self, rank: int, world: int, device: torch.device, backend: str | None = None):
        self.rank = rank
        self.world = world
        self.device = device
        self.distributed = world > 1
        if self.distributed and not dist.is_initialized():
            if backend is None:
                backend = "nccl" if device.type == "cuda" else "gloo"
            os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
            os.environ.setdefault("MASTER_PORT", "29517")
            dist.init_process_group(
                backend=backend,
                rank=rank,
                world_size=world,
                timeout=datetime.timedelta(seconds=300),
            )
        self.backend = dist.get_backend() if self.distributed else "local"

    @classmethod
    def from_env(cls, device: torch.device, backend: str | None = None) -> "Communicator":
        rank = int(os.environ.get("RANK", 0))
        world = int(os.environ.get("WORLD_SIZE", 1))
        return cls(rank, world, device, backend)

    # ------------------------------------------------------------------ collectives
    def barrier(self):
        if self.distributed:
            if self.backend == "nccl":
                dist.barrier(device_ids=[self.device.index])
            else:
                dist.barrier()

    def all_to_all_rows(self, payload: torch.Tensor) -> torch.Tensor:
        """payload: (L, world * shard) -> received (world * L, shard).

        Row w = src * L + l of the result is rank src's local worker l's shard for
        this rank.  world == 1 degenerates to a reshape (no copy).
        """
        L, d_pad = payload.shape
        shard = d_pad // self.world
        if not self.distributed:
            return payload.view(L, shard)
        # (L, world, shard) -> (world, L, shard) so the send buffer is contiguous per
        # destination rank
        send = payload.view(L, self.world, shard).transpose(0, 1).contiguous()
        recv = torch.empty_like(send)
        dist.all_to_all_single(recv.view(-1), send.view(-1))
        return recv.view(self.world * L, shard)

    def reduce_scatter_sum(self, payload_sum: torch.Tensor) -> torch.Tensor:
        """payload_sum: (d_pad,) local sum -> (shard,) global sum of this rank's shard."""
        shard = payload_sum.shape[0] // self.world
        if not self.distributed:
            return payload_sum
        out = torch.empty(shard, dtype=payload_sum.dtype, device=payload_sum.device)
        dist.reduce_scatter_tensor(out, payload_sum)
        return out

    def all_gather_shard(self, shard: torch.Tensor, out: torch.Tensor) -> None:
        """shard: (shard,) -> out: (d_pad,) gathered from all ranks."""
        if not self.distributed:
            out.copy_(shard)
            return
        dist.all_gather_into_tensor(out, shard.contiguous())

    def all_reduce(self, t: torch.Tensor, op: str = "sum") -> torch.Tensor:
        if not self.distributed:
            return t
        red = {"sum": dist.ReduceOp.SUM, "min": dist.ReduceOp.MIN, "max": dist.ReduceOp.MAX}[op]
        dist.all_reduce(t, op=red)
        return t

    def broadcast(self, t: torch.Tensor, src: int = 0) -> torch.Tensor:
        if self.distributed:
            dist.broadcast(t, src=src)
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
