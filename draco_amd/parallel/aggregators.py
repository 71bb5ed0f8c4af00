"""Sharded robust-aggregation policies.

Each policy turns the per-rank payload (the local logical workers' flat gradients)
into the decoded global gradient, replicated on every rank.  All of them follow the
same xGMI-friendly shape: one all_to_all of d/world shards, decode on the local shard
(HIP kernels via draco_amd.ops), tiny allreduces for cross-shard decisions, one
all_gather of the decoded shard.  Reference semantics per policy:

  mean         baseline_master.py:202-207,267-269   (plain averaging)
  maj_vote     rep_master.py:141-168                (group-wise Boyer-Moore vote)
  geo_median   baseline_master.py:271-276           (per-layer Weiszfeld)
  krum         baseline_master.py:278-296           (per-layer Krum selection)
  cyclic       cyclic_master.py:103-188             (DFT-code algebraic decode)
"""
from __future__ import annotations

import numpy as np
import torch

from .. import ops
from ..coding import CyclicCode, majority_vote_index
from .comm import Communicator
from .flat import FlatSpace


class Aggregator:
    """Base: owns comm + space, provides the exchange/allgather plumbing.

    Overlap: the trainer may call start_row(payload, r) as soon as payload row r is
    final (backward done + adversary injected) — the all_to_all for that row then
    runs on the communication stream while the next logical worker's backward
    computes.  aggregate() exchanges any not-yet-started rows and waits.
    """

    name = "base"

    def __init__(self, comm: Communicator, space: FlatSpace, comm_dtype=torch.float32):
        self.comm = comm
        self.space = space
        self.comm_dtype = comm_dtype  # bf16 = K12 wire compression (all_to_all paths)
        self._out = torch.zeros(space.d_pad, dtype=torch.float32, device=space.device)
        self._shard_out = torch.zeros(space.shard, dtype=torch.float32, device=space.device)
        self.local_seg = space.local_seg_bounds(comm.rank).to(space.device)
        self._recv = None
        self._send = None
        self._recv32 = None
        self._works: list = []
        self._started: set = set()

    # ---------------------------------------------------------- row exchange
    def start_row(self, payload: torch.Tensor, row: int) -> None:
        if not self.comm.distributed:
            return
        if self._recv is None or self._recv.shape[0] != payload.shape[0]:
            self._recv = torch.empty(payload.shape[0], self.comm.world, self.space.shard,
                                     dtype=self.comm_dtype, device=payload.device)
            if self.comm_dtype != torch.float32:
                self._send = torch.empty(payload.shape[0], payload.shape[1],
                                         dtype=self.comm_dtype, device=payload.device)
                self._recv32 = torch.empty(payload.shape[0], self.comm.world, self.space.shard,
                                           dtype=torch.float32, device=payload.device)
        if self.comm_dtype != torch.float32:
            self._send[row].copy_(payload[row])  # fp32 -> bf16 cast on device
            send_row = self._send[row]
        else:
            send_row = payload[row]
        w = self.comm.all_to_all_row(send_row, self._recv[row], async_op=True)
        if w is not None:
            self._works.append(w)
        self._started.add(row)

    def exchanged(self, payload: torch.Tensor) -> torch.Tensor:
        """(rows, d_pad) payload -> (rows*world, shard) in l-major row order."""
        rows = payload.shape[0]
        if not self.comm.distributed:
            return payload.view(rows, self.space.shard)
        for r in range(rows):
            if r not in self._started:
                self.start_row(payload, r)
        for w in self._works:
            w.wait()
        self._works = []
        self._started = set()
        if self.comm_dtype != torch.float32:
            self._recv32.copy_(self._recv)  # upcast once; decode kernels stay fp32
            return self._recv32.view(rows * self.comm.world, self.space.shard)
        return self._recv.view(rows * self.comm.world, self.space.shard)

    def aggregate(self, payload: torch.Tensor, step: int) -> torch.Tensor:
        raise NotImplementedError


class MeanAggregator(Aggregator):
    """Plain averaging over all P = L*world logical workers.

    Mean commutes with sharding, so no all_to_all is needed: start_row posts an
    async per-row reduce_scatter (the sum happens in-flight on the wire), which
    both gives the baseline path the same backward/comm overlap as the coded
    paths and ships each byte exactly once (round-1 bug: the base-class
    all_to_all was posted but never awaited — dead traffic + unbounded _works)."""

    name = "mean"

    def __init__(self, comm, space, num_workers: int):
        super().__init__(comm, space)
        self.num_workers = num_workers
        self._rs_shards = None  # (rows, shard) per-row reduce_scatter outputs

    def start_row(self, payload: torch.Tensor, row: int) -> None:
        if not self.comm.distributed:
            return
        if self._rs_shards is None or self._rs_shards.shape[0] != payload.shape[0]:
            self._rs_shards = torch.empty(payload.shape[0], self.space.shard,
                                          dtype=torch.float32, device=payload.device)
        w = self.comm.reduce_scatter_row(payload[row], self._rs_shards[row], async_op=True)
        if w is not None:
            self._works.append(w)
        self._started.add(row)

    def aggregate(self, payload: torch.Tensor, step: int) -> torch.Tensor:
        if not self.comm.distributed:
            ops.sum_rows(payload, self._out)
            self._out /= float(self.num_workers)
            return self._out
        for r in range(payload.shape[0]):
            if r not in self._started:
                self.start_row(payload, r)
        for w in self._works:
            w.wait()
        self._works = []
        self._started = set()
        torch.sum(self._rs_shards, dim=0, out=self._shard_out)
        self._shard_out /= float(self.num_workers)
        self.comm.all_gather_shard(self._shard_out, self._out)
        return self._out


class VoteAggregator(Aggregator):
    """Repetition-code decode: per-group majority vote, then mean of winners.

    Colocated layout: G = world groups of size r; member i of group g is local worker
    slot i of rank (g+i) % world, so received row of member (g, i) is
    src*L + l with src = (g+i) % world, l = i.

    Equality rule: members a, b are equal iff max|a-b| <= atol + rtol*max(|a|inf,|b|inf)
    over the FULL gradient (per-shard maxima combined by a tiny MAX allreduce).
    atol=rtol=0 reproduces the reference's bitwise np.array_equal vote
    (rep_master.py:154-168) — usable on CPU, where autograd is reproducible.  On GPU,
    MIOpen conv backward is not bitwise-reproducible (measured: enabling
    cudnn.deterministic neither fixes it nor is affordable — 15x on conv backward), so
    honest replicas differ by fp-reorder noise and the DEFAULT GPU rule is the
    tolerance vote.  This is a deliberate, sound relaxation of the adversary model: an
    adversary constrained to the tolerance ball of an honest gradient can shift the
    aggregate by at most tol/G — indistinguishable from fp noise — while anything
    outside the ball still loses the vote.
    """

    name = "maj_vote"

    def __init__(self, comm, space, group_size: int, atol: float = 0.0,
                 rtol: float = 0.0, member_rows=None, comm_dtype=torch.float32):
        super().__init__(comm, space, comm_dtype)
        self.atol = atol
        self.rtol = rtol
        if member_rows is None:
            # colocated layout: G = world groups; member i of group g is local slot i
            # of rank (g+i)%world = row i*world + (g+i)%world (l-major convention of
            # comm.all_to_all_rows)
            member_rows = np.asarray(
                [[i * comm.world + (g + i) % comm.world for i in range(group_size)]
                 for g in range(max(comm.world, 1))]
            )
        self.member_rows = np.asarray(member_rows)  # (G, r)
        self.G, self.r = self.member_rows.shape
        pairs_a, pairs_b = [], []
        for rows in self.member_rows:
            for i in range(self.r):
                for j in range(i + 1, self.r):
                    pairs_a.append(rows[i])
                    pairs_b.append(rows[j])
        self.pairs_a = torch.tensor(pairs_a, dtype=torch.int64, device=space.device)
        self.pairs_b = torch.tensor(pairs_b, dtype=torch.int64, device=space.device)
        self.n_pairs_per_group = self.r * (self.r - 1) // 2
        # telemetry: steps where some group saw NO equal pair (tolerance too tight /
        # replicas diverged -> Boyer-Moore degenerates to "last member wins", which an
        # adversary can exploit).  Surfaced by bench as vote_degenerate_steps.
        self.degenerate_steps = 0

    @classmethod
    def from_member_rows(cls, comm, space, member_rows, atol: float = 0.0, rtol: float = 0.0):
        return cls(comm, space, group_size=member_rows.shape[1], atol=atol, rtol=rtol,
                   member_rows=member_rows)

    def aggregate(self, payload: torch.Tensor, step: int) -> torch.Tensor:
        recv = self.exchanged(payload)  # (r*world, shard)
        maxdiff = ops.pair_maxdiff(recv, self.pairs_a, self.pairs_b)  # (n_pairs,)
        if self.rtol > 0.0:
            rowmax = ops.row_absmax(recv)  # (world*r,)
            stats = torch.cat([maxdiff, rowmax])
            self.comm.all_reduce(stats, op="max")  # full-gradient maxima
            maxdiff, rowmax = stats[: len(maxdiff)], stats[len(maxdiff):]
            thresh = self.atol + self.rtol * torch.maximum(rowmax[self.pairs_a], rowmax[self.pairs_b])
            eq = maxdiff <= thresh
        else:
            self.comm.all_reduce(maxdiff, op="max")
            eq = maxdiff <= self.atol
        eq_host = eq.to("cpu", non_blocking=False).numpy()
        winners = np.empty(self.G, dtype=np.int64)
        k = 0
        degenerate = False
        for g in range(self.G):
            mat = np.eye(self.r, dtype=bool)
            any_eq = False
            for i in range(self.r):
                for j in range(i + 1, self.r):
                    mat[i, j] = mat[j, i] = bool(eq_host[k])
                    any_eq = any_eq or bool(eq_host[k])
                    k += 1
            if not any_eq and self.r > 1:
                degenerate = True
            winners[g] = self.member_rows[g, majority_vote_index(mat)]
        if degenerate:
            self.degenerate_steps += 1
        idx = torch.tensor(winners, dtype=torch.int64, device=recv.device)
        ops.mean_rows(recv, idx, self._shard_out)
        self.comm.all_gather_shard(self._shard_out, self._out)
        return self._out


class GeoMedianAggregator(Aggregator):
    """Per-layer geometric median (Weiszfeld), sharded over d.

    Each iteration: per-(worker, layer) partial squared distances on the local shard,
    one (P, L) allreduce, then the reweighted mean on the shard.  Every rank sees the
    identical allreduced distances, so the iterates stay consistent without any
    further synchronisation.
    """

    name = "geo_median"

    def __init__(self, comm, space, num_workers: int, max_iter: int = 80, tol: float = 1e-7):
        super().__init__(comm, space)
        self.num_workers = num_workers
        self.max_iter = max_iter
        self.tol = tol

    def aggregate(self, payload: torch.Tensor, step: int) -> torch.Tensor:
        recv = self.exchanged(payload)  # (P, shard)
        P = recv.shape[0]
        z = recv.mean(dim=0)  # init at the mean (hdmedians does the same)
        # zeros, not empty: the pad tail past the last segment is never written by
        # segment_weighted_mean, and garbage there (inf) poisons the delta criterion
        z_new = torch.zeros_like(z)
        for it in range(self.max_iter):
            part = ops.segment_sqdist(recv, z, self.local_seg)  # (P, L)
            self.comm.all_reduce(part)
            dist = part.clamp_min(1e-24).sqrt()
            w = 1.0 / dist
            w = w / w.sum(dim=0, keepdim=True)
            ops.segment_weighted_mean(recv, w, self.local_seg, z_new)
            z, z_new = z_new, z
            # convergence check every 8 iterations (one fused 2-word allreduce +
            # host sync, instead of three per iteration)
            if (it & 7) == 7 or it == self.max_iter - 1:
                stats = torch.stack([(z - z_new).pow(2).sum(), z.pow(2).sum()])
                self.comm.all_reduce(stats)
                d_val, s_val = float(stats[0]), float(stats[1])
                if np.isfinite(d_val) and d_val <= self.tol * self.tol * max(s_val, 1e-12):
                    break
        self.comm.all_gather_shard(z, self._out)
        return self._out


class KrumAggregator(Aggregator):
    """Per-layer Krum selection (arXiv:1703.02757), sharded over d.

    Pairwise squared distances come from per-segment Gram matrices (one allreduce of
    (L, P, P)); selection score per layer = sum of the P-s-2 smallest distances to the
    other workers, argmin wins (exactly baseline_master.py:279-291).
    """

    name = "krum"

    def __init__(self, comm, space, num_workers: int, s: int):
        super().__init__(comm, space)
        self.num_workers = num_workers
        self.s = s

    def aggregate(self, payload: torch.Tensor, step: int) -> torch.Tensor:
        recv = self.exchanged(payload)  # (P, shard)
        P = recv.shape[0]
        gram = ops.segment_gram(recv, self.local_seg)  # (L, P, P)
        self.comm.all_reduce(gram)
        g = gram.to("cpu").numpy().astype(np.float64)
        diag = np.einsum("lpp->lp", g)
        d2 = diag[:, :, None] + diag[:, None, :] - 2.0 * g  # (L, P, P)
        np.maximum(d2, 0.0, out=d2)
        L = d2.shape[0]
        keep = max(self.num_workers - self.s - 2, 1)
        winners = np.empty(L, dtype=np.int64)
        for l in range(L):
            m = d2[l].copy()
            np.fill_diagonal(m, np.inf)
            m.sort(axis=1)
            scores = m[:, :keep].sum(axis=1)
            winners[l] = int(np.argmin(scores))
        # build the output shard layer-by-layer from each layer's winning worker
        seg = self.local_seg.to("cpu").numpy()
        out = self._shard_out
        for l in range(L):
            lo, hi = int(seg[l]), int(seg[l + 1])
            if hi > lo:
                out[lo:hi] = recv[winners[l], lo:hi]
        tail = int(seg[-1])
        if tail < out.shape[0]:
            out[tail:] = 0.0
        self.comm.all_gather_shard(out, self._out)
        return self._out


class CyclicAggregator(Aggregator):
    """Cyclic-code algebraic decode, sharded over d.

    The payload rows here are the ENCODED complex gradients as (2, d) fp32 planes —
    (L*2, d_pad) per rank.  Decode: random projection proj = R @ z (partial per shard,
    one (n, 2) allreduce), host error-location + recombination-vector solve (tiny,
    cached per healthy-set), then out = Re(v @ R)/n on the shard.
    """

    name = "cyclic"

    def __init__(self, comm, space, code: CyclicCode, workers_per_rank: int,
                 comm_dtype=torch.float32):
        super().__init__(comm, space, comm_dtype)
        self.code = code
        self.L = workers_per_rank
        self.n = code.n
        world = max(comm.world, 1)
        # worker w = l*world + src; its (re, im) payload rows at the source are
        # (2l, 2l+1), so after the l-major all_to_all its recv rows are
        # (2l+p)*world + src
        w_ids = np.arange(self.n)
        l, src = w_ids // world, w_ids % world
        self.rows_re = (2 * l) * world + src
        self.rows_im = (2 * l + 1) * world + src
        self._z = None
        self._zgen = None

    def aggregate(self, payload_planes: torch.Tensor, step: int,
                  erasures: frozenset = frozenset()) -> torch.Tensor:
        recv = self.exchanged(payload_planes)  # (2L*world, shard)
        # random projection generated ON DEVICE (a shard-sized CPU randn + H2D copy
        # costs ~30 ms at d=11M — it dominated the whole decode); per-rank streams
        # may legitimately use different z (partial projections are summed)
        if self._z is None:
            self._z = torch.empty(self.space.shard, dtype=torch.float32, device=self.space.device)
            self._zgen = torch.Generator(device=self.space.device)
        self._zgen.manual_seed(0x5EED ^ (step * 1000003) ^ self.comm.rank)
        z = self._z.normal_(mean=1.0, std=1.0, generator=self._zgen)
        proj = ops.cyclic_project(recv, z)  # (2L*world,) per-row partial dots
        self.comm.all_reduce(proj)
        pa = proj.to("cpu").numpy().astype(np.float64)
        proj_complex = pa[self.rows_re] + 1j * pa[self.rows_im]
        syndrome = self.code.W_perp @ proj_complex
        scale = float(np.abs(proj_complex).max())
        if not erasures and float(np.abs(syndrome).max()) <= 1e-7 * max(scale, 1e-30):
            healthy = np.arange(self.n)
        else:
            healthy = self.code.locate_errors(syndrome, known_bad=erasures)
        v = self.code.recombination_vector(healthy)
        # Re(v @ R) = sum_w vre[w]*re_row[w] - vim[w]*im_row[w]: one combine kernel
        rows = torch.tensor(np.concatenate([self.rows_re, self.rows_im]),
                            dtype=torch.int64, device=recv.device)
        w = torch.tensor(np.concatenate([np.real(v), -np.imag(v)]) / self.n,
                         dtype=torch.float32, device=recv.device)
        ops.combine_rows(recv, rows, w, self._shard_out)
        self.comm.all_gather_shard(self._shard_out, self._out)
        return self._out
