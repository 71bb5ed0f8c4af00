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
from ..coding import CyclicCode
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
        self._pending_copies: list = []  # (tmp2d, row, slo, shi) from bucket exchanges

    # ---------------------------------------------------------- row exchange
    def _ensure_recv(self, rows: int, device) -> None:
        if self._recv is None or self._recv.shape[0] != rows:
            # zeros, not empty: the bucketed path never transmits the [d, d_pad)
            # padding tail (payload padding is always zero), so recv must start zero
            self._recv = torch.zeros(rows, self.comm.world, self.space.shard,
                                     dtype=self.comm_dtype, device=device)
            if self.comm_dtype != torch.float32:
                self._send = torch.empty(rows, self.space.d_pad,
                                         dtype=self.comm_dtype, device=device)
                self._recv32 = torch.empty(rows, self.comm.world, self.space.shard,
                                           dtype=torch.float32, device=device)

    def start_row(self, payload: torch.Tensor, row: int) -> None:
        if not self.comm.distributed:
            return
        self._ensure_recv(payload.shape[0], payload.device)
        if self.comm_dtype != torch.float32:
            self._send[row].copy_(payload[row])  # fp32 -> bf16 cast on device
            send_row = self._send[row]
        else:
            send_row = payload[row]
        w = self.comm.all_to_all_row(send_row, self._recv[row], async_op=True)
        if w is not None:
            self._works.append(w)
        self._started.add(row)

    def start_bucket(self, payload: torch.Tensor, row: int, lo: int, hi: int) -> None:
        """Exchange just the [lo, hi) flat range of one payload row (posted from a
        post-accumulate-grad hook while the rest of backward still computes — the
        per-layer comm/compute overlap, reference lenet.py:114-218 interleaved
        isends).  Same wire format as start_row, delivered into the same recv slots;
        the trainer calls mark_row_started(row) once every bucket of the row is
        posted."""
        if not self.comm.distributed:
            return
        self._ensure_recv(payload.shape[0], payload.device)
        world, shard, rank = self.comm.world, self.space.shard, self.comm.rank
        if self.comm_dtype != torch.float32:
            self._send[row, lo:hi].copy_(payload[row, lo:hi])
            src = self._send[row, lo:hi]
        else:
            src = payload[row, lo:hi]
        in_splits = [max(0, min(hi, (j + 1) * shard) - max(lo, j * shard)) for j in range(world)]
        mlo, mhi = max(lo, rank * shard), min(hi, (rank + 1) * shard)
        m = max(0, mhi - mlo)
        tmp = torch.empty(world * m, dtype=self.comm_dtype, device=payload.device)
        w = self.comm.all_to_all_bucket(src, tmp, in_splits, m)
        if w is not None:
            self._works.append(w)
        if m > 0:
            self._pending_copies.append((tmp.view(world, m), row, mlo - rank * shard, mhi - rank * shard))

    def mark_row_started(self, row: int) -> None:
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
        for tmp2d, row, slo, shi in self._pending_copies:
            self._recv[row, :, slo:shi].copy_(tmp2d)
        self._pending_copies = []
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
        self._bucketed = False

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

    def start_bucket(self, payload: torch.Tensor, row: int, lo: int, hi: int) -> None:
        """Mean commutes with summing in-flight: the bucket is an in-place async
        all_reduce of the payload slice (the hook guarantees autograd is done with
        the range), so aggregate() just rescales the row."""
        if not self.comm.distributed:
            return
        w = self.comm.all_reduce(payload[row, lo:hi], async_op=True)
        if w is not None:
            self._works.append(w)
        self._bucketed = True

    def aggregate(self, payload: torch.Tensor, step: int) -> torch.Tensor:
        if not self.comm.distributed:
            ops.sum_rows(payload, self._out)
            self._out /= float(self.num_workers)
            return self._out
        if self._bucketed:
            for w in self._works:
                w.wait()
            self._works = []
            self._started = set()
            self._bucketed = False
            ops.sum_rows(payload, self._out)  # rows now hold cross-rank sums
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
    tolerance vote.  This is a deliberate, MEASURED relaxation of the adversary
    model (tests/test_vote_security.py): a within-ball adversary that wins its
    group's tie-break shifts that group's contribution by at most the ball radius
    (atol + rtol*max|g|) — so the aggregate over G groups by at most s/G of it —
    and training under a persistent within-ball attack still tracks the clean
    curve; anything outside the ball loses the vote.  granularity="segment"
    tightens the ball to per parameter tensor (atol + rtol*max|g_seg| on EVERY
    segment; honest noise margin measured in profiles/segnoise_r02.txt).
    """

    name = "maj_vote"

    def __init__(self, comm, space, group_size: int, atol: float = 0.0,
                 rtol: float = 0.0, member_rows=None, comm_dtype=torch.float32,
                 member_mask=None, granularity: str = "row"):
        super().__init__(comm, space, comm_dtype)
        self.atol = atol
        self.rtol = rtol
        # granularity="segment": the tolerance ball is PER PARAMETER TENSOR
        # (max|a-b| <= atol + rtol*max(segmax_a, segmax_b) must hold on EVERY
        # segment).  This shrinks a within-tolerance adversary's reach from
        # rtol*max|g| over the whole gradient to rtol*max|g_seg| per tensor —
        # orders of magnitude tighter for late/small layers — at the cost of a
        # slightly larger (still tiny) stats allreduce.
        assert granularity in ("row", "segment"), granularity
        self.granularity = granularity
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
        # member_mask[g, i] False = forfeited member (its host rank died; the vote
        # proceeds over the remaining members — the repetition code's erasure form)
        if member_mask is None:
            member_mask = np.ones((self.G, self.r), dtype=bool)
        self.member_mask = np.asarray(member_mask, dtype=bool)
        pairs_a, pairs_b, pg, pi, pj = [], [], [], [], []
        for g, rows in enumerate(self.member_rows):
            for i in range(self.r):
                for j in range(i + 1, self.r):
                    if self.member_mask[g, i] and self.member_mask[g, j]:
                        pairs_a.append(rows[i])
                        pairs_b.append(rows[j])
                        pg.append(g)
                        pi.append(i)
                        pj.append(j)
        self.pairs_a = torch.tensor(pairs_a, dtype=torch.int64, device=space.device)
        self.pairs_b = torch.tensor(pairs_b, dtype=torch.int64, device=space.device)
        # device-side winner selection state (no host round-trip in aggregate():
        # at N=8 a per-step .to("cpu") serialises all ranks on one readback)
        dev = space.device
        self._pg = torch.tensor(pg, dtype=torch.int64, device=dev)
        self._pi = torch.tensor(pi, dtype=torch.int64, device=dev)
        self._pj = torch.tensor(pj, dtype=torch.int64, device=dev)
        eye = torch.eye(self.r, device=dev).expand(self.G, self.r, self.r).clone()
        maskt = torch.tensor(self.member_mask, device=dev)
        eye *= maskt[:, :, None].float()  # forfeited members have empty classes
        self._eye = eye
        self._eqm = torch.empty_like(self._eye)
        # forfeited members can never win the argmax
        self._class_bias = torch.where(maskt, 0.0, -1e9)
        self._alive_count = maskt.sum(dim=1).float()  # (G,)
        self._member_rows_t = torch.tensor(self.member_rows, dtype=torch.int64, device=dev)
        # telemetry: steps where some group saw NO equal pair (tolerance too tight /
        # replicas diverged -> the vote degenerates to an arbitrary pick, which an
        # adversary can exploit).  Accumulated ON DEVICE; read via degenerate_steps.
        self._deg_counter = torch.zeros((), dtype=torch.int64, device=dev)

    @property
    def degenerate_steps(self) -> int:
        return int(self._deg_counter.item())

    @classmethod
    def from_member_rows(cls, comm, space, member_rows, atol: float = 0.0, rtol: float = 0.0):
        return cls(comm, space, group_size=member_rows.shape[1], atol=atol, rtol=rtol,
                   member_rows=member_rows)

    def aggregate(self, payload: torch.Tensor, step: int) -> torch.Tensor:
        recv = self.exchanged(payload)  # (r*world, shard)
        if self.rtol > 0.0 and self.granularity == "segment":
            npairs = self.pairs_a.numel()
            segdiff = ops.segment_pair_maxdiff(recv, self.pairs_a, self.pairs_b,
                                               self.local_seg)  # (n_pairs, L)
            segmax = ops.segment_absmax(recv, self.local_seg)  # (rows, L)
            stats = torch.cat([segdiff.reshape(-1), segmax.reshape(-1)])
            self.comm.all_reduce(stats, op="max")  # per-segment global maxima
            segdiff = stats[: segdiff.numel()].view(npairs, -1)
            segmax = stats[segdiff.numel():].view(recv.shape[0], -1)
            thresh = self.atol + self.rtol * torch.maximum(segmax[self.pairs_a],
                                                           segmax[self.pairs_b])
            eq = (segdiff <= thresh).all(dim=1)
        elif self.rtol > 0.0:
            maxdiff = ops.pair_maxdiff(recv, self.pairs_a, self.pairs_b)  # (n_pairs,)
            rowmax = ops.row_absmax(recv)  # (world*r,)
            stats = torch.cat([maxdiff, rowmax])
            self.comm.all_reduce(stats, op="max")  # full-gradient maxima
            maxdiff, rowmax = stats[: len(maxdiff)], stats[len(maxdiff):]
            thresh = self.atol + self.rtol * torch.maximum(rowmax[self.pairs_a], rowmax[self.pairs_b])
            eq = maxdiff <= thresh
        else:
            maxdiff = ops.pair_maxdiff(recv, self.pairs_a, self.pairs_b)
            self.comm.all_reduce(maxdiff, op="max")
            eq = maxdiff <= self.atol
        # winner selection ON DEVICE (every rank computes the identical winners from
        # the identical allreduced bits — zero host round-trips in the decode):
        # the winner of group g is a member of its largest equality class.  When a
        # true majority class exists (the coded guarantee: > r/2 honest members)
        # this is exactly the reference's Boyer-Moore result (rep_master.py:154-168);
        # with no majority both picks are arbitrary (the reference takes the last
        # surviving candidate, we take the plurality class — strictly no worse).
        eqf = eq.float()
        eqm = self._eqm
        eqm.copy_(self._eye)
        eqm[self._pg, self._pi, self._pj] = eqf
        eqm[self._pg, self._pj, self._pi] = eqf
        class_size = eqm.sum(dim=2)  # (G, r); forfeited members contribute/score 0
        winner_member = (class_size + self._class_bias).argmax(dim=1)  # deterministic tie-break
        winners = self._member_rows_t.gather(1, winner_member.unsqueeze(1)).squeeze(1)
        if self.r > 1:
            # degenerate: a group with >= 2 alive members and NO equal pair
            deg = (self._alive_count >= 2) & (class_size.max(dim=1).values <= 1.5)
            self._deg_counter += deg.any().to(torch.int64)
        ops.mean_rows(recv, winners, self._shard_out)
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

    def __init__(self, comm, space, num_workers: int, max_iter: int = 80, tol: float = 1e-7,
                 gpu_iters: int = 24):
        super().__init__(comm, space)
        self.num_workers = num_workers
        self.max_iter = max_iter
        self.tol = tol
        # On GPU: a FIXED iteration count with no convergence readback — each
        # early-exit check is a full host round-trip that serialises all ranks
        # (Weiszfeld converges geometrically; 24 iterations measured ample for
        # P <= 64 points, and the CPU early-exit lane cross-checks the fixpoint).
        self.gpu_iters = gpu_iters

    def aggregate(self, payload: torch.Tensor, step: int) -> torch.Tensor:
        recv = self.exchanged(payload)  # (P, shard)
        P = recv.shape[0]
        on_gpu = recv.is_cuda
        n_iter = self.gpu_iters if on_gpu else self.max_iter
        z = recv.mean(dim=0)  # init at the mean (hdmedians does the same)
        # zeros, not empty: the pad tail past the last segment is never written by
        # segment_weighted_mean, and garbage there (inf) poisons the delta criterion
        z_new = torch.zeros_like(z)
        for it in range(n_iter):
            part = ops.segment_sqdist(recv, z, self.local_seg)  # (P, L)
            self.comm.all_reduce(part)
            dist = part.clamp_min(1e-24).sqrt()
            w = 1.0 / dist
            w = w / w.sum(dim=0, keepdim=True)
            ops.segment_weighted_mean(recv, w, self.local_seg, z_new)
            z, z_new = z_new, z
            # convergence check every 8 iterations (CPU lane only: one fused 2-word
            # allreduce + host sync instead of three per iteration)
            if not on_gpu and ((it & 7) == 7 or it == n_iter - 1):
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
        if num_workers > 32:
            # CPU-side config check: the k_seg_gram LDS tile is sized for P <= 32
            # (draco_kernels.hip GRAM_CHUNK layout); fail at construction, not at
            # first kernel launch on the GPU
            raise ValueError(
                f"KrumAggregator supports at most 32 workers (got {num_workers}): "
                "the segment-Gram HIP kernel stages a (P, chunk) LDS tile sized for P <= 32")
        # per-column segment id of the local shard (for device-side winner assembly:
        # out[c] = recv[winner[seg_of(c)], c]); zero-length padding tail maps to 0
        # and is zeroed after the gather
        seg = self.local_seg
        lengths = (seg[1:] - seg[:-1]).clamp(min=0)
        dev = space.device
        self._col_seg = torch.repeat_interleave(
            torch.arange(len(lengths), dtype=torch.int64, device=dev), lengths.to(dev))
        self._tail = int(seg[-1])

    def aggregate(self, payload: torch.Tensor, step: int) -> torch.Tensor:
        recv = self.exchanged(payload)  # (P, shard)
        P = recv.shape[0]
        gram = ops.segment_gram(recv, self.local_seg)  # (L, P, P)
        self.comm.all_reduce(gram)
        # selection entirely ON DEVICE (fp64 — CDNA4 fp64 is strong and the matrix
        # is tiny; identical allreduced Gram -> identical winners on every rank)
        g = gram.double()
        diag = g.diagonal(dim1=1, dim2=2)  # (L, P)
        d2 = (diag.unsqueeze(2) + diag.unsqueeze(1) - 2.0 * g).clamp_(min=0.0)
        eye = torch.eye(P, dtype=torch.bool, device=d2.device)
        d2.masked_fill_(eye, float("inf"))
        keep = max(self.num_workers - self.s - 2, 1)
        vals, _ = torch.sort(d2, dim=2)
        scores = vals[:, :, :keep].sum(dim=2)  # (L, P)
        winners = scores.argmin(dim=1)  # (L,)
        out = self._shard_out
        if self._tail > 0:
            col_rows = winners[self._col_seg]  # (tail,)
            torch.gather(recv, 0, col_rows.unsqueeze(0), out=out[: self._tail].unsqueeze(0))
        if self._tail < out.shape[0]:
            out[self._tail:] = 0.0
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
                 comm_dtype=torch.float32, world0: int | None = None, alive=None):
        super().__init__(comm, space, comm_dtype)
        self.code = code
        self.L = workers_per_rank
        self.n = code.n
        # Logical worker w = l*world0 + src0 where world0 is the ORIGINAL world size
        # (the code is built for n = L*world0 and never changes).  `alive` lists the
        # surviving original ranks; workers hosted on dead ranks are permanent
        # ERASURES — known-bad rows the decode removes (<= s of them total with the
        # step's adversaries).  Recv rows are in survivor-world coordinates:
        # worker w's (re, im) payload rows (2l, 2l+1) land at (2l+p)*W' + pos(src0).
        world0 = world0 if world0 is not None else max(comm.world, 1)
        alive = list(alive) if alive is not None else list(range(world0))
        pos = {rk: i for i, rk in enumerate(alive)}
        Wp = max(len(alive), 1)
        w_ids = np.arange(self.n)
        l, src = w_ids // world0, w_ids % world0
        self.alive_w = np.array([int(s) in pos for s in src])
        posv = np.array([pos.get(int(s), 0) for s in src])
        self.rows_re = (2 * l) * Wp + posv
        self.rows_im = (2 * l + 1) * Wp + posv
        self.erased = frozenset(int(w) for w in w_ids[~self.alive_w])
        aw = w_ids[self.alive_w]
        self._alive_rows = torch.tensor(
            np.concatenate([self.rows_re[aw], self.rows_im[aw]]),
            dtype=torch.int64, device=space.device)
        self._alive_idx = aw
        self._z = None
        self._zgen = None

    def aggregate(self, payload_planes: torch.Tensor, step: int,
                  erasures: frozenset = frozenset()) -> torch.Tensor:
        recv = self.exchanged(payload_planes)  # (2L*world', shard)
        # random projection generated ON DEVICE (a shard-sized CPU randn + H2D copy
        # costs ~30 ms at d=11M — it dominated the whole decode); per-rank streams
        # may legitimately use different z (partial projections are summed)
        if self._z is None:
            self._z = torch.empty(self.space.shard, dtype=torch.float32, device=self.space.device)
            self._zgen = torch.Generator(device=self.space.device)
        self._zgen.manual_seed(0x5EED ^ (step * 1000003) ^ self.comm.rank)
        z = self._z.normal_(mean=1.0, std=1.0, generator=self._zgen)
        proj = ops.cyclic_project(recv, z)  # (2L*world',) per-row partial dots
        self.comm.all_reduce(proj)
        pa = proj.to("cpu").numpy().astype(np.float64)
        proj_complex = np.where(self.alive_w, pa[self.rows_re] + 1j * pa[self.rows_im], 0.0)
        known_bad = frozenset(erasures) | self.erased
        syndrome = self.code.W_perp @ proj_complex
        scale = float(np.abs(proj_complex).max())
        if not known_bad and float(np.abs(syndrome).max()) <= 1e-7 * max(scale, 1e-30):
            healthy = np.arange(self.n)
        else:
            healthy = self.code.locate_errors(syndrome, known_bad=known_bad)
        v = self.code.recombination_vector(healthy)
        # Re(v @ R) = sum_w vre[w]*re_row[w] - vim[w]*im_row[w]: one combine kernel
        # over the ALIVE workers' rows (v is zero on erased/adversarial rows anyway)
        va = v[self._alive_idx]
        w = torch.tensor(np.concatenate([np.real(va), -np.imag(va)]) / self.n,
                         dtype=torch.float32, device=recv.device)
        ops.combine_rows(recv, self._alive_rows, w, self._shard_out)
        self.comm.all_gather_shard(self._shard_out, self._out)
        return self._out
