"""Colocated trainer: every rank computes gradients AND participates in the sharded
robust decode.  This replaces the reference's dedicated parameter-server process
(/root/reference/src/master/*, worker/*) — on an 8-GPU xGMI node a pure PS GPU would
idle 1/8 of the compute and bottleneck all gradient traffic on one GPU's links, so
the decode is sharded across all ranks instead (parallel/aggregators.py).  A
reference-parity PS topology lives in parallel/ps.py.

Logical-worker layout per approach (P = total logical workers):
  baseline   L=1 per rank, P=world; every worker draws its own batch.
  maj_vote   L=r per rank, G=world groups; member i of group g is slot i of rank
             (g+i)%world, so group members sit on different GPUs (the vote actually
             crosses the wire) and per-GPU work is r forward/backward per step.
  cyclic     L=n/world workers per rank (n=P); each computes the 2s+1 sub-batch
             gradients of its cyclic band and ships one encoded complex gradient.
"""
from __future__ import annotations

import math
import os
import time

import numpy as np
import torch
import torch.nn.functional as F

from .. import ops
from ..coding import AdversarySchedule, build_cyclic_code
from ..config import Config
from ..data import GlobalBatchSource, GroupBatchSource, SyntheticClassification
from ..models import build_model
from ..optim import FlatSGD
from ..utils.checkpoint import load_checkpoint, save_checkpoint
from ..utils.logging import MetricsLogger
from .aggregators import (
    CyclicAggregator,
    GeoMedianAggregator,
    KrumAggregator,
    MeanAggregator,
    VoteAggregator,
)
from .comm import Communicator
from .flat import FlatSpace


def _resolve_device(cfg: Config, rank: int) -> torch.device:
    if cfg.device == "cpu":
        return torch.device("cpu")
    if cfg.device == "cuda" or (cfg.device == "auto" and torch.cuda.is_available()):
        local = int(os.environ.get("LOCAL_RANK", rank % max(torch.cuda.device_count(), 1)))
        torch.cuda.set_device(local)
        return torch.device("cuda", local)
    return torch.device("cpu")


class Trainer:
    def __init__(self, cfg: Config, comm: Communicator | None = None):
        cfg.sanity()
        self.cfg = cfg
        rank = int(os.environ.get("RANK", 0))
        device = _resolve_device(cfg, rank)
        self.comm = comm or Communicator.from_env(device)
        self.device = device
        self.rank = self.comm.rank
        self.world = self.comm.world

        if cfg.deterministic:
            torch.backends.cudnn.deterministic = True
            torch.backends.cudnn.benchmark = False
        else:
            torch.backends.cudnn.benchmark = True

        # identical init on every rank
        torch.manual_seed(cfg.seed)
        self.model = build_model(cfg.network, cfg.dataset).to(device)
        self.model.train()
        if cfg.compile and device.type == "cuda":
            import torch._dynamo as _dynamo

            _dynamo.config.suppress_errors = True  # fall back to eager per-op
            self.model = torch.compile(self.model)
        self.use_cl = cfg.channels_last and device.type == "cuda"
        will_graph = bool(cfg.hip_graphs) and device.type == "cuda" and not cfg.deterministic
        if self.use_cl and will_graph and cfg.dtype == "bf16" and cfg.dataset == "ImageNetSynthetic":
            # Measured interaction on this stack (tools/diag_cfg5_isolate.py): a
            # bf16 NHWC conv kernel captured for the 224x224 ResNet-50 shapes starts
            # producing garbage after ~50 graph REPLAYS (eager NHWC clean, graph
            # NCHW clean, graph fp32-NHWC clean, ResNet-18 NHWC graphs clean over
            # 1000+ steps).  Until the kernel-level culprit is fixed upstream, the
            # 224^2 bf16 graph path runs NCHW.
            import warnings

            warnings.warn("bf16+channels_last+hipGraphs is unstable for 224x224 shapes "
                          "on this stack; forcing channels_last=False (see KNOWN_ISSUES.md)")
            self.use_cl = False
        self.space = FlatSpace(self.model, self.world, device, channels_last=self.use_cl)
        if cfg.optimizer == "adam":
            from ..optim import FlatAdam

            self.opt = FlatAdam(self.space.flat_param, lr=cfg.lr)
        else:
            self.opt = FlatSGD(self.space.flat_param, lr=cfg.lr, momentum=cfg.momentum)

        self.autocast_dtype = torch.bfloat16 if (cfg.dtype == "bf16" and device.type == "cuda") else None

        # ------------------ failure detection (colocated topology) ---------------
        # Heartbeats + pre-created survivor subgroups: a dead rank's logical workers
        # become erasures and training continues (parallel/health.py; the reference
        # hangs forever, baseline_master.py:112-116).  Worker identity below is
        # defined by the ORIGINAL (rank, world) and never changes after a failure;
        # only the comm-layer layout (self.comm.rank/world) is rebuilt.
        self.health = None
        if cfg.health_timeout > 0 and self.comm.distributed:
            from .health import HealthMonitor

            self.health = HealthMonitor(self.rank, self.world, cfg.health_timeout)

        # ---------------- logical workers & aggregator ----------------
        approach = cfg.approach
        self.approach = approach
        if approach == "baseline":
            self.L = 1
            self.P = self.world
            self.data = GroupBatchSource(self._dataset(), cfg.batch_size, n_groups=self.P)
        elif approach == "maj_vote":
            self.r = cfg.group_size
            self.L = self.r
            self.G = self.world
            self.P = self.L * self.world
            rtol = cfg.vote_rtol
            if rtol < 0:  # auto: bitwise on CPU, tolerance on GPU (see VoteAggregator doc)
                if device.type != "cuda":
                    rtol = 0.0  # CPU autograd is reproducible; identical fp32 rows
                    # cast to identical bf16 rows, so a bf16 wire stays bitwise too
                elif cfg.dtype == "bf16":
                    # bf16 autocast: replica noise measured at 1.2-3.5% of the row max
                    # under MIOpen algo/order variation (tools/diag_det.py); 0.2 keeps
                    # >5x margin while a rev_grad adversary is ~500x outside
                    rtol = 2e-1
                elif self._comm_dtype() == torch.bfloat16:
                    # fp32 compute but bf16 wire: GPU fp-reorder noise can push
                    # honest replicas across bf16 rounding boundaries (1 ulp =
                    # 2^-8 relative), far above the fp32 reorder level
                    rtol = 1e-2
                else:
                    rtol = 1e-4  # fp32 reorder noise
            self.vote_rtol = rtol
            self.data = GroupBatchSource(self._dataset(), cfg.batch_size, n_groups=self.G)
        elif approach == "cyclic":
            wpr = cfg.workers_per_rank or max(1, math.ceil((2 * cfg.worker_fail + 2) / self.world))
            self.L = wpr
            self.n = self.L * self.world
            self.P = self.n
            self.code = build_cyclic_code(self.n, cfg.worker_fail)
            self.s_hat = self.code.s_hat
            self.data = GlobalBatchSource(self._dataset(), cfg.batch_size, n_workers=self.n)
            W = self.code.W
            sup = self.code.support
            # Sub-batch dedup within a rank: logical workers hosted on the SAME rank
            # have overlapping cyclic bands; their shared sub-batch gradients are
            # computed ONCE (the reference recomputes per worker because its workers
            # are separate machines — same-host recompute adds no fault isolation,
            # and all transmitted encodings are identical either way).  At N=1 this
            # is n fwd/bwd per step instead of n*(2s+1); at N=world=n (L=1) it is
            # the usual 2s+1.
            local_w = [l * self.world + self.rank for l in range(self.L)]
            needed = sorted({int(j) for w in local_w for j in sup[w]})
            self._local_subs = needed
            row_of = {j: i for i, j in enumerate(needed)}
            self._w_re, self._w_im, self._enc_rows = [], [], []
            for l, w_global in enumerate(local_w):
                coeff = W[w_global, sup[w_global]]
                self._w_re.append(torch.tensor(np.real(coeff), dtype=torch.float32, device=device))
                self._w_im.append(torch.tensor(np.imag(coeff), dtype=torch.float32, device=device))
                self._enc_rows.append(torch.tensor([row_of[int(j)] for j in sup[w_global]],
                                                   dtype=torch.int64, device=device))
        else:
            raise ValueError(f"unknown approach {approach!r}")
        self._setup_decode()

        self.use_graphs = bool(cfg.hip_graphs) and device.type == "cuda" and not cfg.deterministic
        self._graphs = {}
        self._streams = []
        # ---------------- per-layer (bucketed) comm/compute overlap --------------
        # Eager path only: python hooks do not fire inside a hipGraph replay, and
        # the graph path already overlaps across logical workers' streams.  With
        # L=1 per rank (baseline at any N, maj_vote/cyclic at N=P) this is the only
        # mechanism that starts the gradient exchange BEFORE backward finishes —
        # the reference's signature per-layer interleaving (lenet.py:114-218,
        # resnet_split.py:431-623), rebuilt as post-accumulate-grad hooks.
        self.use_buckets = (cfg.bucket_mb > 0 and not self.use_graphs
                            and approach in ("baseline", "maj_vote", "cyclic")
                            and self.comm.distributed
                            and hasattr(self.agg, "start_bucket"))
        if self.use_buckets:
            self._buckets = self.space.build_buckets(cfg.bucket_mb)
            self._param_bucket = {}
            for bi, (_, _, idxs) in enumerate(self._buckets):
                for i in idxs:
                    self._param_bucket[i] = bi
            self._bucket_left: list = []
            self._hook_row = None
            self._hook_adversary = False
            for i, p in enumerate(self.space.params):
                p.register_post_accumulate_grad_hook(self._make_bucket_hook(i))
        self.step_sync = True     # per-step device sync (bench may disable; driver
                                  # brackets with its own barrier+synchronize)
        self.collect_loss = True  # read losses to host each step
        self.n_fail = min(cfg.worker_fail, self.P)
        self.schedule = AdversarySchedule(self.P, self.n_fail, cfg.max_steps)
        self.step_num = 0
        self._skip_counter = torch.zeros((), dtype=torch.int64, device=device)
        self.logger = MetricsLogger(cfg.log_dir, self.rank)
        self.criterion = F.cross_entropy

        if cfg.checkpoint_step > 0:
            self.load(cfg.checkpoint_step)

    # ------------------------------------------------------------------ helpers
    @property
    def skipped_updates(self) -> int:
        return int(self._skip_counter.item())

    def _event(self):
        if self.device.type != "cuda" or not self.cfg.gpu_timing:
            return None
        ev = torch.cuda.Event(enable_timing=True)
        ev.record()
        return ev

    def _comm_dtype(self):
        # "compress" = reference CLI alias (README.md:118, blosc) -> bf16 wire dtype
        return torch.bfloat16 if self.cfg.compress_grad in ("bf16", "compress") else torch.float32

    def _dataset(self, train: bool = True):
        from ..data.real import RealClassification, dataset_available

        if dataset_available(self.cfg.dataset, self.cfg.data_root):
            return RealClassification(self.cfg.dataset, self.cfg.data_root, self.device,
                                      train=train, augment=train)
        # held-out synthetic stream: same task function, disjoint index space
        # (the 2**40 offset in evaluate(); splitmix64 seeding cannot collide)
        return SyntheticClassification(self.cfg.dataset, self.device, seed=1234,
                                       task=self.cfg.synthetic_task)

    def _baseline_aggregator(self, cfg: Config, num_workers: int):
        if cfg.mode == "normal":
            return MeanAggregator(self.comm, self.space, num_workers=num_workers)
        if cfg.mode == "geometric_median":
            return GeoMedianAggregator(self.comm, self.space, num_workers=num_workers)
        if cfg.mode == "krum":
            return KrumAggregator(self.comm, self.space, num_workers=num_workers, s=cfg.worker_fail)
        raise ValueError(f"baseline approach supports modes normal/geometric_median/krum, got {cfg.mode!r}")

    def _setup_decode(self) -> None:
        """Build the aggregator + comm-layout buffers for the CURRENT communicator
        (called at init, and again with the survivor communicator after a rank
        failure — worker identity keys off the ORIGINAL self.rank/self.world)."""
        cfg = self.cfg
        alive = self.health.alive if self.health is not None else list(range(self.world))
        if self.approach == "baseline":
            self.agg = self._baseline_aggregator(cfg, num_workers=len(alive))
            self.payload = self.space.alloc_payload(self.L)
        elif self.approach == "maj_vote":
            from ..coding import colocated_member_rows

            rows, mask = colocated_member_rows(self.G, self.r, self.world, alive)
            self.agg = VoteAggregator(self.comm, self.space, group_size=self.r,
                                      atol=cfg.vote_atol, rtol=self.vote_rtol,
                                      member_rows=rows, member_mask=mask,
                                      comm_dtype=self._comm_dtype(),
                                      granularity=cfg.vote_granularity)
            self.payload = self.space.alloc_payload(self.L)
        else:  # cyclic
            self.agg = CyclicAggregator(self.comm, self.space, self.code, self.L,
                                        comm_dtype=self._comm_dtype(),
                                        world0=self.world, alive=alive)
            self.payload = self.space.alloc_payload(self.L * 2)
            self.scratch = self.space.alloc_payload(len(self._local_subs))

    def _handle_failure(self, dead: list) -> None:
        """Commit a rank failure: switch to the survivor subgroup, re-shard the flat
        space, rebuild the decode with the dead rank's workers as erasures, and
        adopt a canonical survivor's (params, optimizer, step) — survivors may have
        stopped on different sides of the failed step's update, so state is
        broadcast from the lowest surviving rank rather than assumed equal.

        Edge semantics: if the failure surfaced in a collective AFTER the step's
        optimizer update (e.g. the checkpoint-boundary sync_buffers), the retry
        re-runs the whole step on the adopted post-update state — that step's
        update lands twice and one step number is skipped.  Consistent across
        survivors (all adopt the same canonical state) and benign for training;
        exactly-once step semantics under mid-collective failure would need an
        update journal, which a Byzantine-tolerant trainer does not require."""
        from ..utils.checkpoint import _flat_to_params, _params_to_flat

        group, alive = self.health.declare_dead(dead)
        self.logger.log({"step": self.step_num, "event": "rank_failure",
                         "dead": [int(r) for r in dead], "alive": alive})
        old_space = self.space
        stash = {}
        for name in ("buf", "exp_avg", "exp_avg_sq", "max_exp_avg_sq"):
            t = getattr(self.opt, name, None)
            if isinstance(t, torch.Tensor):
                stash[name] = _flat_to_params(old_space, t)
        deg = getattr(self.agg, "_deg_counter", None)
        self.comm = self.comm.subgroup_communicator(group, alive)
        self.space = FlatSpace(self.model, self.comm.world, self.device,
                               channels_last=self.use_cl)
        self.opt.param = self.space.flat_param
        for name, tensors in stash.items():
            new = torch.zeros_like(self.space.flat_param)
            _params_to_flat(self.space, tensors, new)
            setattr(self.opt, name, new)
        self._graphs = {}  # captured graphs point into the old payload/space
        if self.use_buckets:
            self._hook_row = None
            self._buckets = self.space.build_buckets(self.cfg.bucket_mb)
        self._setup_decode()
        if deg is not None:
            self.agg._deg_counter.copy_(deg.to(self.agg._deg_counter.device))
        # canonical-state adoption (lowest surviving rank wins)
        src = int(alive[0])
        self.comm.broadcast(self.space.flat_param, src=src)
        for name in stash:
            self.comm.broadcast(getattr(self.opt, name), src=src)
        bufs = [b for b in self.model.buffers() if b.dtype.is_floating_point]
        if bufs:
            flatb = torch.cat([b.reshape(-1).float() for b in bufs])
            self.comm.broadcast(flatb, src=src)
            off = 0
            with torch.no_grad():
                for b in bufs:
                    n = b.numel()
                    b.copy_(flatb[off : off + n].view(b.shape).to(b.dtype))
                    off += n
        meta = torch.tensor([self.step_num], dtype=torch.int64,
                            device=self.device if self.comm.backend == "nccl" else "cpu")
        self.comm.broadcast(meta, src=src)
        self.step_num = int(meta[0])

    def _forward_backward(self, x, y, grad_row: torch.Tensor):
        if self.use_cl and x.dim() == 4:
            x = x.to(memory_format=torch.channels_last)
        self.space.attach_grads(grad_row)
        grad_row.zero_()
        if self.autocast_dtype is not None:
            with torch.autocast("cuda", dtype=self.autocast_dtype):
                logits = self.model(x)
                loss = self.criterion(logits, y)
        else:
            logits = self.model(x)
            loss = self.criterion(logits, y)
        loss.backward()
        return loss.detach()

    # ------------------------------------------------------------ bucketed overlap
    def _make_bucket_hook(self, pidx: int):
        bi = self._param_bucket[pidx]

        def hook(_param):
            if self._hook_row is None:  # not inside a bucketed backward (e.g. eval)
                return
            self._bucket_left[bi] -= 1
            if self._bucket_left[bi] == 0:
                self._ship_bucket(bi)

        return hook

    def _ship_bucket(self, bi: int) -> None:
        if self._hook_row == "cyclic":
            self._ship_bucket_cyclic(bi)
            return
        lo, hi, _ = self._buckets[bi]
        row = self._hook_row
        if self._hook_adversary:
            # injection at the send boundary, per chunk — same placement as the
            # reference's per-layer err_simulation inside backward (lenet.py:129-141)
            self._inject_row(row, lo, hi)
        self.agg.start_bucket(self.payload, row, lo, hi)

    # Cyclic per-layer overlap: the encode needs ALL of a rank's sub-batch
    # gradients, so hooks are armed only during the LAST local sub-batch's
    # backward — when bucket b of that backward lands, bucket b of every earlier
    # sub-batch is long final, so the bucket's slice of each logical worker's
    # encoded planes is computed, injected, and shipped while the rest of the
    # last backward still runs.
    def _ship_bucket_cyclic(self, bi: int) -> None:
        lo, hi, _ = self._buckets[bi]
        n = hi - lo
        for l in range(self.L):
            w_global = l * self.world + self.rank
            enc = self.payload[2 * l : 2 * l + 2]
            ops.combine_rows_slice(self.scratch, self._enc_rows[l], self._w_re[l],
                                   enc[0, lo:hi], lo)
            ops.combine_rows_slice(self.scratch, self._enc_rows[l], self._w_im[l],
                                   enc[1, lo:hi], lo)
            if w_global in self._cyc_advs:
                self._inject_encoded_slice(enc, lo, hi, self.cfg.err_mode)
            self.agg.start_bucket(self.payload, 2 * l, lo, hi)
            self.agg.start_bucket(self.payload, 2 * l + 1, lo, hi)

    def _begin_cyclic_buckets(self, adversaries) -> None:
        self._bucket_left = [len(b[2]) for b in self._buckets]
        self._cyc_advs = adversaries
        self._hook_row = "cyclic"

    def _end_cyclic_buckets(self) -> None:
        for bi, left in enumerate(self._bucket_left):
            if left > 0:
                self._ship_bucket_cyclic(bi)
        for l in range(self.L):
            self.agg.mark_row_started(2 * l)
            self.agg.mark_row_started(2 * l + 1)
        self._hook_row = None

    def _inject_encoded_slice(self, enc: torch.Tensor, lo: int, hi: int, mode: str) -> None:
        s = enc[:, lo:hi]
        if mode == "rev_grad":
            s.add_(s, alpha=ops.fallback.ADVERSARY_)
        elif mode == "constant":
            enc[0, lo:hi].add_(ops.fallback.ADVERSARY_)
        elif mode in ("random", "none", ""):
            pass
        elif mode == "gauss":
            s.add_(torch.randn_like(s) * s.abs().mean().clamp(min=1e-12) * 100.0)
        else:
            raise ValueError(mode)

    def _inject_row(self, row: int, lo: int, hi: int) -> None:
        if self.cfg.err_mode == "within_tol":
            # worst-case TOLERANCE-RESPECTING adversary (security-model test): a
            # perturbation just inside the vote's equality ball — it passes the
            # vote every step and biases the winner by at most the ball radius.
            # Uses the adversary's own |g| maxima (<= the pairwise max in the
            # threshold, so the crafted row always stays inside the ball).
            g = self.payload[row, lo:hi]
            rtol = getattr(self, "vote_rtol", 0.0)
            scale = self.cfg.vote_atol + rtol * g.abs().max()
            gen = torch.Generator(device=g.device)
            gen.manual_seed(0xADD ^ (self.step_num * 7919) ^ row)
            sign = torch.randint(0, 2, g.shape, generator=gen, device=g.device,
                                 dtype=torch.int8).float() * 2.0 - 1.0
            g.add_(sign * (0.99 * scale))
        else:
            ops.inject_(self.payload[row, lo:hi], self.cfg.err_mode, cyclic=False)

    def _begin_bucketed_row(self, row: int, adversary: bool) -> None:
        self._bucket_left = [len(b[2]) for b in self._buckets]
        self._hook_row = row
        self._hook_adversary = adversary

    def _end_bucketed_row(self) -> None:
        # flush buckets whose params never fired (unused-in-forward params keep
        # zero grads; every rank has the identical model so flush order matches)
        for bi, left in enumerate(self._bucket_left):
            if left > 0:
                self._ship_bucket(bi)
        self.agg.mark_row_started(self._hook_row)
        self._hook_row = None

    # ------------------------------------------------------------ hipGraph capture
    # ResNet-18/CIFAR is launch-bound on MI355X (~1.4k kernel dispatches per bench
    # step measured); capturing each logical worker's fwd+bwd into a hipGraph replays
    # the whole thing as one launch.  The flat spaces make this natural: the payload
    # row IS a static buffer, so the captured backward accumulates straight into the
    # comm buffer; batch data is copied into static inputs before each replay.
    # Concurrent replay: the L logical workers' graphs are mutually independent
    # (private mempools, disjoint payload rows, shared READ-ONLY params), and a
    # 128-image CIFAR conv underfills 256 CUs — so each worker's graph replays on its
    # own HIP stream and the three fwd+bwd overlap on the chip.  BN running-stat
    # updates race across concurrent replicas; the race is benign (stats are
    # eval-only, approximately equal across replicas, and rank-averaged before
    # checkpoints).  NOTE: do NOT "fix" the race by freezing BN momentum in replica
    # graphs — under torch.compile a momentum change triggers a RECOMPILE, so the
    # replicas would run different kernels, their gradients would diverge beyond the
    # vote tolerance, and the majority vote breaks (observed: adversary winning 1/3
    # of votes).  Identical code across replicas is what the vote relies on.
    def _run_fwd_bwd(self, key, grad_row: torch.Tensor, x, y, stream=None):
        if not self.use_graphs:
            return self._forward_backward(x, y, grad_row)
        g = self._graphs.get(key)
        if g is None:
            try:
                g = self._capture(grad_row, x, y)
            except RuntimeError as e:  # pragma: no cover - capture unsupported
                import warnings

                warnings.warn(f"hipGraph capture failed ({e}); falling back to eager")
                self.use_graphs = False
                return self._forward_backward(x, y, grad_row)
            self._graphs[key] = g
        if stream is not None:
            stream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(stream):
                g["x"].copy_(x)
                g["y"].copy_(y)
                # x/y were allocated on the DEFAULT stream and their refs die when
                # this call returns; the caching allocator does not track cross-
                # stream consumers, so without record_stream their blocks can be
                # recycled and overwritten while the side-stream copy is still in
                # flight (intermittent garbage batches -> inf/NaN gradients)
                x.record_stream(stream)
                y.record_stream(stream)
                g["graph"].replay()
        else:
            g["x"].copy_(x)
            g["y"].copy_(y)
            g["graph"].replay()
        return g["loss"]

    def _worker_stream(self, idx: int):
        if not self.use_graphs:
            return None
        while len(self._streams) <= idx:
            self._streams.append(torch.cuda.Stream())
        return self._streams[idx]

    def _capture(self, grad_row: torch.Tensor, x, y):
        self.space.attach_grads(grad_row)
        static_x = x.clone()
        if self.use_cl and static_x.dim() == 4:
            # capture with channels_last inputs so MIOpen sees NHWC end-to-end
            # (per-replay copy_ converts the incoming contiguous batch)
            static_x = static_x.contiguous(memory_format=torch.channels_last)
        static_y = y.clone()

        def body():
            grad_row.zero_()
            if self.autocast_dtype is not None:
                with torch.autocast("cuda", dtype=self.autocast_dtype):
                    loss = self.criterion(self.model(static_x), static_y)
            else:
                loss = self.criterion(self.model(static_x), static_y)
            loss.backward()
            return loss

        # warm up on a side stream (MIOpen find, dynamo compile, autograd warmup)
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                body()
        torch.cuda.current_stream().wait_stream(s)

        # private mempool per graph: graphs sharing a pool may alias activation
        # memory, which forbids concurrent replay
        graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(graph):
            loss = body()
        return {"graph": graph, "x": static_x, "y": static_y, "loss": loss.detach()}

    # ------------------------------------------------------------------ one step
    def train_step(self) -> dict:
        if self.health is None:
            return self._train_step_inner()
        self.health.beat(self.step_num)
        dead = self.health.check()
        if dead:
            self._handle_failure(dead)
        try:
            return self._train_step_inner()
        except RuntimeError as err:
            # a peer dying mid-collective surfaces as a RuntimeError on the op;
            # confirm via heartbeats, commit the failure, re-run the step (the
            # canonical-state adoption in _handle_failure makes the retry exact
            # whichever side of the update each survivor stopped on)
            try:
                dead = self.health.wait_for_dead()
            except RuntimeError:
                raise err  # not a failure we recognise — propagate the original
            self._handle_failure(dead)
            return self._train_step_inner()

    def _train_step_inner(self) -> dict:
        cfg = self.cfg
        step = self.step_num
        t0 = time.perf_counter()
        self._ev_step = self._event()
        losses = []
        adversaries = self.schedule.adversaries_at(step) if self.n_fail > 0 else frozenset()

        if self.approach in ("baseline", "maj_vote"):
            pending = []
            for l in range(self.L):
                if self.approach == "baseline":
                    group = self.rank  # every worker draws its own stream
                    worker_id = self.rank
                else:
                    group = (self.rank - l) % self.world
                    worker_id = l * self.world + self.rank  # l-major global worker id
                x, y = self.data.batch_for(group, step)
                if self.use_buckets:
                    # per-layer overlap: hooks ship ~bucket_mb chunks of this row
                    # (inject-at-send included) while backward still computes
                    self._begin_bucketed_row(l, worker_id in adversaries)
                    losses.append(self._forward_backward(x, y, self.payload[l]))
                    self._end_bucketed_row()
                    continue
                st = self._worker_stream(l)
                losses.append(self._run_fwd_bwd(("slot", l), self.payload[l], x, y, stream=st))
                pending.append((l, worker_id, st))
            for l, worker_id, st in pending:
                if st is not None:  # order the default stream after worker l's graph
                    torch.cuda.current_stream().wait_stream(st)
                if worker_id in adversaries:
                    self._inject_row(l, 0, self.space.d_pad)
                # overlap: this row's all_to_all runs while other backwards compute
                self.agg.start_row(self.payload, l)
        else:  # cyclic
            # phase 1: every DISTINCT local sub-batch fwd/bwd replays concurrently
            streams = []
            last = len(self._local_subs) - 1
            for i, j in enumerate(self._local_subs):
                x, y = self.data.sub_batch(j, step)
                if self.use_buckets and i == last:
                    # per-layer overlap: bucketed encode+exchange fires from the
                    # last sub-batch's backward hooks (see _ship_bucket_cyclic)
                    self._begin_cyclic_buckets(adversaries)
                    losses.append(self._forward_backward(x, y, self.scratch[i]))
                    self._end_cyclic_buckets()
                    break
                st = self._worker_stream(i)
                losses.append(self._run_fwd_bwd(("sub", i), self.scratch[i], x, y, stream=st))
                streams.append(st)
            for st in streams:
                if st is not None:
                    torch.cuda.current_stream().wait_stream(st)
            if not self.use_buckets:
                # phase 2: per logical worker, encode its band (gathered rows),
                # inject, start the exchange
                for l in range(self.L):
                    w_global = l * self.world + self.rank
                    enc = self.payload[2 * l : 2 * l + 2]
                    ops.combine_rows(self.scratch, self._enc_rows[l], self._w_re[l], enc[0])
                    ops.combine_rows(self.scratch, self._enc_rows[l], self._w_im[l], enc[1])
                    if w_global in adversaries:
                        self._inject_encoded(enc, cfg.err_mode)
                    self.agg.start_row(self.payload, 2 * l)
                    self.agg.start_row(self.payload, 2 * l + 1)

        t_comp = time.perf_counter()
        ev_comp = self._event()
        grad = self.agg.aggregate(self.payload, step)
        ev_agg = self._event()
        t_agg = time.perf_counter()
        skipped = False
        if self.cfg.nan_guard:
            # failure detection: never apply a non-finite decoded gradient.  The
            # guard stays ON DEVICE (a 0-dim bool consumed by the fused optimizer
            # kernel) — no host round-trip on the hot path; the skip count
            # accumulates in a device counter read lazily via skipped_updates.
            finite = torch.isfinite(grad).all()
            self._skip_counter += (~finite).to(torch.int64)
            self.opt.step(grad, guard=finite)
            if self.collect_loss and not bool(finite):  # loss readback syncs anyway
                skipped = True
                self.logger.log({"step": step, "event": "nan_grad_skipped"})
        else:
            self.opt.step(grad)
        ev_end = self._event()
        if self.device.type == "cuda" and self.step_sync:
            torch.cuda.synchronize()
        t1 = time.perf_counter()

        self.step_num += 1
        if self.collect_loss:
            loss_val = float(sum(float(v) for v in losses) / len(losses))
        else:
            loss_val = None  # skip the device sync of reading losses (bench hot loop)
        rec = {
            "step": step,
            "loss": loss_val,
            "time": t1 - t0,
            "comp": t_comp - t0,
            "agg": t_agg - t_comp,
            "update": t1 - t_agg,
        }
        if skipped:
            rec["skipped_update"] = True
        if self._ev_step is not None and ev_end is not None:
            # device-accurate spans (host timers only measure launch time with
            # async collectives): step-start -> agg-start -> agg-end -> step-end
            rec["gpu_comp"] = self._ev_step.elapsed_time(ev_comp) / 1e3
            rec["gpu_agg"] = ev_comp.elapsed_time(ev_agg) / 1e3
            rec["gpu_update"] = ev_agg.elapsed_time(ev_end) / 1e3
        if cfg.eval_freq > 0 and self.step_num % cfg.eval_freq == 0:
            # collective on ALL ranks: save() runs sync_buffers (an all_reduce);
            # only the file write inside is rank-0-gated.  A rank-0-only call here
            # deadlocks world>1 (the other ranks enter the next step's collectives).
            self.save()
        self.logger.log(rec)
        return rec

    def _inject_encoded(self, enc: torch.Tensor, mode: str) -> None:
        # reference injects on the encoded complex combination with cyclic=True
        # (additive error, cyclic_worker.py:181-183 / model_ops/utils.py:8-11)
        if mode == "rev_grad":
            enc.add_(enc, alpha=ops.fallback.ADVERSARY_)
        elif mode == "constant":
            enc[0].add_(ops.fallback.ADVERSARY_)
        elif mode in ("random", "none", ""):
            pass
        elif mode == "gauss":
            enc.add_(torch.randn_like(enc) * enc.abs().mean().clamp(min=1e-12) * 100.0)
        else:
            raise ValueError(mode)

    # ------------------------------------------------------------------ eval / io
    @torch.no_grad()
    def evaluate(self, n_batches: int = 8) -> dict:
        """Held-out Prec@1/Prec@5 + loss (reference eval cadence,
        baseline_worker.py:148-155 / distributed_evaluator.py:92-110)."""
        self.model.eval()
        data = self._dataset(train=False)  # test split for real data (held out)
        correct = correct5 = total = 0
        loss_sum = 0.0
        for b in range(n_batches):
            # held-out stream: index space never touched by training
            x, y = data.get_batch(2**40 + b * self.cfg.test_batch_size, self.cfg.test_batch_size)
            logits = self.model(x)
            loss_sum += float(self.criterion(logits, y))
            k = min(5, logits.shape[1])
            topk = logits.topk(k, dim=1).indices
            correct += int((topk[:, 0] == y).sum())
            correct5 += int((topk == y[:, None]).any(dim=1).sum())
            total += y.numel()
        self.model.train()
        return {"prec1": correct / total, "prec5": correct5 / total, "loss": loss_sum / n_batches}

    def _ckpt_path(self, step: int | None = None) -> str:
        step = self.step_num if step is None else step
        return os.path.join(self.cfg.train_dir, f"model_step_{step}")

    def sync_buffers(self):
        """Average BN running stats across ranks (they drift: each rank normalises its
        own group batches; parameters stay bit-identical by construction)."""
        if not self.comm.distributed:
            return
        bufs = [b for b in self.model.buffers() if b.dtype.is_floating_point]
        if not bufs:
            return
        flat = torch.cat([b.reshape(-1).float() for b in bufs])
        self.comm.all_reduce(flat)
        flat /= self.comm.world  # survivors only, after a failure
        off = 0
        with torch.no_grad():
            for b in bufs:
                n = b.numel()
                b.copy_(flat[off : off + n].view(b.shape).to(b.dtype))
                off += n

    def save(self):
        self.sync_buffers()  # collective — every rank must call save() together
        if self.rank == 0:
            save_checkpoint(self._ckpt_path(), self.model, self.space, self.opt, self.step_num, self.cfg)

    def load(self, step: int):
        load_checkpoint(self._ckpt_path(step), self.model, self.space, self.opt)
        self.step_num = step

    def close(self):
        if self.health is not None:
            self.health.close()
        self.logger.close()
