"""Reference-parity parameter-server topology: rank 0 = PS (no compute), ranks 1..P =
workers (one logical worker each; maj_vote groups are contiguous blocks of group_size
workers, as in /root/reference/src/util.py:90-97).

This mode reproduces the reference's process layout (SyncReplicasMaster_NN /
DistributedWorker, baseline_master.py:64-146 / baseline_worker.py:67-158) on
torch.distributed: the PS broadcasts ONE fused flat parameter buffer per step
(replacing the per-layer float64 MPI.Bcast, C2 in SURVEY §2.4), gathers one flat
gradient payload per worker over p2p (replacing the P×L tagged Irecvs, C4), decodes
with the same aggregator kernels (run unsharded on a world-1 communicator), and steps
the fused optimizer.  The colocated topology (trainer.py) is the fast path; this one
exists for CLI/behavior parity and the CPU/gloo logic lane (BASELINE config 1).
"""
from __future__ import annotations

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
from .trainer import _resolve_device


class _LocalComm:
    """World-1 stand-in so the sharded aggregators run unsharded on the PS."""

    rank = 0
    world = 1
    distributed = False

    def all_to_all_rows(self, payload):
        L, d = payload.shape
        return payload.view(L, d)

    def reduce_scatter_sum(self, x):
        return x

    def all_gather_shard(self, shard, out):
        out.copy_(shard)

    def all_reduce(self, t, op="sum"):
        return t


class _PSBase:
    def __init__(self, cfg: Config):
        cfg.sanity()
        self.cfg = cfg
        rank = int(os.environ.get("RANK", 0))
        device = _resolve_device(cfg, rank)
        self.comm = Communicator.from_env(device)
        self.device = device
        self.rank = self.comm.rank
        self.world = self.comm.world
        self.P = self.world - 1
        if self.P < 1:
            raise RuntimeError("ps topology needs world_size >= 2")
        if self.comm.backend == "nccl" and os.environ.get("DRACO_PS_NCCL_OK") != "1":
            # refuse loudly rather than fail subtly: the PS lane's straggler
            # timeout relies on host-blocking p2p waits (gloo semantics) and the
            # mode has not been validated on RCCL hardware.  The colocated
            # topology (trainer.py) is the GPU path.  Set DRACO_PS_NCCL_OK=1 to
            # run it anyway (plain broadcast+send/recv are RCCL-supported).
            raise RuntimeError(
                "topology=ps has not been validated on the nccl/RCCL backend; "
                "use topology=colocated on GPUs or set DRACO_PS_NCCL_OK=1")
        # backend-independent out-of-band signalling (mid-step preemption/abort):
        # tagged p2p is NOT honored by NCCL (tags are ignored -> mis-matched recvs,
        # not errors), so the abort channel lives in the rendezvous TCPStore
        import torch.distributed as dist

        self._store = dist.distributed_c10d._get_default_store()
        if self.rank == 0:
            self._store.set("ps_abort", "0")

        torch.manual_seed(cfg.seed)
        self.model = build_model(cfg.network, cfg.dataset).to(device)
        self.model.train()
        # world=1 sharding: flat vectors are whole on every rank in this topology
        self.space = FlatSpace(self.model, 1, device)
        self.criterion = F.cross_entropy

        self.approach = cfg.approach
        if self.approach == "cyclic":
            self.n = self.P
            self.code = build_cyclic_code(self.n, cfg.worker_fail)
            self.payload_rows = 2  # encoded planes per worker
        else:
            self.payload_rows = 1
        self.n_fail = min(cfg.worker_fail, self.P)
        self.schedule = AdversarySchedule(self.P, self.n_fail, cfg.max_steps)

    def _data(self):
        from ..data.real import RealClassification, dataset_available

        if dataset_available(self.cfg.dataset, self.cfg.data_root):
            return RealClassification(self.cfg.dataset, self.cfg.data_root, self.device)
        return SyntheticClassification(self.cfg.dataset, self.device, seed=1234)


class Master(_PSBase):
    def __init__(self, cfg: Config):
        super().__init__(cfg)
        assert self.rank == 0
        self.opt = FlatSGD(self.space.flat_param, lr=cfg.lr, momentum=cfg.momentum)
        local = _LocalComm()
        if cfg.approach == "baseline":
            if cfg.mode == "normal":
                self.agg = MeanAggregator(local, self.space, num_workers=self.P)
            elif cfg.mode == "geometric_median":
                self.agg = GeoMedianAggregator(local, self.space, num_workers=self.P)
            elif cfg.mode == "krum":
                self.agg = KrumAggregator(local, self.space, num_workers=self.P, s=cfg.worker_fail)
            else:
                raise ValueError(cfg.mode)
        elif cfg.approach == "maj_vote":
            r = cfg.group_size
            if self.P % r != 0:
                raise ValueError(f"ps maj_vote needs P % group_size == 0 (P={self.P}, r={r})")
            member_rows = np.arange(self.P).reshape(self.P // r, r)
            rtol = cfg.vote_rtol
            if rtol < 0:  # auto (no autocast in ps workers -> fp32 noise on GPU)
                rtol = 1e-4 if self.device.type == "cuda" else 0.0
            self.agg = VoteAggregator.from_member_rows(local, self.space, member_rows,
                                                       atol=cfg.vote_atol, rtol=rtol)
        elif cfg.approach == "cyclic":
            self.agg = CyclicAggregator(local, self.space, self.code, workers_per_rank=self.n)
        self.gather_buf = self.space.alloc_payload(self.P * self.payload_rows)
        self.logger = MetricsLogger(cfg.log_dir, self.rank)
        self.step_num = 0
        self._abort = False
        if cfg.checkpoint_step > 0:
            load_checkpoint(
                os.path.join(cfg.train_dir, f"model_step_{cfg.checkpoint_step}"),
                self.model, self.space, self.opt,
            )
            self.step_num = cfg.checkpoint_step

    def request_abort(self):
        """Preempt workers MID-STEP (the reference's tag-77 kill channel,
        lenet.py:237-240 / resnet_split.py:636-640 — which had no master-side sender;
        this is it).  Delivered through the TCPStore, not tagged p2p: NCCL silently
        ignores tags (a mis-matched recv, not an exception, would be the failure
        mode) and a pending p2p recv wedges gloo teardown.  Workers poll the flag
        between redundant sub-batches, ship what they have, and exit at the next
        step boundary."""
        if self._abort:
            return
        self._abort = True
        self._store.set("ps_abort", "1")

    def run(self, max_steps: int | None = None):
        cfg = self.cfg
        steps = max_steps or cfg.max_steps
        ctrl = torch.zeros(2, dtype=torch.int64,
                           device=self.device if self.comm.backend == "nccl" else "cpu")
        for _ in range(steps):
            if self._abort:
                break
            t0 = time.perf_counter()
            # step announce (reference tag-10 broadcast, baseline_master.py:156-162) +
            # abort flag (the reference's vestigial tag-77 kill channel, done properly)
            ctrl[0] = self.step_num
            ctrl[1] = 0
            self.comm.broadcast(ctrl, src=0)
            self.comm.broadcast(self.space.flat_param, src=0)
            erasures = self._gather_grads()
            t_gather = time.perf_counter()
            if erasures and self.approach == "cyclic":
                grad = self.agg.aggregate(self.gather_buf, self.step_num, erasures=erasures)
            else:
                grad = self.agg.aggregate(self.gather_buf, self.step_num)
                if erasures and cfg.approach == "baseline" and cfg.mode == "normal":
                    # missing rows were zeroed; rescale the mean to stay unbiased
                    grad = grad * (self.P / max(self.P - len(erasures), 1))
            t_agg = time.perf_counter()
            self.opt.step(grad)
            t1 = time.perf_counter()
            self.step_num += 1
            self.logger.log({
                "step": self.step_num - 1, "role": "master",
                "gather": t_gather - t0, "agg": t_agg - t_gather,
                "update": t1 - t_agg, "time": t1 - t0,
            })
            if cfg.eval_freq > 0 and self.step_num % cfg.eval_freq == 0:
                save_checkpoint(
                    os.path.join(cfg.train_dir, f"model_step_{self.step_num}"),
                    self.model, self.space, self.opt, self.step_num, cfg,
                )
        # tell workers to stop (clean shutdown; replaces the reference's missing
        # master-side kill sender, SURVEY §2.3 straggler row)
        ctrl[0] = self.step_num
        ctrl[1] = 1
        self.comm.broadcast(ctrl, src=0)
        if not self._abort:
            self.request_abort()  # store flag: lets worker poll threads exit
        self.logger.close()

    def _gather_grads(self) -> frozenset:
        """Gather one gradient payload per worker; with straggler_timeout > 0, give up
        on workers that have not arrived within the timeout after the first arrival
        and treat them as ERASURES (the cyclic decode tolerates them as known error
        locations; a missing vote member simply loses its group's vote).

        NOTE: the timeout path relies on blocking Work.wait() observed via waiter
        threads, which is host-blocking on gloo (the CPU logic lane); NCCL p2p waits
        are stream-ordered, so straggler_timeout is a gloo-lane feature.

        Protocol care: a timed-out worker's send still completes later (matched by the
        orphaned irecv).  The request is kept in self._stale[w]; at the next step the
        stale payload is DISCARDED and a fresh irecv posted, so the per-step message
        stream stays aligned.  A worker whose stale request never completes is treated
        as dead and stays erased (its irecv is never reposted)."""
        timeout = self.cfg.straggler_timeout
        if timeout <= 0:
            reqs = [
                self.comm.irecv(self.gather_buf[w * self.payload_rows : (w + 1) * self.payload_rows], src=w + 1)
                for w in range(self.P)
            ]
            for r in reqs:
                r.wait()
            return frozenset()
        # gloo Work.is_completed() never fires for p2p recvs without wait(), so
        # completion is observed through one waiter thread per outstanding request
        # (blocking wait() + Event; threads are daemonic and bounded by P).
        import threading

        if not hasattr(self, "_stale"):
            self._stale = {}
        events = {}
        dead = set()
        for w in range(self.P):
            buf = self.gather_buf[w * self.payload_rows : (w + 1) * self.payload_rows]
            old = self._stale.get(w)
            if old is not None:
                if old.is_set():
                    del self._stale[w]  # late (previous-step) payload arrived: discard
                else:
                    dead.add(w)  # still missing: remains erased, no new irecv
                    continue
            req = self.comm.irecv(buf, src=w + 1)
            ev = threading.Event()

            def waiter(r=req, e=ev):
                try:
                    r.wait()
                finally:
                    e.set()

            threading.Thread(target=waiter, daemon=True).start()
            events[w] = ev
        deadline = None
        pending = set(events)
        while pending:
            done = {w for w in pending if events[w].is_set()}
            pending -= done
            if not pending:
                break
            now = time.perf_counter()
            if deadline is None:
                if done or dead:
                    deadline = now + timeout
            elif now > deadline:
                break
            time.sleep(0.0005)
        for w in pending:
            self._stale[w] = events[w]
        missing = frozenset(pending | dead)
        for w in missing:
            self.gather_buf[w * self.payload_rows : (w + 1) * self.payload_rows].zero_()
        return missing


class Worker(_PSBase):
    def __init__(self, cfg: Config):
        super().__init__(cfg)
        assert self.rank >= 1
        self.worker_id = self.rank - 1
        if cfg.approach == "maj_vote":
            self.group = self.worker_id // cfg.group_size
            self.data = GroupBatchSource(self._data(), cfg.batch_size, n_groups=max(self.P // cfg.group_size, 1))
        elif cfg.approach == "cyclic":
            self.data = GlobalBatchSource(self._data(), cfg.batch_size, n_workers=self.n)
            coeff = self.code.W[self.worker_id, self.code.support[self.worker_id]]
            self._w_re = torch.tensor(np.real(coeff), dtype=torch.float32, device=self.device)
            self._w_im = torch.tensor(np.imag(coeff), dtype=torch.float32, device=self.device)
            self.scratch = self.space.alloc_payload(self.code.s_hat)
        else:
            self.group = self.worker_id
            self.data = GroupBatchSource(self._data(), cfg.batch_size, n_groups=self.P)
        self.payload = self.space.alloc_payload(self.payload_rows)
        self.logger = MetricsLogger(cfg.log_dir, self.rank)
        self.step_num = 0
        self._preempt = self._arm_preempt()

    def _arm_preempt(self):
        """Out-of-band preemption: a daemon thread polls the master's store-side
        abort flag (backend-independent — tagged p2p would be silently broken on
        NCCL) and sets a local event the compute loop checks between redundant
        sub-batches."""
        import threading

        ev = threading.Event()
        store = self._store

        def poller():
            while not ev.is_set():
                try:
                    if store.get("ps_abort").decode() == "1":
                        ev.set()
                        return
                except Exception:
                    return  # store gone: process is shutting down
                time.sleep(0.05)

        th = threading.Thread(target=poller, daemon=True)
        th.start()
        self._preempt_thread = th
        return ev

    def run(self, max_steps: int | None = None):
        cfg = self.cfg
        steps = max_steps or cfg.max_steps
        ctrl = torch.zeros(2, dtype=torch.int64,
                           device=self.device if self.comm.backend == "nccl" else "cpu")
        for _ in range(steps + 1):
            t0 = time.perf_counter()
            self.comm.broadcast(ctrl, src=0)
            if int(ctrl[1]) != 0:
                break  # master sent the kill signal
            self.step_num = int(ctrl[0])
            self.comm.broadcast(self.space.flat_param, src=0)
            adversaries = self.schedule.adversaries_at(self.step_num)
            if cfg.approach == "cyclic":
                sup = self.code.support[self.worker_id]
                losses = []
                for k in range(self.code.s_hat):
                    if self._preempt.is_set():
                        break  # preempted: ship what we have, exit at next ctrl
                    x, y = self.data.sub_batch(int(sup[k]), self.step_num)
                    losses.append(self._fwd_bwd(x, y, self.scratch[k]))
                ops.cyclic_encode(self.scratch, self._w_re, self._w_im, self.payload)
                if self.worker_id in adversaries:
                    _inject_encoded(self.payload, cfg.err_mode)
                loss = float(np.mean(losses)) if losses else float("nan")
            else:
                x, y = self.data.batch_for(self.group, self.step_num)
                loss = self._fwd_bwd(x, y, self.payload[0])
                if self.worker_id in adversaries:
                    ops.inject_(self.payload[0], cfg.err_mode, cyclic=False)
            t_comp = time.perf_counter()
            self.comm.send(self.payload, dst=0)
            t1 = time.perf_counter()
            self.logger.log({
                "step": self.step_num, "role": "worker", "loss": loss,
                "comp": t_comp - t0, "comm": t1 - t_comp, "time": t1 - t0,
            })
        # the master completes the tag-77 irecv at shutdown; join the waiter so the
        # recv is fully consumed before the process tears gloo down
        th = getattr(self, "_preempt_thread", None)
        if th is not None:
            th.join(timeout=30)
        self.logger.close()

    def _fwd_bwd(self, x, y, row):
        self.space.attach_grads(row)
        row.zero_()
        logits = self.model(x)
        loss = self.criterion(logits, y)
        loss.backward()
        return float(loss.detach())


def _inject_encoded(enc, mode):
    if mode == "rev_grad":
        enc.add_(enc, alpha=ops.fallback.ADVERSARY_)
    elif mode == "constant":
        enc[0].add_(ops.fallback.ADVERSARY_)
    elif mode in ("random", "none", ""):
        pass
    else:
        raise ValueError(mode)


def run_ps(cfg: Config, max_steps: int | None = None):
    rank = int(os.environ.get("RANK", 0))
    role = Master(cfg) if rank == 0 else Worker(cfg)
    role.run(max_steps)
    return role
