"""Rank-failure recovery in the colocated topology (gloo, CPU).

A killed rank's logical workers become erasures: the vote drops the forfeited
members, the cyclic decode removes the known-bad rows, and training continues on
the survivor subgroup with bit-identical params across survivors.  (The reference
hangs forever on a dead worker — baseline_master.py:112-116.)

Capability note: erasures spend the same redundancy budget as adversaries
(repetition: forfeits+adversaries < r/2 per group; cyclic: erased+adversarial
rows <= s), so these tests run with err_mode="none" — one dead rank IS the fault.
Rank 0 hosts the rendezvous TCPStore, so rank 0 death is unrecoverable by design;
tests kill a non-zero rank.
"""
import os
import time

import pytest
import torch

from tests.dist_util import run_dist

pytestmark = pytest.mark.timeout(300)


def _failure_worker(rank, world, approach, kw, die_rank, die_step):
    from draco_amd.config import Config
    from draco_amd.parallel.trainer import Trainer

    base = dict(network="FC", dataset="MNIST", batch_size=4, device="cpu", lr=0.05,
                approach=approach, err_mode="none", max_steps=100, eval_freq=0,
                log_dir="", train_dir="/tmp/draco_fail", health_timeout=1.5)
    cfg = Config(**{**base, **kw})
    t = Trainer(cfg)
    t.logger.stdout_every = 0
    losses = []
    for i in range(12):
        if rank == die_rank and i == die_step:
            # abrupt death mid-run: report to the harness, then hard-exit
            time.sleep(0.2)
            os._exit(0)
        losses.append(t.train_step()["loss"])
    h = float(t.space.flat_param.double().sum())
    alive = list(t.health.alive)
    t.close()
    return (losses, h, alive)


def _run_failure(world, approach, kw, die_rank=None):
    die_rank = world - 1 if die_rank is None else die_rank
    res = run_dist(_failure_worker, world, approach, kw, die_rank, 3,
                   timeout=240.0, expect_missing={die_rank})
    survivors = [r for r in range(world) if r != die_rank]
    hashes = [res[r][1] for r in survivors]
    assert all(h == hashes[0] for h in hashes), "survivor params diverged"
    for r in survivors:
        assert res[r][2] == survivors, f"rank {r} alive set wrong: {res[r][2]}"
        losses = res[r][0]
        assert len(losses) == 12
        assert losses[-1] < losses[0], "training did not keep converging after the failure"
    return res


def test_vote_survives_rank_death():
    _run_failure(3, "maj_vote", dict(mode="maj_vote", group_size=3, worker_fail=1))


def test_cyclic_survives_rank_death():
    # world=3, L=2 -> n=6 workers, s=2: killing one rank erases its 2 workers,
    # within the decode's known-bad budget (<= s)
    _run_failure(3, "cyclic", dict(mode="cyclic", worker_fail=2, workers_per_rank=2))


def test_baseline_mean_survives_rank_death():
    _run_failure(3, "baseline", dict(mode="normal", worker_fail=0))


def test_vote_survives_rank_death_with_many_buckets():
    """Rank dies while MANY per-bucket collectives are in flight (bucket_mb tiny):
    the retry must abandon the orphaned bucket works cleanly and keep training."""
    _run_failure(3, "maj_vote",
                 dict(mode="maj_vote", group_size=3, worker_fail=1, bucket_mb=0.01))


def _combined_worker(rank, world, die_rank, die_step):
    """Erasure AND adversary together: r=5 groups lose <=2 members to the dead
    rank while a rev_grad adversary keeps striking the survivors — each group
    still has an honest majority among its alive members, so the vote holds."""
    from draco_amd.config import Config
    from draco_amd.parallel.trainer import Trainer

    cfg = Config(network="FC", dataset="MNIST", batch_size=4, device="cpu", lr=0.05,
                 approach="maj_vote", mode="maj_vote", group_size=5, worker_fail=1,
                 err_mode="rev_grad", max_steps=100, eval_freq=0, log_dir="",
                 train_dir="/tmp/draco_fail2", health_timeout=1.5)
    t = Trainer(cfg)
    t.logger.stdout_every = 0
    losses = []
    for i in range(12):
        if rank == die_rank and i == die_step:
            time.sleep(0.2)
            os._exit(0)
        losses.append(t.train_step()["loss"])
    h = float(t.space.flat_param.double().sum())
    deg = t.agg.degenerate_steps
    t.close()
    return (losses, h, deg)


def test_vote_survives_rank_death_with_active_adversary():
    world, die_rank = 3, 2
    res = run_dist(_combined_worker, world, die_rank, 3, timeout=240.0,
                   expect_missing={die_rank})
    survivors = [0, 1]
    assert res[0][1] == res[1][1], "survivor params diverged"
    for r in survivors:
        losses = res[r][0]
        assert losses[-1] < losses[0], "did not keep converging"
        assert res[r][2] == 0, "vote degenerated under combined erasure+adversary"


def _adam_failure_worker(rank, world, die_rank, die_step):
    """Failure recovery must re-shard ALL optimizer state (Adam moments), not just
    params: survivors' flat buffers change size with the survivor world."""
    from draco_amd.config import Config
    from draco_amd.parallel.trainer import Trainer

    cfg = Config(network="FC", dataset="MNIST", batch_size=4, device="cpu", lr=0.005,
                 approach="maj_vote", mode="maj_vote", group_size=3, worker_fail=1,
                 err_mode="none", optimizer="adam", max_steps=100, eval_freq=0,
                 log_dir="", train_dir="/tmp/draco_fail_adam", health_timeout=1.5)
    t = Trainer(cfg)
    t.logger.stdout_every = 0
    losses = []
    for i in range(12):
        if rank == die_rank and i == die_step:
            time.sleep(0.2)
            os._exit(0)
        losses.append(t.train_step()["loss"])
    h = float(t.space.flat_param.double().sum())
    m = float(t.opt.exp_avg.double().sum())
    v = float(t.opt.exp_avg_sq.double().sum())
    t.close()
    return (losses, h, m, v)


def test_adam_state_survives_rank_death():
    res = run_dist(_adam_failure_worker, 3, 2, 3, timeout=240.0, expect_missing={2})
    for k in (1, 2, 3):  # params + both Adam moments identical across survivors
        assert res[0][k] == res[1][k], f"survivor state field {k} diverged"
    assert res[0][0][-1] < res[0][0][0]
