"""PS-topology logic lane (BASELINE config 1): 1 PS + 3 workers over gloo on CPU,
LeNet/MNIST-shaped synthetic, repetition r=3, no adversary — plus an attacked run."""
import numpy as np
import pytest
import torch

from tests.dist_util import run_dist


def _ps_worker(rank, world, approach, kw, steps):
    from draco_amd.config import Config
    from draco_amd.parallel.ps import Master, Worker

    cfg = Config(network="LeNet", dataset="MNIST", batch_size=4, device="cpu", lr=0.05,
                 topology="ps", approach=approach, mode="maj_vote" if approach == "maj_vote" else "normal",
                 max_steps=50, eval_freq=0, log_dir="", train_dir="/tmp/draco_ps_ckpt", **kw)
    role = Master(cfg) if rank == 0 else Worker(cfg)
    role.run(max_steps=steps)
    if rank == 0:
        return float(role.space.flat_param.double().sum())
    return None


def test_ps_repetition_r3():
    # 1 PS + 3 workers, one group of size 3, s=1 adversary
    res = run_dist(_ps_worker, 4, "maj_vote", dict(group_size=3, worker_fail=1, err_mode="rev_grad"), 4)
    assert np.isfinite(res[0])


def test_ps_baseline_mean():
    res = run_dist(_ps_worker, 3, "baseline", dict(worker_fail=0), 4)
    assert np.isfinite(res[0])


def test_ps_cyclic():
    # 1 PS + 4 cyclic workers, s=1
    res = run_dist(_ps_worker, 5, "cyclic", dict(worker_fail=1, err_mode="rev_grad"), 3)
    assert np.isfinite(res[0])
