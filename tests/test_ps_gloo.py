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


def _ps_two_groups_worker(rank, world):
    """PS maj_vote with TWO groups (P=6, r=3) + adversaries: master's weights after a
    few steps must match an adversary-free run (vote removes the corruption)."""
    import torch

    from draco_amd.config import Config
    from draco_amd.parallel.ps import Master, Worker

    out = {}
    for fail in (1, 0):
        cfg = Config(network="FC", dataset="MNIST", batch_size=4, device="cpu", lr=0.05,
                     topology="ps", approach="maj_vote", group_size=3, worker_fail=fail,
                     err_mode="rev_grad", max_steps=50, eval_freq=0, log_dir="",
                     train_dir="/tmp/draco_2g")
        role = Master(cfg) if rank == 0 else Worker(cfg)
        role.run(max_steps=3)
        if rank == 0:
            out[fail] = role.space.flat_param.clone()
    if rank == 0:
        diff = float((out[0] - out[1]).abs().max())
        assert diff < 1e-6, f"vote failed to remove adversary: {diff}"
        return diff
    return None


def test_ps_two_groups_vote_removes_adversary():
    run_dist(_ps_two_groups_worker, 7, timeout=300)


def _ps_cyclic_clean_equiv_worker(rank, world):
    """PS cyclic decode with s=1 adversaries reproduces the adversary-free weights."""
    import torch

    from draco_amd.config import Config
    from draco_amd.parallel.ps import Master, Worker

    out = {}
    for err in ("rev_grad", "none"):
        cfg = Config(network="FC", dataset="MNIST", batch_size=4, device="cpu", lr=0.05,
                     topology="ps", approach="cyclic", mode="cyclic", worker_fail=1,
                     err_mode=err, max_steps=50, eval_freq=0, log_dir="",
                     train_dir="/tmp/draco_ce")
        role = Master(cfg) if rank == 0 else Worker(cfg)
        role.run(max_steps=3)
        if rank == 0:
            out[err] = role.space.flat_param.clone()
    if rank == 0:
        diff = float((out["none"] - out["rev_grad"]).abs().max())
        scale = float(out["none"].abs().max())
        assert diff < 1e-4 * max(scale, 1.0), f"cyclic decode drifted under attack: {diff}"
    return None


def test_ps_cyclic_decode_equals_clean():
    run_dist(_ps_cyclic_clean_equiv_worker, 5, timeout=300)
