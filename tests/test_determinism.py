"""Deterministic-mode lane: the whole pipeline — data generation,
adversary schedule, encode/decode, fused optimizer — must be a pure function of the
config, so two identical runs produce BIT-IDENTICAL parameters.  This is the
repeatability property the repetition code's CPU bitwise vote builds on, and the
regression net for any nondeterminism creeping into the decode (unordered
reductions, RNG misuse, stale buffers)."""
import pytest
import torch

from tests.dist_util import run_dist


def _run_once(approach, mode, kw, steps=6):
    from draco_amd.config import Config
    from draco_amd.parallel.trainer import Trainer

    cfg = Config(network="FC", dataset="MNIST", batch_size=8, device="cpu", lr=0.05,
                 approach=approach, mode=mode, err_mode="rev_grad", deterministic=True,
                 max_steps=50, eval_freq=0, log_dir="", train_dir="/tmp/det_ck", **kw)
    t = Trainer(cfg)
    t.logger.stdout_every = 0
    for _ in range(steps):
        t.train_step()
    out = t.space.flat_param.clone()
    t.close()
    return out


@pytest.mark.parametrize("approach,mode,kw", [
    ("baseline", "normal", dict(worker_fail=1)),
    ("baseline", "geometric_median", dict(worker_fail=1)),
    ("baseline", "krum", dict(worker_fail=1)),
    ("maj_vote", "maj_vote", dict(group_size=3, worker_fail=1)),
    ("cyclic", "cyclic", dict(worker_fail=1, workers_per_rank=4)),
])
def test_repeat_run_bitwise_identical(approach, mode, kw):
    a = _run_once(approach, mode, kw)
    b = _run_once(approach, mode, kw)
    assert torch.equal(a, b), f"{approach}/{mode} is not run-to-run deterministic"


def _dist_det_worker(rank, world):
    from draco_amd.config import Config
    from draco_amd.parallel.trainer import Trainer

    outs = []
    for _ in range(2):
        cfg = Config(network="FC", dataset="MNIST", batch_size=8, device="cpu", lr=0.05,
                     approach="maj_vote", mode="maj_vote", group_size=3, worker_fail=1,
                     err_mode="rev_grad", deterministic=True, max_steps=50, eval_freq=0,
                     log_dir="", train_dir="/tmp/det_d")
        t = Trainer(cfg)
        t.logger.stdout_every = 0
        for _ in range(5):
            t.train_step()
        outs.append(t.space.flat_param.clone())
        t.close()
    return bool(torch.equal(outs[0], outs[1]))


def test_distributed_repeat_run_identical():
    res = run_dist(_dist_det_worker, 2)
    assert all(res.values()), "distributed run is not repeat-deterministic"
