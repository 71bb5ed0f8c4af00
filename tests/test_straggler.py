"""Straggler-as-erasure + kill-signal tests (PS topology, gloo CPU)."""
import time

import numpy as np
import pytest
import torch

from tests.dist_util import run_dist


def test_erasure_decode_oracle():
    """Cyclic decode with one zeroed (erased) row and one corrupted row, s=2."""
    from draco_amd.coding import build_cyclic_code

    rng = np.random.default_rng(5)
    code = build_cyclic_code(8, 2)
    d = 300
    G = rng.normal(size=(8, d))
    R = np.stack([code.encode_oracle(w, G[code.support[w]]) for w in range(8)])
    R[3] = 0.0  # erased (straggler timed out)
    R[6] *= -99.0  # Byzantine
    z = rng.normal(loc=1.0, size=d)
    healthy = code.locate_errors(code.W_perp @ (R @ z), known_bad={3})
    assert 3 not in healthy and 6 not in healthy
    v = code.recombination_vector(healthy)
    dec = np.real(v @ R)
    ref = G.sum(axis=0)
    assert np.abs(dec - ref).max() < 1e-6 * np.abs(ref).max()


def test_erasure_only_decode():
    from draco_amd.coding import build_cyclic_code

    rng = np.random.default_rng(6)
    code = build_cyclic_code(6, 2)
    d = 100
    G = rng.normal(size=(6, d))
    R = np.stack([code.encode_oracle(w, G[code.support[w]]) for w in range(6)])
    R[[1, 4]] = 0.0
    z = rng.normal(loc=1.0, size=d)
    healthy = code.locate_errors(code.W_perp @ (R @ z), known_bad={1, 4})
    v = code.recombination_vector(healthy)
    dec = np.real(v @ R)
    ref = G.sum(axis=0)
    assert np.abs(dec - ref).max() < 1e-6 * np.abs(ref).max()


def _ps_straggler_worker(rank, world, straggle_rank, steps):
    from draco_amd.config import Config
    from draco_amd.parallel.ps import Master, Worker

    cfg = Config(network="FC", dataset="MNIST", batch_size=4, device="cpu", lr=0.05,
                 topology="ps", approach="cyclic", mode="cyclic", worker_fail=1,
                 err_mode="none", straggler_timeout=1.0,
                 max_steps=50, eval_freq=0, log_dir="", train_dir="/tmp/draco_strag")
    if rank == 0:
        m = Master(cfg)
        m.run(max_steps=steps)
        return float(m.space.flat_param.double().abs().sum())
    w = Worker(cfg)
    if rank == straggle_rank:
        orig = w._fwd_bwd

        def slow(x, y, row):
            out = orig(x, y, row)
            if w.step_num == 1:
                time.sleep(4.0)  # way past the 1 s straggler timeout
            return out

        w._fwd_bwd = slow
    w.run(max_steps=steps)
    return None


def test_ps_straggler_becomes_erasure():
    # 1 PS + 4 cyclic workers (s=1); worker 2 stalls on step 1 -> erased, training continues
    res = run_dist(_ps_straggler_worker, 5, 2, 3, timeout=240)
    assert np.isfinite(res[0]) and res[0] > 0


def _ps_abort_worker(rank, world):
    """Master preempts mid-run after step 1; everyone exits long before max_steps."""
    import time as _t

    from draco_amd.config import Config
    from draco_amd.parallel.ps import Master, Worker

    cfg = Config(network="FC", dataset="MNIST", batch_size=4, device="cpu", lr=0.05,
                 topology="ps", approach="cyclic", mode="cyclic", worker_fail=1,
                 err_mode="none", max_steps=500, eval_freq=0, log_dir="",
                 train_dir="/tmp/draco_abort")
    t0 = _t.time()
    if rank == 0:
        m = Master(cfg)
        orig = m._gather_grads

        def gg():
            e = orig()
            if m.step_num == 1:
                m.request_abort()
            return e

        m._gather_grads = gg
        m.run()  # max_steps=500; abort must cut it short
        return _t.time() - t0
    w = Worker(cfg)
    w.run()
    return _t.time() - t0


def test_ps_master_abort_preempts():
    res = run_dist(_ps_abort_worker, 5, timeout=180)
    # 500 steps would take minutes; abort after step 1 ends the run in seconds
    assert max(res.values()) < 90
