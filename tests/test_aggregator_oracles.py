"""Geo-median / Krum aggregators vs independent brute-force oracles (world=1)."""
import numpy as np
import torch
import torch.nn as nn

from draco_amd.parallel.aggregators import GeoMedianAggregator, KrumAggregator
from draco_amd.parallel.comm import Communicator
from draco_amd.parallel.flat import FlatSpace


def _setup(d_in=30, d_out=7):
    torch.manual_seed(3)
    comm = Communicator(0, 1, torch.device("cpu"))
    model = nn.Sequential(nn.Linear(d_in, d_out), nn.Linear(d_out, 4))
    space = FlatSpace(model, 1, torch.device("cpu"))
    return comm, space


def _weiszfeld_oracle(X, iters=200):
    """Straight textbook Weiszfeld on (P, d) points."""
    z = X.mean(0)
    for _ in range(iters):
        dist = np.linalg.norm(X - z, axis=1)
        dist = np.maximum(dist, 1e-12)
        w = 1.0 / dist
        z_new = (w[:, None] * X).sum(0) / w.sum()
        if np.linalg.norm(z_new - z) <= 1e-9 * max(np.linalg.norm(z_new), 1e-12):
            z = z_new
            break
        z = z_new
    return z


def test_geomedian_matches_oracle_per_layer():
    comm, space = _setup()
    P = 5
    agg = GeoMedianAggregator(comm, space, num_workers=P, max_iter=200, tol=1e-9)
    rng = np.random.default_rng(0)
    payload = torch.tensor(rng.normal(size=(P, space.d_pad)), dtype=torch.float32)
    payload[:, space.d:] = 0.0
    out = agg.aggregate(payload, 0).numpy()
    # oracle: per-parameter-tensor Weiszfeld (the reference's per-layer semantics)
    segs = space.seg_bounds.numpy()
    for l in range(len(segs) - 1):
        lo, hi = segs[l], segs[l + 1]
        ref = _weiszfeld_oracle(payload[:, lo:hi].double().numpy())
        assert np.abs(out[lo:hi] - ref).max() < 1e-3 * max(np.abs(ref).max(), 1.0), l


def test_geomedian_robust_to_outlier():
    comm, space = _setup()
    P = 5
    agg = GeoMedianAggregator(comm, space, num_workers=P)
    rng = np.random.default_rng(1)
    base = rng.normal(size=space.d_pad)
    payload = torch.tensor(
        np.stack([base + 0.01 * rng.normal(size=space.d_pad) for _ in range(P)]),
        dtype=torch.float32)
    payload[2] *= -100.0  # one Byzantine row
    out = agg.aggregate(payload, 0).numpy()
    # geometric median stays near the honest cluster
    assert np.abs(out[: space.d] - base[: space.d]).max() < 0.2


def test_krum_matches_oracle():
    comm, space = _setup()
    P, s = 6, 1
    agg = KrumAggregator(comm, space, num_workers=P, s=s)
    rng = np.random.default_rng(2)
    payload = torch.tensor(rng.normal(size=(P, space.d_pad)), dtype=torch.float32)
    payload[:, space.d:] = 0.0
    out = agg.aggregate(payload, 0).numpy()
    X = payload.double().numpy()
    segs = space.seg_bounds.numpy()
    keep = P - s - 2
    for l in range(len(segs) - 1):
        lo, hi = segs[l], segs[l + 1]
        Xl = X[:, lo:hi]
        scores = []
        for i in range(P):
            d2 = sorted(np.sum((Xl[i] - Xl[j]) ** 2) for j in range(P) if j != i)
            scores.append(sum(d2[:keep]))
        winner = int(np.argmin(scores))  # reference __krum: first argmin
        assert np.allclose(out[lo:hi], Xl[winner], atol=1e-5), l


def test_krum_excludes_adversary():
    comm, space = _setup()
    P, s = 6, 1
    agg = KrumAggregator(comm, space, num_workers=P, s=s)
    rng = np.random.default_rng(4)
    base = rng.normal(size=space.d_pad).astype(np.float32)
    payload = torch.tensor(
        np.stack([base + 0.01 * rng.normal(size=space.d_pad).astype(np.float32) for _ in range(P)]))
    payload[4] = torch.tensor(base) * -100.0
    out = agg.aggregate(payload, 0)
    assert float((out[: space.d] - torch.tensor(base[: space.d])).abs().max()) < 0.1
