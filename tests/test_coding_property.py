"""Property-based coding tests (hypothesis): encode->decode exactness over random
(n, s, adversary-set, error-pattern) draws."""
import numpy as np
import pytest
from hypothesis import given, settings, strategies as st

from draco_amd.coding import build_cyclic_code, majority_vote_index

_CODES = {}


def _code(n, s):
    if (n, s) not in _CODES:
        _CODES[(n, s)] = build_cyclic_code(n, s)
    return _CODES[(n, s)]


@settings(max_examples=40, deadline=None)
@given(
    ns=st.sampled_from([(4, 1), (6, 1), (6, 2), (8, 2), (10, 3)]),
    seed=st.integers(0, 2**31 - 1),
    err_scale=st.sampled_from([1e-3, 1.0, 1e4]),
)
def test_cyclic_roundtrip_random(ns, seed, err_scale):
    n, s = ns
    code = _code(n, s)
    rng = np.random.default_rng(seed)
    d = 64
    G = rng.normal(size=(n, d))
    R = np.stack([code.encode_oracle(w, G[code.support[w]]) for w in range(n)])
    nbad = int(rng.integers(0, s + 1))
    bad = rng.choice(n, size=nbad, replace=False)
    R[bad] += err_scale * (rng.normal(size=(nbad, d)) + 1j * rng.normal(size=(nbad, d)))
    dec = code.decode_oracle(R, rng)
    ref = G.sum(axis=0)
    assert np.abs(dec - ref).max() < 1e-5 * max(np.abs(ref).max(), 1.0)


@settings(max_examples=60, deadline=None)
@given(
    r=st.sampled_from([3, 5, 7]),
    seed=st.integers(0, 2**31 - 1),
)
def test_majority_vote_honest_majority_wins(r, seed):
    """With a strict honest majority of IDENTICAL members, a corrupt member can
    never win, regardless of order."""
    rng = np.random.default_rng(seed)
    n_bad = int(rng.integers(0, (r - 1) // 2 + 1))
    labels = np.array([0] * (r - n_bad) + list(range(1, n_bad + 1)))
    rng.shuffle(labels)
    eq = labels[:, None] == labels[None, :]
    w = majority_vote_index(eq)
    assert labels[w] == 0


@settings(max_examples=20, deadline=None)
@given(seed=st.integers(0, 2**31 - 1))
def test_cyclic_erasure_plus_error(seed):
    code = _code(8, 2)
    rng = np.random.default_rng(seed)
    d = 48
    G = rng.normal(size=(8, d))
    R = np.stack([code.encode_oracle(w, G[code.support[w]]) for w in range(8)])
    erase, err = rng.choice(8, size=2, replace=False)
    R[erase] = 0.0
    R[err] *= -99.0
    z = rng.normal(loc=1.0, size=d)
    healthy = code.locate_errors(code.W_perp @ (R @ z), known_bad={int(erase)})
    v = code.recombination_vector(healthy)
    dec = np.real(v @ R)
    ref = G.sum(axis=0)
    assert np.abs(dec - ref).max() < 1e-6 * max(np.abs(ref).max(), 1.0)


@given(
    st.integers(min_value=2, max_value=8),   # world
    st.integers(min_value=2, max_value=6),   # r
    st.integers(min_value=0, max_value=7),   # dead rank (mod world)
)
@settings(max_examples=60, deadline=None)
def test_colocated_member_rows_properties(world, r, dead_mod):
    """Survivor member-row mapping invariants for every (world, r, dead) shape:
    rows in range, alive members map to distinct recv slots per rank-slot pair,
    each group loses at most ceil(r/world) members per dead rank."""
    import numpy as np

    from draco_amd.coding import colocated_member_rows

    G = world
    dead = dead_mod % world
    alive = [x for x in range(world) if x != dead]
    rows, mask = colocated_member_rows(G, r, world, alive)
    Wp = len(alive)
    assert rows.shape == (G, r) and mask.shape == (G, r)
    for g in range(G):
        lost = int((~mask[g]).sum())
        assert lost <= -(-r // world), (g, lost)
        for i in range(r):
            if mask[g, i]:
                assert 0 <= rows[g, i] < r * Wp
                # recv slot (i, src) is unique per (group) within member index i
    # two different groups' members with the same slot i map to different src
    # positions unless they share the host rank
    for i in range(r):
        hosts = [(g + i) % world for g in range(G)]
        for g in range(G):
            if mask[g, i]:
                assert rows[g, i] == i * Wp + alive.index(hosts[g])


@given(
    st.lists(st.integers(min_value=1, max_value=300), min_size=2, max_size=8),
    st.integers(min_value=1, max_value=4),
    st.floats(min_value=0.0005, max_value=2.0),
)
@settings(max_examples=40, deadline=None)
def test_bucket_tiling_random_models(widths, world, bucket_mb):
    """Bucket ranges tile [0, d_pad) top-down with aligned boundaries for any
    layer-size profile / world / bucket size."""
    import torch
    import torch.nn as nn

    from draco_amd.parallel.flat import ALIGN, FlatSpace

    layers = []
    prev = 7
    for w in widths:
        layers.append(nn.Linear(prev, w))
        prev = w
    space = FlatSpace(nn.Sequential(*layers), world, torch.device("cpu"))
    buckets = space.build_buckets(bucket_mb)
    expect_hi = space.d_pad
    seen = set()
    for lo, hi, idxs in buckets:
        assert hi == expect_hi and lo < hi
        assert lo % ALIGN == 0 or lo == 0
        seen.update(idxs)
        expect_hi = lo
    assert expect_hi == 0
    assert seen == set(range(len(space.params)))


@given(
    st.integers(min_value=2, max_value=5),   # world
    st.integers(min_value=3, max_value=6),   # r
    st.integers(min_value=0, max_value=31),  # adversary pattern seed
)
@settings(max_examples=40, deadline=None)
def test_vote_with_forfeits_recovers_honest_mean(world, r, adv_seed):
    """Unsharded VoteAggregator with a dead rank's members forfeited: as long as
    each group keeps an honest majority among its ALIVE members, the aggregate is
    exactly the mean of honest gradients."""
    import numpy as np
    import torch
    import torch.nn as nn

    from draco_amd.coding import colocated_member_rows
    from draco_amd.parallel.aggregators import VoteAggregator
    from draco_amd.parallel.comm import Communicator
    from draco_amd.parallel.flat import FlatSpace

    rng = np.random.default_rng(adv_seed)
    dead = int(rng.integers(0, world))
    alive = [x for x in range(world) if x != dead]
    G = world
    rows, mask = colocated_member_rows(G, r, world, alive)
    torch.manual_seed(1)
    model = nn.Linear(30, 4)
    space = FlatSpace(model, 1, torch.device("cpu"))
    comm = Communicator(0, 1, torch.device("cpu"))
    # recv layout: row i*Wp + pos(src) over ALL alive (simulate post-exchange)
    Wp = len(alive)
    recv = torch.zeros(r * Wp, space.d_pad)
    honest = {}
    expected = torch.zeros(space.d_pad)
    for g in range(G):
        torch.manual_seed(100 + g)
        hg = torch.randn(space.d_pad)
        honest[g] = hg
        expected += hg
        alive_members = [i for i in range(r) if mask[g, i]]
        n_alive = len(alive_members)
        max_adv = (n_alive - 1) // 2
        advs = set(rng.choice(alive_members, size=rng.integers(0, max_adv + 1),
                              replace=False).tolist()) if max_adv > 0 else set()
        for i in alive_members:
            recv[rows[g, i]] = hg * (-100.0 if i in advs else 1.0)
    expected /= G
    agg = VoteAggregator(comm, space, group_size=r, atol=0.0,
                         member_rows=rows, member_mask=mask)
    # feed the pre-exchanged rows directly (world-1 comm: exchanged() is a view,
    # so hand aggregate() a payload shaped (r*Wp, d_pad))
    out = agg.aggregate(recv, step=0)
    assert torch.allclose(out, expected, atol=1e-5), \
        float((out - expected).abs().max())


@given(
    st.integers(min_value=3, max_value=4),   # world0
    st.integers(min_value=1, max_value=2),   # L (workers per rank)
    st.integers(min_value=0, max_value=97),  # seed
)
@settings(max_examples=25, deadline=None)
def test_cyclic_survivor_mapping_decodes_exact(world0, L, seed):
    """CyclicAggregator with a dead original rank: recv rows in survivor-world
    coordinates + erased workers as known-bad must still decode the exact sum,
    for every (world0, L, dead) geometry within capability (L erasures <= s)."""
    import numpy as np
    import torch
    import torch.nn as nn

    from draco_amd import ops
    from draco_amd.coding import build_cyclic_code
    from draco_amd.parallel.aggregators import CyclicAggregator
    from draco_amd.parallel.comm import Communicator
    from draco_amd.parallel.flat import FlatSpace

    rng = np.random.default_rng(seed)
    n = L * world0
    s = L  # one dead rank erases exactly L workers; stay within the locator budget
    if n < 2 * s + 2:
        return  # geometry outside code constraints
    dead = int(rng.integers(0, world0))
    alive = [x for x in range(world0) if x != dead]
    code = build_cyclic_code(n, s)
    torch.manual_seed(2)
    model = nn.Linear(40, 5)
    space = FlatSpace(model, 1, torch.device("cpu"))
    comm = Communicator(0, 1, torch.device("cpu"))
    agg = CyclicAggregator(comm, space, code, workers_per_rank=L,
                           world0=world0, alive=alive)

    def sub_grad(j):
        torch.manual_seed(700 + j)
        return torch.randn(space.d_pad)

    Wp = len(alive)
    recv = torch.zeros(2 * L * Wp, space.d_pad)
    for w in range(n):
        l, src = w // world0, w % world0
        if src == dead:
            continue  # erased: its rows never arrive
        sup = code.support[w]
        grads = torch.stack([sub_grad(int(j)) for j in sup])
        wre = torch.tensor(np.real(code.W[w, sup]), dtype=torch.float32)
        wim = torch.tensor(np.imag(code.W[w, sup]), dtype=torch.float32)
        enc = torch.zeros(2, space.d_pad)
        ops.cyclic_encode(grads, wre, wim, enc)
        pos = alive.index(src)
        recv[(2 * l) * Wp + pos] = enc[0]
        recv[(2 * l + 1) * Wp + pos] = enc[1]
    out = agg.aggregate(recv, step=int(rng.integers(0, 1000)))
    ref = torch.stack([sub_grad(j) for j in range(n)]).sum(0) / n
    assert torch.allclose(out, ref, atol=1e-4), float((out - ref).abs().max())
