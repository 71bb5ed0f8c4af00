"""Coding-math unit tests (SURVEY §4 implication (a)): orthogonality, encode->decode
round trips under corruption, vote recovery, schedule determinism."""
import numpy as np
import pytest

from draco_amd.coding import (
    AdversarySchedule,
    build_cyclic_code,
    group_membership,
    majority_vote_index,
)


@pytest.mark.parametrize("n,s", [(4, 1), (8, 1), (8, 2), (7, 2), (15, 3), (6, 2)])
def test_cyclic_orthogonality(n, s):
    code = build_cyclic_code(n, s)
    assert np.abs(code.W_perp @ code.W).max() < 1e-10
    # band support: exactly 2s+1 nonzeros per row at the cyclic band
    nz = np.abs(code.W) > 1e-9
    assert (nz.sum(axis=1) == 2 * s + 1).all()
    for i in range(n):
        assert set(np.nonzero(nz[i])[0]) == set(code.support[i].tolist())


@pytest.mark.parametrize("n,s", [(4, 1), (8, 2), (15, 3)])
@pytest.mark.parametrize("err", ["rev", "const", "huge"])
def test_cyclic_roundtrip_with_corruption(n, s, err):
    rng = np.random.default_rng(42)
    code = build_cyclic_code(n, s)
    d = 257
    for nbad in range(s + 1):
        G = rng.normal(size=(n, d)) * 10.0
        R = np.stack([code.encode_oracle(w, G[code.support[w]]) for w in range(n)])
        bad = rng.choice(n, size=nbad, replace=False)
        if err == "rev":
            R[bad] += -100.0 * R[bad]
        elif err == "const":
            R[bad] += -100.0
        else:
            R[bad] = 1e6
        dec = code.decode_oracle(R, rng)
        ref = G.sum(axis=0)
        assert np.abs(dec - ref).max() < 1e-6 * max(np.abs(ref).max(), 1.0)


def test_cyclic_fp32_transport_precision():
    rng = np.random.default_rng(0)
    code = build_cyclic_code(8, 2)
    G = rng.normal(size=(8, 500)) * 1e3
    R = np.stack([code.encode_oracle(w, G[code.support[w]]) for w in range(8)])
    R = R.astype(np.complex64).astype(np.complex128)
    R[[2, 5]] *= -99.0
    dec = code.decode_oracle(R, rng)
    ref = G.sum(axis=0)
    assert np.abs(dec - ref).max() < 1e-3 * np.abs(ref).max()


def test_majority_vote():
    # honest majority wins regardless of order
    eq = np.eye(3, dtype=bool)
    eq[0, 2] = eq[2, 0] = True  # members 0 and 2 agree; 1 is corrupt
    assert majority_vote_index(eq) in (0, 2)
    eq = np.eye(5, dtype=bool)
    for a in (0, 2, 3):
        for b in (0, 2, 3):
            eq[a, b] = True
    w = majority_vote_index(eq)
    assert w in (0, 2, 3)
    # all agree
    assert majority_vote_index(np.ones((3, 3), dtype=bool)) == 0


def test_group_membership_layout():
    host, slot = group_membership(4, 3, 4)
    # each rank hosts exactly r slots, one per group
    for rank in range(4):
        served = [(g, i) for g in range(4) for i in range(3) if host[g, i] == rank]
        assert len(served) == 3
        assert sorted(i for _, i in served) == [0, 1, 2]


def test_schedule_deterministic_and_reference_seed():
    s1 = AdversarySchedule(8, 2, 100)
    s2 = AdversarySchedule(8, 2, 100)
    assert all(s1.adversaries_at(t) == s2.adversaries_at(t) for t in range(100))
    assert all(len(s1.adversaries_at(t)) == 2 for t in range(100))
    # varies over steps
    assert len({tuple(sorted(s1.adversaries_at(t))) for t in range(50)}) > 1


def test_cyclic_rejects_bad_params():
    with pytest.raises(ValueError):
        build_cyclic_code(3, 1)  # n < 2s+2
    with pytest.raises(ValueError):
        build_cyclic_code(8, 0)


def test_colocated_member_rows_full_and_survivor():
    from draco_amd.coding import colocated_member_rows

    G, r, world = 4, 3, 4
    rows, mask = colocated_member_rows(G, r, world, alive=list(range(world)))
    assert mask.all()
    # full world: member i of group g at row i*world + (g+i)%world
    for g in range(G):
        for i in range(r):
            assert rows[g, i] == i * world + (g + i) % world
    # rank 2 dead: its hosted members are forfeited, others re-indexed over [0,1,3]
    alive = [0, 1, 3]
    rows2, mask2 = colocated_member_rows(G, r, world, alive)
    pos = {0: 0, 1: 1, 3: 2}
    for g in range(G):
        for i in range(r):
            h = (g + i) % world
            if h == 2:
                assert not mask2[g, i]
            else:
                assert mask2[g, i] and rows2[g, i] == i * 3 + pos[h]
    # every group loses at most one member with one dead rank
    assert (mask2.sum(axis=1) >= r - 1).all()
