"""Repetition (group-wise majority vote) gradient code.

Reference semantics: workers are partitioned into groups of size r; members of a group
draw identical batches (shared RNG seed, /root/reference/src/util.py:69-97 and
rep_worker.py:89) so honest members produce *identical* gradients; the decoder runs a
Boyer-Moore majority vote over each group's r gradients and averages the group winners
(/root/reference/src/master/rep_master.py:141-168).

MI355X-native layout (colocated topology): with N ranks and G = N groups, member i of
group g lives on rank (g + i) % N, i.e. every rank hosts exactly r logical workers
(one per group it serves) — per-GPU work is fixed (weak scaling) and the gradient
exchange is an all-to-all of d/N shards, so the vote itself runs sharded on every GPU.

The vote over full vectors factorises over shards: members a, b are equal iff they are
equal on every shard, so ranks compute r x r per-shard equality bits (HIP kernel /
torch fallback), AND-reduce them (a tiny allreduce), and then every rank runs the same
Boyer-Moore pass below on the combined bits — selecting the same winner everywhere.
"""
from __future__ import annotations

import numpy as np


def majority_vote_index(eq: np.ndarray) -> int:
    """Boyer-Moore majority vote over r members given their pairwise equality matrix.

    eq: (r, r) boolean, eq[a, b] == True iff member a's and b's gradients are equal
    (eq must be reflexive/symmetric).  Returns the index of the winning member,
    replicating the reference's pass exactly (rep_master.py:154-168): no verification
    pass — with no true majority the last surviving candidate wins, as in the reference.
    """
    r = eq.shape[0]
    maj = 0
    counter = 1
    for i in range(1, r):
        if eq[i, maj]:
            counter += 1
        elif counter == 1:
            maj = i
        else:
            counter -= 1
    return maj


def colocated_member_rows(n_groups: int, group_size: int, world0: int, alive):
    """Recv-row coordinates of every group member over a (possibly reduced) rank set.

    Member i of group g is hosted at ORIGINAL rank (g + i) % world0 (the colocated
    layout, which never changes); after the all_to_all over the survivor group its
    recv row is i * W' + pos(host) with W' = len(alive).  Members hosted on dead
    ranks are forfeited (mask False) — the erasure form of the repetition decode.

    Returns (member_rows (G, r) int64, member_mask (G, r) bool).
    """
    pos = {rk: i for i, rk in enumerate(alive)}
    Wp = max(len(alive), 1)
    G, r = n_groups, group_size
    rows = np.zeros((G, r), dtype=np.int64)
    mask = np.zeros((G, r), dtype=bool)
    for g in range(G):
        for i in range(r):
            h = (g + i) % world0
            if h in pos:
                rows[g, i] = i * Wp + pos[h]
                mask[g, i] = True
    return rows, mask


def group_membership(n_groups: int, group_size: int, world: int):
    """Member i of group g is hosted as local worker i of rank (g + i) % world.

    Returns (host_rank, local_slot) arrays of shape (n_groups, group_size).
    The local slot of a member on its host rank is its member index i, so each rank
    hosts exactly group_size logical workers: slot i serves group (rank - i) % n_groups
    when n_groups == world (the colocated bench layout).
    """
    g = np.arange(n_groups)[:, None]
    i = np.arange(group_size)[None, :]
    host = (g + i) % world
    return host, np.broadcast_to(i, host.shape).copy()
