"""Deterministic per-step adversary schedule.

Mirrors /root/reference/src/util.py:100-103 (`_generate_adversarial_nodes`, SEED_=428):
for every step, `worker_fail` distinct logical workers are drawn uniformly; the draw
sequence is fixed by the seed so every rank computes the identical schedule locally
(no communication).  Worker ids here are the framework's logical worker ids 0..P-1
(the reference's MPI ranks 1..P map to 0..P-1).
"""
from __future__ import annotations

import numpy as np

SEED_ = 428  # reference: util.py:17


class AdversarySchedule:
    def __init__(self, num_workers: int, worker_fail: int, max_steps: int, seed: int = SEED_):
        self.num_workers = num_workers
        self.worker_fail = worker_fail
        rng = np.random.RandomState(seed)
        # one draw per step, exactly as the reference does (one list per step, size=fail)
        self._sched = [
            rng.choice(np.arange(num_workers), size=worker_fail, replace=False)
            for _ in range(max_steps + 1)
        ]
        self._sets = [frozenset(int(x) for x in s) for s in self._sched]

    def adversaries_at(self, step: int) -> frozenset:
        if self.worker_fail == 0:
            return frozenset()
        return self._sets[min(step, len(self._sets) - 1)]

    def is_adversarial(self, worker: int, step: int) -> bool:
        return worker in self.adversaries_at(step)
