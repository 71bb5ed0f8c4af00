"""Failure detection for the colocated topology: heartbeats + survivor subgroups.

The reference's PS blocks forever when a worker dies
(/root/reference/src/master/baseline_master.py:112-116 — the gather loop has no
timeout); its only gesture at the problem is a never-sent tag-77 kill signal
(SURVEY §5.3).  Here failure handling is first-class and matches the coding layer:
a dead rank's logical workers become ERASURES — known-bad rows the cyclic decode
removes for free (<= s of them) and forfeited members the vote simply drops.

Mechanism (single 8-GPU node, torch.distributed):
  * every rank heartbeats `hb_{rank} = "{step}:{monotonic-ish wall time}"` into the
    rendezvous TCPStore each step (one store round-trip, microseconds, no collective);
  * before each step, ranks read peers' heartbeats; a peer whose beat is older than
    `timeout` seconds is declared dead.  All survivors reach the same verdict within
    one step of each other because they share the store and the wall clock (one node);
    a deterministic confirmation barrier (the first collective on the survivor group)
    aligns them;
  * leave-one-out subgroups for every rank are pre-created while everyone is alive
    (dist.new_group is collective and cannot be created after a member died), so on
    failure the survivors switch to a ready-made process group and re-shard;
  * a rank that dies MID-collective surfaces as a RuntimeError on the gloo/RCCL op;
    the trainer catches it, waits for the heartbeat verdict, switches groups, and
    RE-RUNS the interrupted step (parameters were not yet updated, batches are
    deterministic functions of (worker, step), so the retry is exact).

Covers one failed rank per event (the leave-one-out set); a second failure raises.
"""
from __future__ import annotations

import threading
import time

import torch.distributed as dist


class HealthMonitor:
    """Heartbeats run on a BACKGROUND daemon thread (every timeout/4), so a rank
    busy in a long step (MIOpen find / graph capture can take minutes) still beats
    — only a DEAD process stops.  A hung-but-alive rank is therefore not detected
    here (that is the PS lane's timeout-as-erasure job, ps.py:203-275); this lane
    handles process death, which is what hangs the reference forever."""

    def __init__(self, rank: int, world: int, timeout: float):
        self.rank = rank
        self.world = world
        self.timeout = timeout
        self.store = dist.distributed_c10d._get_default_store()
        self.alive = list(range(world))
        # leave-one-out subgroups, created NOW while all ranks participate
        self.loo_groups = {}
        for dead in range(world):
            ranks = [r for r in range(world) if r != dead]
            self.loo_groups[dead] = dist.new_group(ranks=ranks)
        self.beat(step=-1)
        self._stop = threading.Event()
        t = threading.Thread(target=self._beat_loop, daemon=True)
        t.start()

    def _beat_loop(self):
        while not self._stop.wait(self.timeout / 4.0):
            self.store.set(f"hb_{self.rank}", f"-1:{time.time()}")

    def close(self):
        self._stop.set()

    def beat(self, step: int) -> None:
        self.store.set(f"hb_{self.rank}", f"{step}:{time.time()}")

    def _age(self, peer: int) -> float:
        try:
            raw = self.store.get(f"hb_{peer}").decode()
            return time.time() - float(raw.split(":")[1])
        except Exception:
            return float("inf")

    def check(self) -> list:
        """Peers (from the currently-alive set) whose heartbeat is stale."""
        return [p for p in self.alive
                if p != self.rank and self._age(p) > self.timeout]

    def wait_for_dead(self, max_wait: float = 60.0) -> list:
        """After a collective error: poll until some peer's heartbeat goes stale
        (it takes up to `timeout` seconds for a just-died peer to look dead)."""
        t0 = time.time()
        while time.time() - t0 < max_wait:
            dead = self.check()
            if dead:
                return dead
            time.sleep(0.05)
        raise RuntimeError("collective failed but no peer heartbeat went stale")

    def declare_dead(self, dead: list):
        """Commit the failure: returns (subgroup, survivor_ranks).  Single-failure
        per event; survivors keep their original rank ids."""
        if len(dead) != 1 or dead[0] not in self.alive:
            raise RuntimeError(f"unsupported failure set {dead} (alive={self.alive})")
        if len(self.alive) < self.world:
            raise RuntimeError("second rank failure: only one failure per run is supported")
        self.alive = [r for r in self.alive if r not in dead]
        return self.loo_groups[dead[0]], list(self.alive)
