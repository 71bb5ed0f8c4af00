"""Structured metrics: jsonl file + stdout line (replaces the reference's bare prints,
SURVEY §5.5 — same span taxonomy: comp/agg/update per step)."""
from __future__ import annotations

import json
import os
import sys
import time


class MetricsLogger:
    def __init__(self, log_dir: str, rank: int, stdout_every: int = 10):
        self.rank = rank
        self.stdout_every = stdout_every
        self._fh = None
        if log_dir:
            os.makedirs(log_dir, exist_ok=True)
            self._fh = open(os.path.join(log_dir, f"rank{rank}.jsonl"), "a")

    def log(self, rec: dict):
        rec = dict(rec, rank=self.rank, ts=time.time())
        if self._fh is not None:
            self._fh.write(json.dumps(rec) + "\n")
            self._fh.flush()
        if self.rank == 0 and self.stdout_every and rec.get("step", 0) % self.stdout_every == 0:
            fields = " ".join(
                f"{k}={v:.4f}" if isinstance(v, float) else f"{k}={v}"
                for k, v in rec.items()
                if k not in ("ts", "rank")
            )
            print(f"[step] {fields}", file=sys.stderr, flush=True)

    def close(self):
        if self._fh is not None:
            self._fh.close()
            self._fh = None
