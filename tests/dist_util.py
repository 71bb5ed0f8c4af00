"""Spawn helper for multi-process gloo tests (world_size > 1 on CPU)."""
from __future__ import annotations

import multiprocessing as mp
import os
import socket
import traceback


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _entry(fn, rank, world, port, q, args, kwargs):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        os.environ["RANK"] = str(rank)
        os.environ["WORLD_SIZE"] = str(world)
        os.environ["LOCAL_RANK"] = str(rank)
        res = fn(rank, world, *args, **kwargs)
        q.put((rank, "ok", res))
    except Exception:
        q.put((rank, "err", traceback.format_exc()))


def run_dist(fn, world: int, *args, timeout: float = 180.0, expect_missing=(), **kwargs):
    """Run fn(rank, world, *args) in `world` processes over gloo; returns
    {rank: result}.  Raises on any child failure.  Ranks in `expect_missing` are
    allowed to die without reporting (failure-injection tests)."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = _free_port()
    procs = [
        ctx.Process(target=_entry, args=(fn, r, world, port, q, args, kwargs))
        for r in range(world)
    ]
    for p in procs:
        p.start()
    results = {}
    errs = []
    for _ in range(world - len(set(expect_missing))):
        rank, status, payload = q.get(timeout=timeout)
        if status == "ok":
            results[rank] = payload
        else:
            errs.append(f"rank {rank}:\n{payload}")
    for p in procs:
        p.join(timeout=30)
        if p.is_alive():
            p.terminate()
    if errs:
        raise AssertionError("\n".join(errs))
    return results
