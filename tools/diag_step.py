"""Step-time diagnostic: where do the milliseconds go (compute / data / agg / update)?"""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from draco_amd.config import Config
from draco_amd.parallel.trainer import Trainer


def timeit(fn, n=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1000


def main():
    det = os.environ.get("DIAG_DET", "1") == "1"
    cfg = Config(network="ResNet18", dataset="Cifar10", batch_size=128,
                 approach="maj_vote", mode="maj_vote", group_size=3, worker_fail=1,
                 device="cuda", dtype=os.environ.get("DIAG_DTYPE", "bf16"),
                 deterministic=det,
                 max_steps=1000, eval_freq=0, log_dir="", train_dir="/tmp/ck")
    t = Trainer(cfg)
    t.logger.stdout_every = 0

    # full step
    full = timeit(lambda: t.train_step(), n=10)

    # data generation only
    data_ms = timeit(lambda: t.data.batch_for(0, 1), n=20)

    # fwd+bwd only (one batch)
    x, y = t.data.batch_for(0, 2)
    row = t.payload[0]
    fb_ms = timeit(lambda: t._forward_backward(x, y, row), n=10)

    # aggregation only
    agg_ms = timeit(lambda: t.agg.aggregate(t.payload, 3), n=20)

    # optimizer only
    g = t.payload[0]
    opt_ms = timeit(lambda: t.opt.step(g), n=20)

    print(f"det={det} dtype={cfg.dtype}: full={full:.1f}ms data={data_ms:.2f}ms "
          f"fwdbwd={fb_ms:.1f}ms (x3={3*fb_ms:.1f}) agg={agg_ms:.2f}ms opt={opt_ms:.2f}ms "
          f"d={t.space.d}")


if __name__ == "__main__":
    main()
