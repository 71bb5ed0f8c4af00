"""Isolate the cfg5 honest-gradient spike: concurrent graph replay vs serialized
replay vs eager.  Spike present in all three -> model-kernel/data numerics; only
in concurrent -> replay race.

  gpurun -- 'python tools/diag_cfg5_isolate.py > gpurun_out/isolate.txt 2>&1'
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from draco_amd.config import Config
from draco_amd.parallel.trainer import Trainer


def run(name, steps, serial=False, **kw):
    base = dict(network="ResNet50", dataset="ImageNetSynthetic", batch_size=32,
                approach="cyclic", mode="cyclic", worker_fail=2, err_mode="rev_grad",
                device="cuda", dtype="bf16", max_steps=steps + 10, eval_freq=0,
                log_dir="", train_dir="/tmp/ck5i")
    cfg = Config(**{**base, **kw})
    t = Trainer(cfg)
    t.logger.stdout_every = 0
    if serial:
        t._worker_stream = lambda i: None  # graphs replay sequentially on stream 0
    spike = None
    for i in range(steps):
        t.train_step()
        gm = float(t.agg._out.abs().max())
        if gm > 1e6 and spike is None:
            spike = i
            print(f"{name}: SPIKE at step {i} grad_max={gm:.3e}")
            break
        if i % 25 == 0:
            print(f"{name} step {i}: grad_max={gm:.3e} "
                  f"param_max={float(t.space.flat_param.abs().max()):.3e}")
    print(f"{name}: done, spike={spike}, skipped={t.skipped_updates}")
    t.close()
    del t
    torch.cuda.empty_cache()
    return spike


def matrix():
    run("graphs-nchw", 120, channels_last=False)
    run("graphs-fp32", 120, dtype="fp32")
    run("graphs-no-adversary", 120, err_mode="none")


if __name__ == "__main__":
    if os.environ.get("MATRIX") == "1":
        matrix()
        raise SystemExit(0)
    run("concurrent-graphs", 120)
    run("serial-graphs", 120, serial=True)
    run("eager", 120, hip_graphs=False)
