"""Is ResNet-18 backward bitwise-reproducible WITHOUT cudnn.deterministic?

The majority vote needs group members (same batch, same weights, different GPUs /
different processes) to produce bit-identical gradients.  cudnn.deterministic=True
guarantees it but costs 15x on MIOpen.  This probe measures, per setting:
  - within-process repeatability (backward twice, same weights/batch)
  - cross-process repeatability (run this script twice, compare saved grads)
"""
import hashlib
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from draco_amd.config import Config
from draco_amd.parallel.trainer import Trainer


def main():
    det = os.environ.get("DIAG_DET", "0") == "1"
    bench = os.environ.get("DIAG_BENCH", "0") == "1"
    cfg = Config(network="ResNet18", dataset="Cifar10", batch_size=128,
                 approach="maj_vote", mode="maj_vote", group_size=3, worker_fail=0,
                 device="cuda", dtype="bf16", deterministic=det,
                 max_steps=100, eval_freq=0, log_dir="", train_dir="/tmp/ck")
    t = Trainer(cfg)
    torch.backends.cudnn.benchmark = bench
    t.logger.stdout_every = 0
    x, y = t.data.batch_for(0, 0)
    g1 = t.space.alloc_payload(1)[0]
    g2 = t.space.alloc_payload(1)[0]
    # warm up MIOpen find
    t._forward_backward(x, y, g1)
    t._forward_backward(x, y, g1)
    t._forward_backward(x, y, g1)
    t._forward_backward(x, y, g2)
    torch.cuda.synchronize()
    within = torch.equal(g1, g2)
    h = hashlib.sha256(g1.cpu().numpy().tobytes()).hexdigest()[:16]
    # timing
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(10):
        t._forward_backward(x, y, g1)
    torch.cuda.synchronize()
    ms = (time.perf_counter() - t0) / 10 * 1000
    print(f"det={det} bench={bench}: within_proc_bitwise={within} grad_sha={h} fwdbwd_ms={ms:.1f}")


if __name__ == "__main__":
    main()
