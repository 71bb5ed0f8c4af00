"""Single-machine sanity runner (parity with /root/reference/src/single_machine.py):
runs the same trainer with world_size 1 — all logical workers and the full decode
pipeline execute locally, so coding/aggregation logic can be exercised without any
distributed launch.

  python -m draco_amd.single_machine --approach maj_vote --group-size 3 \
      --worker-fail 1 --network LeNet --dataset MNIST --max-steps 100
"""
from __future__ import annotations


def main(argv=None):
    import os

    os.environ.setdefault("WORLD_SIZE", "1")
    from .config import parse_cli
    from .parallel.trainer import Trainer

    cfg = parse_cli(argv)
    t = Trainer(cfg)
    try:
        for _ in range(cfg.max_steps):
            rec = t.train_step()
            if cfg.eval_freq > 0 and t.step_num % cfg.eval_freq == 0:
                m = t.evaluate()
                print(f"[eval] step={t.step_num} prec1={m['prec1']:.4f} loss={m['loss']:.4f}")
    finally:
        t.close()


if __name__ == "__main__":
    main()
