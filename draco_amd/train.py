"""Training CLI (entry parity with /root/reference/src/distributed_nn.py).

Launch (one process per GPU over RCCL; gloo on CPU):
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 -m draco_amd.train -- \
      --approach maj_vote --group-size 3 --worker-fail 1 --network ResNet18 \
      --dataset Cifar10 --batch-size 128 --max-steps 1000

Topologies:
  colocated (default) — every rank computes + sharded decode (parallel/trainer.py)
  ps                  — rank 0 parameter server, ranks 1..P workers (parallel/ps.py)
"""
from __future__ import annotations


def main(argv=None):
    from .config import parse_cli
    from .parallel.trainer import Trainer

    cfg = parse_cli(argv)
    if cfg.topology == "ps":
        from .parallel.ps import run_ps

        run_ps(cfg)
        return

    t = Trainer(cfg)
    try:
        for _ in range(cfg.max_steps):
            t.train_step()
            if cfg.eval_freq > 0 and t.step_num % cfg.eval_freq == 0 and t.rank == 0:
                m = t.evaluate()
                print(f"Testset Performance: Cur Step:{t.step_num} "
                      f"Prec@1: {m['prec1']:.4f} Prec@5: {m['prec5']:.4f} Loss: {m['loss']:.4f}",
                      flush=True)
    finally:
        t.close()


if __name__ == "__main__":
    main()
