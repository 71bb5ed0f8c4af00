"""8-rank gloo soak (VERDICT r01 #1 'done' criterion): 200+ steps per approach with
bit-identical params on every rank, zero degenerate votes, zero skipped updates.

  python tools/validate_8rank_cpu.py            # full soak (FC 208 steps x 3 arms
                                                #  + ResNet18 structural check)
  SOAK_STEPS=24 python tools/validate_8rank_cpu.py   # quick mode
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

STEPS = int(os.environ.get("SOAK_STEPS", "208"))
RN_STEPS = int(os.environ.get("SOAK_RN_STEPS", "24"))


def worker(rank, world, network, dataset, steps, approach, kw):
    from draco_amd.config import Config
    from draco_amd.parallel.trainer import Trainer

    base = dict(network=network, dataset=dataset, batch_size=8, device="cpu",
                lr=0.02, approach=approach, err_mode="rev_grad",
                max_steps=steps + 10, eval_freq=0, log_dir="",
                train_dir="/tmp/soak_ck")
    cfg = Config(**{**base, **kw})
    t = Trainer(cfg)
    t.logger.stdout_every = 0
    losses = [t.train_step()["loss"] for _ in range(steps)]
    h = float(t.space.flat_param.double().sum())
    deg = getattr(t.agg, "degenerate_steps", 0)
    skipped = t.skipped_updates
    t.close()
    return (losses[0], losses[-1], h, deg, skipped)


def soak(name, network, dataset, steps, approach, kw):
    from tests.dist_util import run_dist

    res = run_dist(worker, 8, network, dataset, steps, approach, kw, timeout=3600)
    hs = {r: res[r][2] for r in range(8)}
    assert len(set(hs.values())) == 1, f"{name}: param divergence {hs}"
    assert all(res[r][3] == 0 for r in range(8)), f"{name}: degenerate votes"
    assert all(res[r][4] == 0 for r in range(8)), f"{name}: skipped updates"
    assert res[0][1] < res[0][0], f"{name}: loss did not decrease"
    print(f"OK {name}: {steps} steps x 8 ranks, params bit-identical, "
          f"0 degenerate / 0 skipped, loss {res[0][0]:.3f} -> {res[0][1]:.3f}")


def main():
    soak("maj_vote r=3 s=1 FC", "FC", "MNIST", STEPS, "maj_vote",
         dict(mode="maj_vote", group_size=3, worker_fail=1))
    soak("cyclic r=3 s=1 FC", "FC", "MNIST", STEPS, "cyclic",
         dict(mode="cyclic", worker_fail=1, workers_per_rank=1))
    soak("baseline mean (bucketed overlap) FC", "FC", "MNIST", STEPS, "baseline",
         dict(mode="normal", worker_fail=0, err_mode="none", bucket_mb=0.05))
    soak("maj_vote r=3 s=1 ResNet18 (structural)", "ResNet18", "Cifar10", RN_STEPS,
         "maj_vote", dict(mode="maj_vote", group_size=3, worker_fail=1))
    print("8-rank soak PASSED")


if __name__ == "__main__":
    main()
