import sys
sys.path.insert(0, '/root/repo')

def worker(rank, world):
    from draco_amd.config import Config
    from draco_amd.parallel.trainer import Trainer

    cfg = Config(network="FC", dataset="MNIST", batch_size=4, device="cpu", lr=0.03,
                 approach="maj_vote", mode="maj_vote", group_size=3, worker_fail=1,
                 err_mode="rev_grad", max_steps=300, eval_freq=0, log_dir="",
                 train_dir="/tmp/soak_ck")
    t = Trainer(cfg)
    t.logger.stdout_every = 0
    losses = [t.train_step()["loss"] for _ in range(200)]
    h = float(t.space.flat_param.double().sum())
    deg = t.agg.degenerate_steps
    t.close()
    return (losses[0], losses[-1], h, deg)

def main():
    from tests.dist_util import run_dist
    res = run_dist(worker, 3, timeout=600)
    hs = [res[r][2] for r in range(3)]
    assert hs[0] == hs[1] == hs[2], f"params diverged: {hs}"
    assert res[0][3] == 0, f"degenerate vote steps: {res[0][3]}"
    print(f"soak OK: loss {res[0][0]:.3f} -> {res[0][1]:.4f} over 200 attacked steps, params identical, 0 degenerate votes")

if __name__ == "__main__":
    main()
