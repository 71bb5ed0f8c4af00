import sys
sys.path.insert(0, '/root/repo')

def worker(rank, world):
    from draco_amd.config import Config
    from draco_amd.parallel.trainer import Trainer

    cfg = Config(network="ResNet18", dataset="Cifar10", batch_size=8, device="cpu",
                 lr=0.02, approach="maj_vote", mode="maj_vote", group_size=3,
                 worker_fail=1, err_mode="rev_grad", max_steps=50, eval_freq=0,
                 log_dir="", train_dir="/tmp/rn18ck")
    t = Trainer(cfg)
    t.logger.stdout_every = 0
    losses = [t.train_step()["loss"] for _ in range(3)]
    h = float(t.space.flat_param.double().sum())
    deg = t.agg.degenerate_steps
    t.close()
    return (losses, h, deg)

def main():
    from tests.dist_util import run_dist
    res = run_dist(worker, 8, timeout=900)
    hs = {r: res[r][1] for r in range(8)}
    assert len(set(hs.values())) == 1, f"divergence: {hs}"
    assert all(res[r][2] == 0 for r in range(8)), "degenerate votes"
    print("8-rank ResNet18 gloo OK: params identical on all ranks, 0 degenerate votes, losses", [round(x,3) for x in res[0][0]])

if __name__ == "__main__":
    main()
