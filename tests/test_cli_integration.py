"""End-to-end CLI integration: the exact launcher form from the README
(`torch.distributed.run ... -m draco_amd.train`), world 2 over gloo, with eval
printing and checkpointing — exercises train.py's main loop, not just Trainer."""
import os
import socket
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def test_torchrun_train_cli(tmp_path):
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(_free_port()),
         "-m", "draco_amd.train", "--",
         "--approach", "maj_vote", "--group-size", "3", "--worker-fail", "1",
         "--network", "FC", "--dataset", "MNIST", "--batch-size", "8",
         "--device", "cpu", "--max-steps", "6", "--eval-freq", "3",
         "--train-dir", str(tmp_path), "--log-dir", ""],
        cwd=REPO, capture_output=True, text=True, timeout=240)
    assert out.returncode == 0, out.stderr[-3000:]
    # reference-format eval line printed by rank 0 (distributed_evaluator parity)
    assert "Testset Performance:" in out.stdout
    assert "Prec@1:" in out.stdout
    # checkpoints in the reference layout (model_step_N) at both eval boundaries
    assert os.path.exists(tmp_path / "model_step_3")
    assert os.path.exists(tmp_path / "model_step_6")


def test_torchrun_ps_topology_cli(tmp_path):
    """Reference-parity PS topology through the CLI: 1 master + 2 workers on gloo."""
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "3", "--master-addr", "127.0.0.1",
         "--master-port", str(_free_port()),
         "-m", "draco_amd.train", "--",
         "--topology", "ps", "--approach", "maj_vote", "--group-size", "2",
         "--worker-fail", "0", "--err-mode", "none",
         "--network", "FC", "--dataset", "MNIST", "--batch-size", "8",
         "--device", "cpu", "--max-steps", "4", "--eval-freq", "2",
         "--train-dir", str(tmp_path), "--log-dir", ""],
        cwd=REPO, capture_output=True, text=True, timeout=240)
    assert out.returncode == 0, out.stderr[-3000:]
    assert os.path.exists(tmp_path / "model_step_2")
    assert os.path.exists(tmp_path / "model_step_4")


def test_evaluator_cli_once(tmp_path):
    """`python -m draco_amd.evaluate --once` scores checkpoints in the reference
    print format, honoring the checkpoint's synthetic task."""
    from draco_amd.config import Config
    from draco_amd.parallel.trainer import Trainer

    cfg = Config(network="FC", dataset="MNIST", batch_size=16, device="cpu", lr=0.05,
                 approach="baseline", mode="normal", worker_fail=0,
                 synthetic_task="teacher", max_steps=10, eval_freq=0, log_dir="",
                 train_dir=str(tmp_path))
    t = Trainer(cfg)
    t.logger.stdout_every = 0
    for _ in range(4):
        t.train_step()
    t.save()
    t.close()
    out = subprocess.run(
        [sys.executable, "-m", "draco_amd.evaluate", "--model-dir", str(tmp_path),
         "--once", "--eval-batch-size", "32"],
        cwd=REPO, capture_output=True, text=True, timeout=180)
    assert out.returncode == 0, out.stderr[-2000:]
    assert "Prec@1:" in out.stdout and "Prec@5:" in out.stdout
