"""CLI parity: every reference flag name is accepted (README.md:104-121 /
distributed_nn.py:23-77), plus the single-machine runner."""
import subprocess
import sys

import pytest

from draco_amd.config import parse_cli


REFERENCE_FLAGS = [
    ("--batch-size", "32"),
    ("--test-batch-size", "50"),
    ("--max-steps", "10"),
    ("--epochs", "2"),
    ("--lr", "0.1"),
    ("--momentum", "0.9"),
    ("--seed", "7"),
    ("--network", "ResNet18"),
    ("--mode", "maj_vote"),
    ("--dataset", "Cifar10"),
    ("--comm-type", "Bcast"),
    ("--err-mode", "constant"),
    ("--approach", "maj_vote"),
    ("--num-aggregate", "5"),
    ("--eval-freq", "20"),
    ("--train-dir", "/tmp/x"),
    ("--adversarial", "1"),
    ("--worker-fail", "1"),
    ("--group-size", "3"),
    ("--compress-grad", "compress"),
    ("--checkpoint-step", "0"),
    ("--hostfile", "hosts"),
    ("--log-interval", "5"),
]


def test_reference_flags_accepted():
    argv = [a for pair in REFERENCE_FLAGS for a in pair]
    cfg = parse_cli(argv)
    assert cfg.network == "ResNet18"
    assert cfg.batch_size == 32
    assert cfg.err_mode == "constant"
    assert cfg.compress_grad == "compress"


def test_extension_flags():
    cfg = parse_cli(["--topology", "ps", "--vote-rtol", "0.05", "--dtype", "fp32",
                     "--device", "cpu", "--hip-graphs", "false",
                     "--straggler-timeout", "2.5", "--optimizer", "adam"])
    assert cfg.topology == "ps" and cfg.vote_rtol == 0.05
    assert cfg.straggler_timeout == 2.5 and cfg.optimizer == "adam"
    assert cfg.hip_graphs is False


def test_single_machine_runs():
    out = subprocess.run(
        [sys.executable, "-m", "draco_amd.single_machine", "--network", "FC",
         "--dataset", "MNIST", "--batch-size", "4", "--max-steps", "3",
         "--approach", "maj_vote", "--group-size", "3", "--worker-fail", "1",
         "--device", "cpu", "--eval-freq", "0", "--log-dir", "", "--train-dir", "/tmp/sm"],
        capture_output=True, text=True, timeout=180,
    )
    assert out.returncode == 0, out.stderr[-2000:]


def test_train_cli_runs():
    out = subprocess.run(
        [sys.executable, "-m", "draco_amd.train", "--network", "FC", "--dataset", "MNIST",
         "--batch-size", "4", "--max-steps", "3", "--approach", "cyclic",
         "--worker-fail", "1", "--workers-per-rank", "4", "--device", "cpu",
         "--eval-freq", "0", "--log-dir", "", "--train-dir", "/tmp/sm2"],
        capture_output=True, text=True, timeout=180,
    )
    assert out.returncode == 0, out.stderr[-2000:]
