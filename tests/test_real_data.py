"""Real-dataset parsers (IDX / CIFAR pickle), deterministic epoch shuffle, prefetch."""
import gzip
import os
import pickle
import struct

import pytest

import numpy as np
import torch

from draco_amd.data.real import (
    PrefetchLoader,
    RealClassification,
    dataset_available,
    load_cifar10,
    load_mnist_idx,
)


def _write_mnist(root, n=64, prefix="train", seed=0):
    rng = np.random.default_rng(seed)
    imgs = rng.integers(0, 256, size=(n, 28, 28), dtype=np.uint8)
    labels = rng.integers(0, 10, size=n, dtype=np.uint8)
    with open(os.path.join(root, f"{prefix}-images-idx3-ubyte"), "wb") as f:
        f.write(struct.pack(">IIII", 2051, n, 28, 28))
        f.write(imgs.tobytes())
    with gzip.open(os.path.join(root, f"{prefix}-labels-idx1-ubyte.gz"), "wb") as f:
        f.write(struct.pack(">II", 2049, n))
        f.write(labels.tobytes())
    return imgs, labels


def _write_cifar(root, n=50):
    rng = np.random.default_rng(1)
    for b in range(1, 6):
        data = rng.integers(0, 256, size=(n, 3072), dtype=np.uint8)
        labels = rng.integers(0, 10, size=n).tolist()
        with open(os.path.join(root, f"data_batch_{b}"), "wb") as f:
            pickle.dump({b"data": data, b"labels": labels}, f)


def test_mnist_idx_roundtrip(tmp_path):
    root = str(tmp_path)
    imgs, labels = _write_mnist(root)
    assert dataset_available("MNIST", root)
    x, y = load_mnist_idx(root, train=True)
    assert x.shape == (64, 1, 28, 28) and y.shape == (64,)
    assert (y == labels.astype(np.int64)).all()
    # normalisation round-trips
    raw = x[0, 0] * 0.3081 + 0.1307
    assert np.allclose(raw * 255.0, imgs[0], atol=0.51)


def test_cifar_pickle(tmp_path):
    root = str(tmp_path)
    _write_cifar(root)
    assert dataset_available("Cifar10", root)
    x, y = load_cifar10(root, train=True)
    assert x.shape == (250, 3, 32, 32) and y.shape == (250,)


def test_real_batches_deterministic(tmp_path):
    root = str(tmp_path)
    _write_mnist(root)
    a = RealClassification("MNIST", root, torch.device("cpu"))
    b = RealClassification("MNIST", root, torch.device("cpu"))
    xa, ya = a.get_batch(10, 16)
    xb, yb = b.get_batch(10, 16)
    assert torch.equal(xa, xb) and torch.equal(ya, yb)
    # epoch boundary straddle (n=64): indices 60..76 span two epochs
    xs, ys = a.get_batch(60, 16)
    assert xs.shape == (16, 1, 28, 28)
    # different epochs shuffle differently
    x0, _ = a.get_batch(0, 16)
    x1, _ = a.get_batch(64, 16)
    assert not torch.equal(x0, x1)


def test_prefetch_loader_matches_direct(tmp_path):
    root = str(tmp_path)
    _write_mnist(root)
    src = RealClassification("MNIST", root, torch.device("cpu"))
    loader = PrefetchLoader(src, batch_size=8, device=torch.device("cpu"))
    for start in (0, 8, 16, 40):  # sequential + a jump
        x, y = loader.get(start)
        xr, yr = src.get_batch(start, 8)
        assert torch.equal(x, xr) and torch.equal(y, yr)


def test_trainer_uses_real_data(tmp_path):
    from draco_amd.config import Config
    from draco_amd.parallel.trainer import Trainer
    from draco_amd.data.real import RealClassification

    root = str(tmp_path)
    _write_mnist(root)
    cfg = Config(network="FC", dataset="MNIST", batch_size=4, device="cpu", lr=0.05,
                 approach="maj_vote", mode="maj_vote", group_size=3, worker_fail=1,
                 data_root=root, max_steps=20, eval_freq=0, log_dir="",
                 train_dir=str(tmp_path / "ck"))
    t = Trainer(cfg)
    t.logger.stdout_every = 0
    assert isinstance(t.data.data, RealClassification)
    rec = t.train_step()
    assert np.isfinite(rec["loss"])
    t.close()


def test_cifar_augmentation(tmp_path):
    """RandomCrop(32, pad=4) + flip (reference util.py:29-65): deterministic per
    start index, shape-preserving, and every output is a genuine crop/flip of the
    padded original."""
    root = str(tmp_path)
    _write_cifar(root)
    plain = RealClassification("Cifar10", root, torch.device("cpu"), augment=False)
    aug = RealClassification("Cifar10", root, torch.device("cpu"), augment=True)
    aug2 = RealClassification("Cifar10", root, torch.device("cpu"), augment=True)
    xa, ya = aug.get_batch(0, 16)
    xb, yb = aug2.get_batch(0, 16)
    xp, yp = plain.get_batch(0, 16)
    assert torch.equal(xa, xb) and torch.equal(ya, yb), "augmentation not deterministic"
    assert torch.equal(ya, yp), "labels must be untouched"
    assert xa.shape == xp.shape
    assert not torch.equal(xa, xp), "augmentation was a no-op"
    # brute-force verify sample 0 is one of the 9*9*2 crop/flip variants
    src = torch.nn.functional.pad(xp[0:1], (4, 4, 4, 4))[0]
    found = False
    for dy in range(9):
        for dx in range(9):
            crop = src[:, dy : dy + 32, dx : dx + 32]
            if torch.equal(xa[0], crop) or torch.equal(xa[0], crop.flip(2)):
                found = True
    assert found, "augmented sample is not a crop/flip of the original"


def test_mnist_augment_is_noop(tmp_path):
    root = str(tmp_path)
    _write_mnist(root)
    a = RealClassification("MNIST", root, torch.device("cpu"), augment=True)
    b = RealClassification("MNIST", root, torch.device("cpu"), augment=False)
    xa, _ = a.get_batch(0, 8)
    xb, _ = b.get_batch(0, 8)
    assert torch.equal(xa, xb)


@pytest.mark.gpu
def test_real_data_training_on_gpu(tmp_path):
    """Real-file lane on the GPU: parsers + device-side augmentation + trainer."""
    from draco_amd.config import Config
    from draco_amd.parallel.trainer import Trainer
    from draco_amd.data.real import RealClassification

    root = str(tmp_path)
    _write_cifar(root)
    cfg = Config(network="ResNet18", dataset="Cifar10", batch_size=32, device="cuda",
                 lr=0.02, approach="maj_vote", mode="maj_vote", group_size=3,
                 worker_fail=1, data_root=root, max_steps=20, eval_freq=0, log_dir="",
                 train_dir=str(tmp_path / "ck"))
    t = Trainer(cfg)
    t.logger.stdout_every = 0
    assert isinstance(t.data.data, RealClassification)
    assert t.data.data.augment
    for _ in range(3):
        rec = t.train_step()
    assert np.isfinite(rec["loss"])
    assert t.skipped_updates == 0
    t.close()


def test_evaluate_uses_test_split(tmp_path):
    """Regression (round-1 advisor, medium): evaluate() must score the HELD-OUT
    split, not training data."""
    from draco_amd.config import Config
    from draco_amd.parallel.trainer import Trainer

    root = str(tmp_path)
    _write_mnist(root, prefix="train", seed=0)
    test_imgs, test_labels = _write_mnist(root, prefix="t10k", seed=99)
    cfg = Config(network="FC", dataset="MNIST", batch_size=4, device="cpu", lr=0.05,
                 approach="baseline", mode="normal", worker_fail=0, data_root=root,
                 test_batch_size=8, max_steps=20, eval_freq=0, log_dir="",
                 train_dir=str(tmp_path / "ck"))
    t = Trainer(cfg)
    t.logger.stdout_every = 0
    held = t._dataset(train=False)
    x, y = held.get_batch(0, 8)
    # the held-out source must serve t10k content (seed 99), not train content
    norm = (test_imgs.astype(np.float32) / 255.0 - 0.1307) / 0.3081
    pool = torch.from_numpy(norm).unsqueeze(1)
    match = any(torch.allclose(x[0], pool[i], atol=1e-5) for i in range(len(pool)))
    assert match, "evaluate() source does not serve the test split"
    m = t.evaluate(n_batches=2)
    assert 0.0 <= m["prec1"] <= 1.0
    t.close()


def test_data_prepare_tool(tmp_path):
    """Reference-parity dataset prep entry point (data_prepare.py): synthesize mode
    writes real-format files that the loaders then accept."""
    from draco_amd.data.prepare import main

    root = str(tmp_path / "d")
    assert main(["--root", root, "--synthesize"]) == 0
    assert dataset_available("MNIST", root) and dataset_available("Cifar10", root)
    x, y = load_mnist_idx(root, train=False)
    assert x.shape[0] == 1024
    x2, y2 = load_cifar10(root, train=True)
    assert x2.shape[0] == 5 * 1024
    # verify-only mode on an empty dir reports missing
    assert main(["--root", str(tmp_path / "empty")]) != 0
