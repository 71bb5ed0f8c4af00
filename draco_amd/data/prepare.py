"""Dataset preparation entry point (parity with the reference's pre-download step,
/root/reference/src/datasets/data_prepare.py — which fetches MNIST/CIFAR via
torchvision).  This environment has no network, so two modes:

  python -m draco_amd.data.prepare --root DIR                # verify files present
  python -m draco_amd.data.prepare --root DIR --synthesize   # write deterministic
        sample files in the REAL on-disk formats (MNIST IDX + CIFAR pickle batches)
        so the full real-data pipeline (parsers, epoch shuffle, augmentation,
        PrefetchLoader) runs end-to-end without a download

With network access, drop the standard files (train-images-idx3-ubyte[.gz] etc. /
cifar-10-batches-py) into --root and every `--data-root DIR` run uses them.
"""
from __future__ import annotations

import argparse
import gzip
import os
import pickle
import struct

import numpy as np

from .real import dataset_available


def synthesize_mnist(root: str, n_train: int = 4096, n_test: int = 1024, seed: int = 0) -> None:
    rng = np.random.default_rng(seed)
    for prefix, n in (("train", n_train), ("t10k", n_test)):
        imgs = rng.integers(0, 256, size=(n, 28, 28), dtype=np.uint8)
        labels = rng.integers(0, 10, size=n, dtype=np.uint8)
        with open(os.path.join(root, f"{prefix}-images-idx3-ubyte"), "wb") as f:
            f.write(struct.pack(">IIII", 2051, n, 28, 28))
            f.write(imgs.tobytes())
        with gzip.open(os.path.join(root, f"{prefix}-labels-idx1-ubyte.gz"), "wb") as f:
            f.write(struct.pack(">II", 2049, n))
            f.write(labels.tobytes())


def synthesize_cifar(root: str, per_batch: int = 1024, seed: int = 1) -> None:
    rng = np.random.default_rng(seed)
    for name in [f"data_batch_{i}" for i in range(1, 6)] + ["test_batch"]:
        data = rng.integers(0, 256, size=(per_batch, 3072), dtype=np.uint8)
        labels = rng.integers(0, 10, size=per_batch).tolist()
        with open(os.path.join(root, name), "wb") as f:
            pickle.dump({b"data": data, b"labels": labels}, f)


def main(argv=None) -> int:
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--root", type=str, required=True)
    p.add_argument("--dataset", type=str, default="all", choices=["MNIST", "Cifar10", "all"])
    p.add_argument("--synthesize", action="store_true",
                   help="write deterministic sample files in the real formats")
    args = p.parse_args(argv)
    os.makedirs(args.root, exist_ok=True)
    want = ["MNIST", "Cifar10"] if args.dataset == "all" else [args.dataset]
    if args.synthesize:
        if "MNIST" in want:
            synthesize_mnist(args.root)
        if "Cifar10" in want:
            synthesize_cifar(args.root)
    rc = 0
    for ds in want:
        ok = dataset_available(ds, args.root)
        print(f"{ds}: {'ready' if ok else 'MISSING'} in {args.root}")
        rc |= 0 if ok else 1
    return rc


if __name__ == "__main__":
    raise SystemExit(main())
