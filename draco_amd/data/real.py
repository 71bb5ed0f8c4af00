"""Real-dataset loading without torchvision (none in this environment, and no
network): pure-numpy parsers for the standard on-disk formats, exposed through the
same index-addressed `get_batch` contract as the synthetic sources, so every
coding-layer determinism property (group-identical batches, global sub-batches)
holds for real data too.

Formats (parity with the reference's torchvision datasets, util.py:23-66):
  MNIST    IDX ubyte files (train-images-idx3-ubyte / train-labels-idx1-ubyte,
           optionally .gz)
  Cifar10  python-pickle batches (data_batch_1..5 / test_batch)
Point --data-root at a directory containing them; the trainer falls back to the
synthetic stream when the files are absent.
"""
from __future__ import annotations

import gzip
import os
import pickle
import struct

import numpy as np
import torch

MNIST_MEAN, MNIST_STD = 0.1307, 0.3081
CIFAR_MEAN = np.array([125.3, 123.0, 113.9], dtype=np.float32) / 255.0
CIFAR_STD = np.array([63.0, 62.1, 66.7], dtype=np.float32) / 255.0


def _open_maybe_gz(path: str):
    if os.path.exists(path + ".gz"):
        return gzip.open(path + ".gz", "rb")
    return open(path, "rb")


def load_mnist_idx(root: str, train: bool = True):
    """-> images (N,1,28,28) float32 normalised, labels (N,) int64."""
    prefix = "train" if train else "t10k"
    with _open_maybe_gz(os.path.join(root, f"{prefix}-images-idx3-ubyte")) as f:
        magic, n, rows, cols = struct.unpack(">IIII", f.read(16))
        if magic != 2051:
            raise ValueError(f"bad MNIST image magic {magic}")
        images = np.frombuffer(f.read(n * rows * cols), dtype=np.uint8)
        images = images.reshape(n, 1, rows, cols).astype(np.float32) / 255.0
    with _open_maybe_gz(os.path.join(root, f"{prefix}-labels-idx1-ubyte")) as f:
        magic, n2 = struct.unpack(">II", f.read(8))
        if magic != 2049:
            raise ValueError(f"bad MNIST label magic {magic}")
        labels = np.frombuffer(f.read(n2), dtype=np.uint8).astype(np.int64)
    images = (images - MNIST_MEAN) / MNIST_STD
    return images, labels


def load_cifar10(root: str, train: bool = True):
    """-> images (N,3,32,32) float32 normalised, labels (N,) int64."""
    names = [f"data_batch_{i}" for i in range(1, 6)] if train else ["test_batch"]
    xs, ys = [], []
    for name in names:
        path = os.path.join(root, name)
        if not os.path.exists(path):
            path = os.path.join(root, "cifar-10-batches-py", name)
        with open(path, "rb") as f:
            d = pickle.load(f, encoding="bytes")
        xs.append(np.asarray(d[b"data"], dtype=np.uint8))
        ys.append(np.asarray(d[b"labels"], dtype=np.int64))
    x = np.concatenate(xs).reshape(-1, 3, 32, 32).astype(np.float32) / 255.0
    y = np.concatenate(ys)
    x = (x - CIFAR_MEAN[None, :, None, None]) / CIFAR_STD[None, :, None, None]
    return x, y


def dataset_available(dataset: str, root: str) -> bool:
    if not root:
        return False
    try:
        if dataset == "MNIST":
            with _open_maybe_gz(os.path.join(root, "train-images-idx3-ubyte")):
                return True
        if dataset == "Cifar10":
            return (os.path.exists(os.path.join(root, "data_batch_1"))
                    or os.path.exists(os.path.join(root, "cifar-10-batches-py", "data_batch_1")))
    except OSError:
        return False
    return False


class RealClassification:
    """Index-addressed deterministic view of an on-disk dataset (same `get_batch`
    contract as SyntheticClassification; indices wrap modulo the dataset with a
    per-epoch deterministic shuffle, mirroring the reference's seeded reshuffle,
    rep_worker.py:89)."""

    def __init__(self, dataset: str, root: str, device: torch.device, train: bool = True,
                 seed: int = 428, dtype=torch.float32, augment: bool = False):
        if dataset == "MNIST":
            x, y = load_mnist_idx(root, train)
        elif dataset == "Cifar10":
            x, y = load_cifar10(root, train)
        else:
            raise ValueError(f"real data loader supports MNIST/Cifar10, got {dataset}")
        self.x = torch.from_numpy(np.ascontiguousarray(x)).to(device=device, dtype=dtype)
        self.y = torch.from_numpy(np.ascontiguousarray(y)).to(device)
        self.n = len(self.y)
        self.seed = seed
        # train-time augmentation (reference util.py:29-65: RandomCrop(32, pad=4) +
        # RandomHorizontalFlip for CIFAR).  Applied only to 3-channel 32x32 inputs;
        # deterministic per batch start so repetition-group members on different
        # ranks still draw bit-identical (augmented) batches.
        self.augment = augment and train and dataset == "Cifar10"
        self._perm_epoch = -1
        self._perm = None

    def _perm_for(self, epoch: int) -> torch.Tensor:
        if epoch != self._perm_epoch:
            g = torch.Generator()
            g.manual_seed(self.seed + 23 * epoch)  # reference _FACTOR=23 reseed cadence
            self._perm = torch.randperm(self.n, generator=g).to(self.y.device)
            self._perm_epoch = epoch
        return self._perm

    def get_batch(self, start: int, batch: int):
        idx0 = torch.arange(start, start + batch, device=self.y.device)
        epoch = int(start // self.n)
        pos = idx0 % self.n
        # a batch can straddle an epoch boundary; map each half through its epoch's perm
        if int((start + batch - 1) // self.n) != epoch:
            sel = torch.empty(batch, dtype=torch.int64, device=self.y.device)
            for e in (epoch, epoch + 1):
                m = (idx0 // self.n) == e
                sel[m] = self._perm_for(e)[pos[m]]
        else:
            sel = self._perm_for(epoch)[pos]
        x, y = self.x[sel], self.y[sel]
        if self.augment:
            x = self._augment_batch(x, start)
        return x, y

    def _augment_batch(self, x: torch.Tensor, start: int) -> torch.Tensor:
        """RandomCrop(32, padding=4) + RandomHorizontalFlip, on device, seeded by the
        batch start index — a pure function of (data, start) like get_batch itself."""
        B, C, H, W = x.shape
        g = torch.Generator(device=x.device)
        g.manual_seed((self.seed * 0x9E3779B1 + start * 2654435761) & ((1 << 63) - 1))
        xp = torch.nn.functional.pad(x, (4, 4, 4, 4))
        offs = torch.randint(0, 9, (B, 2), generator=g, device=x.device)
        flip = torch.rand(B, generator=g, device=x.device) < 0.5
        ar = torch.arange(H, device=x.device)
        idx_r = (offs[:, 0].view(B, 1, 1, 1) + ar.view(1, 1, H, 1)).expand(B, C, H, W + 8)
        xr = xp.gather(2, idx_r)
        idx_c = (offs[:, 1].view(B, 1, 1, 1) + ar.view(1, 1, 1, W)).expand(B, C, H, W)
        xc = xr.gather(3, idx_c)
        return torch.where(flip.view(B, 1, 1, 1), xc.flip(3), xc)


class PrefetchLoader:
    """Double-buffered prefetch over an index-addressed source: batch k+1 is staged
    (host->device on a side stream when on GPU) while batch k is consumed — the
    replacement for the reference's multiprocess DataLoader fork
    (data_loader_ops/my_data_loader.py), which existed for CPU-side image decode."""

    def __init__(self, source, batch_size: int, device: torch.device):
        self.source = source
        self.B = batch_size
        self.device = device
        self._stream = torch.cuda.Stream() if device.type == "cuda" else None
        self._next = None
        self._next_start = None

    def _stage(self, start: int):
        if self._stream is not None:
            with torch.cuda.stream(self._stream):
                self._next = self.source.get_batch(start, self.B)
        else:
            self._next = self.source.get_batch(start, self.B)
        self._next_start = start

    def get(self, start: int):
        if self._next_start == start:
            if self._stream is not None:
                torch.cuda.current_stream().wait_stream(self._stream)
                # staged tensors were allocated on the side stream; record their use
                # on the consumer stream so the allocator cannot recycle them early
                for t in self._next:
                    t.record_stream(torch.cuda.current_stream())
            out = self._next
        else:
            out = self.source.get_batch(start, self.B)
        self._stage(start + self.B)  # prefetch the sequential successor
        return out
