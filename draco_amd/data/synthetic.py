"""Deterministic synthetic datasets (no network access in this environment).

Index-addressed like the reference's deterministic global batches
(/root/reference/src/datasets/utils.py:21-29 `get_batch`): batch contents are a pure
function of (seed, indices), so
  * repetition-code group members on DIFFERENT ranks draw bit-identical batches
    (the group-seed mechanism of util.py:69-97 / rep_worker.py:89), and
  * cyclic-code sub-batches are globally addressable slices, exactly like
    cyclic_worker.py:94.

The task is learnable: labels are drawn from the index stream and images are
class-mean + noise (fixed class means), so convergence-under-attack experiments mirror
the reference methodology without torchvision downloads.
"""
from __future__ import annotations

import torch

from ..models import dataset_shape


class SyntheticClassification:
    """Infinite deterministic synthetic image stream, two tasks:

    task="means"   class-conditional Gaussians (class-mean + noise).  Separable —
                   a discrimination demo for adversary experiments; clean training
                   saturates Prec@1 quickly.
    task="teacher" labels are argmax of a fixed random 2-layer teacher network on
                   N(0,1) inputs — a genuinely non-trivial decision boundary the
                   student must LEARN (clean Prec@1 climbs through ~0.5-0.8 over
                   hundreds of steps instead of saturating), used for the
                   convergence-under-attack evidence (tools/convergence.py).
    """

    def __init__(self, dataset: str, device: torch.device, seed: int = 1234, noise: float = 1.0,
                 dtype=torch.float32, task: str = "means"):
        c, h, w, classes = dataset_shape(dataset)
        self.shape = (c, h, w)
        self.classes = classes
        self.device = device
        self.noise = noise
        self.dtype = dtype
        self.task = task
        g = torch.Generator(device="cpu")
        g.manual_seed(seed)
        # fixed class means, modest separation so training has to work for it
        self.means = (torch.randn(classes, c, h, w, generator=g) * 0.7).to(device=device, dtype=dtype)
        if task == "teacher":
            din = c * h * w
            hid = 64
            # Kaiming-ish scaling keeps the teacher logits O(1); the margin
            # distribution makes a fraction of samples genuinely hard
            self.t_w1 = (torch.randn(din, hid, generator=g) / din ** 0.5).to(device, dtype)
            self.t_b1 = (torch.randn(hid, generator=g) * 0.1).to(device, dtype)
            self.t_w2 = (torch.randn(hid, classes, generator=g) / hid ** 0.5).to(device, dtype)
        elif task != "means":
            raise ValueError(f"unknown synthetic task {task!r}")

    @staticmethod
    def _mix(start: int) -> int:
        """splitmix64 of the full 64-bit start index: distinct starts map to
        distinct seeds with no structured collisions (the previous xor-fold let the
        2**40 held-out eval offset collide with small training starts)."""
        m = (1 << 64) - 1
        z = (start + 0x9E3779B97F4A7C15) & m
        z = ((z ^ (z >> 30)) * 0xBF58476D1CE4E5B9) & m
        z = ((z ^ (z >> 27)) * 0x94D049BB133111EB) & m
        return (z ^ (z >> 31)) & ((1 << 63) - 1)  # manual_seed wants non-negative int64

    def get_batch(self, start: int, batch: int):
        """Deterministic batch for global sample indices [start, start+batch).

        Generated ON DEVICE (Philox counter-based, seeded by the index block), so
        group members on different GPUs draw bit-identical batches with no host
        round-trip — generation costs <1 ms vs ~25 ms for host randn + H2D copy.
        """
        seed = self._mix(start)
        g = torch.Generator(device=self.device)
        g.manual_seed(seed)
        if self.task == "teacher":
            x = torch.randn(batch, *self.shape, generator=g, device=self.device,
                            dtype=self.dtype)
            h = torch.relu(x.reshape(batch, -1) @ self.t_w1 + self.t_b1)
            y = (h @ self.t_w2).argmax(dim=1)
            return x, y
        y = torch.randint(0, self.classes, (batch,), generator=g, device=self.device)
        x = torch.randn(batch, *self.shape, generator=g, device=self.device,
                        dtype=self.dtype) * self.noise
        x += self.means[y]
        return x, y


class GroupBatchSource:
    """Batch stream keyed by (group, step): all members of a group draw the identical
    batch (repetition code), with disjoint data across groups and steps."""

    def __init__(self, data: SyntheticClassification, batch_size: int, n_groups: int):
        self.data = data
        self.B = batch_size
        self.G = n_groups

    def batch_for(self, group: int, step: int):
        start = (step * self.G + group) * self.B
        return self.data.get_batch(start, self.B)


class GlobalBatchSource:
    """Cyclic-code stream: a global batch of n*B samples per step, sub-batch j is the
    j-th slice (cyclic_worker.py:94,122-126)."""

    def __init__(self, data: SyntheticClassification, batch_size: int, n_workers: int):
        self.data = data
        self.B = batch_size
        self.n = n_workers

    def sub_batch(self, sub_index: int, step: int):
        start = (step * self.n + sub_index) * self.B
        return self.data.get_batch(start, self.B)
