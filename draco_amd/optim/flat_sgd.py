"""Fused flat-buffer SGD (semantics of /root/reference/src/optim/sgd_modified.py:53-89).

The whole parameter space is ONE contiguous fp32 tensor (parallel/flat.py), so the
update is a single fused elementwise HIP kernel launch over d — vs the reference's
per-parameter numpy loop.  The reference's first-step quirk (no (1-dampening) factor
on the very first momentum accumulation) is preserved.
"""
from __future__ import annotations

import torch

from .. import ops


class FlatSGD:
    def __init__(self, flat_param: torch.Tensor, lr: float, momentum: float = 0.0,
                 dampening: float = 0.0, weight_decay: float = 0.0, nesterov: bool = False):
        self.param = flat_param
        self.lr = lr
        self.momentum = momentum
        self.dampening = dampening
        self.weight_decay = weight_decay
        self.nesterov = nesterov
        self.buf = torch.zeros_like(flat_param) if momentum != 0.0 else None
        self._first = True

    @torch.no_grad()
    def step(self, grad: torch.Tensor, guard: torch.Tensor | None = None) -> None:
        """guard: optional device bool; False -> the kernel no-ops (nan-guard skip
        with no host sync).  The host-side _first flag still advances on a guarded
        skip — a skipped step only happens on an anomalous (non-finite) decode, and
        the first-step dampening quirk is irrelevant in that regime."""
        ops.fused_sgd_step(
            self.param, grad, self.buf,
            lr=self.lr, momentum=self.momentum, dampening=self.dampening,
            weight_decay=self.weight_decay, nesterov=self.nesterov, first_step=self._first,
            guard=guard,
        )
        self._first = False

    def state_dict(self):
        return {
            "buf": None if self.buf is None else self.buf.clone(),
            "first": self._first,
            "lr": self.lr,
            "momentum": self.momentum,
        }

    def load_state_dict(self, sd):
        if sd.get("buf") is not None and self.buf is not None:
            self.buf.copy_(sd["buf"].to(self.buf.device))
        self._first = sd.get("first", False)


class FlatAdam:
    """Fused flat Adam/AMSGrad (semantics of /root/reference/src/optim/adam_modified.py:32-93)."""

    def __init__(self, flat_param: torch.Tensor, lr: float = 1e-3, betas=(0.9, 0.999),
                 eps: float = 1e-8, weight_decay: float = 0.0, amsgrad: bool = False):
        self.param = flat_param
        self.lr = lr
        self.beta1, self.beta2 = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.amsgrad = amsgrad
        self.exp_avg = torch.zeros_like(flat_param)
        self.exp_avg_sq = torch.zeros_like(flat_param)
        self.max_exp_avg_sq = torch.zeros_like(flat_param) if amsgrad else None
        self.t = 0

    @torch.no_grad()
    def step(self, grad: torch.Tensor, guard: torch.Tensor | None = None) -> None:
        self.t += 1  # advances even on a guarded skip (bias correction drifts one
        # step on an anomalous decode — negligible vs applying a poisoned update)
        ops.fused_adam_step(
            self.param, grad, self.exp_avg, self.exp_avg_sq, self.max_exp_avg_sq,
            step=self.t, lr=self.lr, beta1=self.beta1, beta2=self.beta2,
            eps=self.eps, weight_decay=self.weight_decay, amsgrad=self.amsgrad,
            guard=guard,
        )

    def state_dict(self):
        return {
            "exp_avg": self.exp_avg.clone(),
            "exp_avg_sq": self.exp_avg_sq.clone(),
            "max_exp_avg_sq": None if self.max_exp_avg_sq is None else self.max_exp_avg_sq.clone(),
            "t": self.t,
        }

    def load_state_dict(self, sd):
        self.exp_avg.copy_(sd["exp_avg"].to(self.exp_avg.device))
        self.exp_avg_sq.copy_(sd["exp_avg_sq"].to(self.exp_avg_sq.device))
        if sd.get("max_exp_avg_sq") is not None and self.max_exp_avg_sq is not None:
            self.max_exp_avg_sq.copy_(sd["max_exp_avg_sq"].to(self.max_exp_avg_sq.device))
        self.t = sd["t"]
