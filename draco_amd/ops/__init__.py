"""Op dispatch: hand-written HIP/CDNA4 kernels on GPU, pure-torch on CPU.

Every op has identical semantics in both paths; the CPU path doubles as the
numerics oracle for the GPU kernels (tests/test_kernels_gpu.py).
"""
from __future__ import annotations

import torch

from . import fallback
from .native import available, get_extension, require_extension

__all__ = [
    "fused_sgd_step",
    "fused_adam_step",
    "inject_",
    "rows_equal",
    "pair_maxdiff",
    "row_absmax",
    "mean_rows",
    "sum_rows",
    "cyclic_encode",
    "combine_rows",
    "cyclic_project",
    "cyclic_recombine",
    "segment_absmax",
    "segment_pair_maxdiff",
    "segment_sqdist",
    "segment_weighted_mean",
    "segment_gram",
    "available",
]


def _native_for(t: torch.Tensor):
    """Return the HIP extension for CUDA tensors (or None -> fallback allowed)."""
    if t.is_cuda:
        return require_extension()
    return None


def fused_sgd_step(param, grad, momentum_buf, *, lr, momentum, dampening, weight_decay,
                   nesterov, first_step, guard=None):
    """guard: optional 0-dim bool tensor ON DEVICE; False -> the update is a no-op.
    Lets the nan_guard skip poisoned updates without a host round-trip."""
    ext = _native_for(param)
    if ext is not None:
        ext.fused_sgd_step(
            param, grad,
            momentum_buf if momentum_buf is not None else param.new_empty(0),
            lr, momentum, dampening, weight_decay, bool(nesterov), bool(first_step),
            guard if guard is not None else param.new_empty(0, dtype=torch.bool),
        )
        return
    fallback.fused_sgd_step(
        param, grad, momentum_buf,
        lr=lr, momentum=momentum, dampening=dampening,
        weight_decay=weight_decay, nesterov=nesterov, first_step=first_step, guard=guard,
    )


def fused_adam_step(param, grad, exp_avg, exp_avg_sq, max_exp_avg_sq, *, step, lr, beta1,
                    beta2, eps, weight_decay, amsgrad, guard=None):
    ext = _native_for(param)
    if ext is not None:
        ext.fused_adam_step(
            param, grad, exp_avg, exp_avg_sq,
            max_exp_avg_sq if max_exp_avg_sq is not None else param.new_empty(0),
            int(step), lr, beta1, beta2, eps, weight_decay, bool(amsgrad),
            guard if guard is not None else param.new_empty(0, dtype=torch.bool),
        )
        return
    fallback.fused_adam_step(
        param, grad, exp_avg, exp_avg_sq, max_exp_avg_sq,
        step=step, lr=lr, beta1=beta1, beta2=beta2, eps=eps,
        weight_decay=weight_decay, amsgrad=amsgrad, guard=guard,
    )


def inject_(grad, mode, cyclic=False):
    ext = _native_for(grad)
    if mode in ("random", "none", ""):  # passthrough modes never touch memory
        return
    if ext is not None and mode in ("rev_grad", "constant"):
        ext.inject(grad, mode, bool(cyclic))
        return
    fallback.inject_(grad, mode, cyclic)


def rows_equal(x, a_idx, b_idx, atol):
    ext = _native_for(x)
    if ext is not None:
        return ext.rows_equal(x, a_idx.to(x.device), b_idx.to(x.device), float(atol))
    return fallback.rows_equal(x, a_idx, b_idx, atol)


def pair_maxdiff(x, a_idx, b_idx):
    ext = _native_for(x)
    if ext is not None:
        return ext.pair_maxdiff(x, a_idx.to(x.device), b_idx.to(x.device))
    return fallback.pair_maxdiff(x, a_idx, b_idx)


def row_absmax(x):
    ext = _native_for(x)
    if ext is not None:
        return ext.row_absmax(x)
    return fallback.row_absmax(x)


def mean_rows(x, idx, out):
    ext = _native_for(x)
    if ext is not None:
        ext.mean_rows(x, idx.to(x.device), out)
        return
    fallback.mean_rows(x, idx, out)


def sum_rows(x, out):
    ext = _native_for(x)
    if ext is not None:
        ext.sum_rows(x, out)
        return
    fallback.sum_rows(x, out)


def cyclic_encode(grads, w_re, w_im, out):
    ext = _native_for(grads)
    if ext is not None:
        ext.cyclic_encode(grads, w_re.to(grads.device), w_im.to(grads.device), out)
        return
    fallback.cyclic_encode(grads, w_re, w_im, out)


def cyclic_project(rows, z):
    ext = _native_for(rows)
    if ext is not None:
        return ext.cyclic_project(rows, z)
    return fallback.cyclic_project(rows, z)


def combine_rows(x, rows, w, out):
    ext = _native_for(x)
    if ext is not None:
        ext.combine_rows(x, rows.to(x.device), w.to(x.device), out)
        return
    fallback.combine_rows(x, rows, w, out)


def combine_rows_slice(x, rows, w, out, col_off):
    """out = sum_i w[i] * x[rows[i], col_off : col_off + len(out)] (bucketed encode)."""
    ext = _native_for(x)
    if ext is not None:
        ext.combine_rows_slice(x, rows.to(x.device), w.to(x.device), out, int(col_off))
        return
    fallback.combine_rows_slice(x, rows, w, out, col_off)


def cyclic_recombine(r_planes, v_re, v_im, out):
    ext = _native_for(r_planes)
    if ext is not None:
        ext.cyclic_recombine(r_planes, v_re.to(r_planes.device), v_im.to(r_planes.device), out)
        return
    fallback.cyclic_recombine(r_planes, v_re, v_im, out)


def segment_absmax(x, seg):
    ext = _native_for(x)
    if ext is not None:
        return ext.segment_absmax(x, seg.to(x.device))
    return fallback.segment_absmax(x, seg)


def segment_pair_maxdiff(x, a_idx, b_idx, seg):
    ext = _native_for(x)
    if ext is not None:
        return ext.segment_pair_maxdiff(x, a_idx.to(x.device), b_idx.to(x.device), seg.to(x.device))
    return fallback.segment_pair_maxdiff(x, a_idx, b_idx, seg)


def segment_sqdist(x, z, seg):
    ext = _native_for(x)
    if ext is not None:
        return ext.segment_sqdist(x, z, seg.to(x.device))
    return fallback.segment_sqdist(x, z, seg)


def segment_weighted_mean(x, w, seg, out):
    ext = _native_for(x)
    if ext is not None:
        ext.segment_weighted_mean(x, w, seg.to(x.device), out)
        return
    fallback.segment_weighted_mean(x, w, seg, out)


def segment_gram(x, seg):
    ext = _native_for(x)
    if ext is not None:
        return ext.segment_gram(x, seg.to(x.device))
    return fallback.segment_gram(x, seg)
