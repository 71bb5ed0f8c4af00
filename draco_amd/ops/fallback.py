"""Pure-PyTorch implementations of every Draco op.

These are (a) the CPU execution path for tests/gloo lanes and (b) the numerics oracle
the HIP kernels (ops/csrc/) are tested against.  Semantics mirror the reference:
  - fused_sgd_step:  /root/reference/src/optim/sgd_modified.py:53-89
  - fused_adam_step: /root/reference/src/optim/adam_modified.py:32-93
  - inject_:         /root/reference/src/model_ops/utils.py:6-23
  - rows_equal / mean_rows: the majority-vote decode, rep_master.py:154-168
  - encode/proj/recombine:  cyclic code hot GEMVs, cyclic_worker.py:165-194 and
    cyclic_master.py:146-173
"""
from __future__ import annotations

import torch

ADVERSARY_ = -100.0  # reference: model_ops/utils.py:3


# --------------------------------------------------------------------- optimizers
@torch.no_grad()
def fused_sgd_step(
    param: torch.Tensor,
    grad: torch.Tensor,
    momentum_buf: torch.Tensor | None,
    *,
    lr: float,
    momentum: float,
    dampening: float,
    weight_decay: float,
    nesterov: bool,
    first_step: bool,
    guard: torch.Tensor | None = None,
) -> None:
    if guard is not None and not bool(guard):
        return  # nan-guard skip (host read is fine on the CPU path)
    d_p = grad
    if weight_decay != 0.0:
        d_p = d_p.add(param, alpha=weight_decay)
    if momentum != 0.0:
        assert momentum_buf is not None
        if first_step:
            # reference quirk kept deliberately: the very first momentum update has no
            # (1 - dampening) factor (sgd_modified.py:80-82)
            momentum_buf.mul_(momentum).add_(d_p)
        else:
            momentum_buf.mul_(momentum).add_(d_p, alpha=1.0 - dampening)
        if nesterov:
            d_p = d_p.add(momentum_buf, alpha=momentum)
        else:
            d_p = momentum_buf
    param.add_(d_p, alpha=-lr)


@torch.no_grad()
def fused_adam_step(
    param: torch.Tensor,
    grad: torch.Tensor,
    exp_avg: torch.Tensor,
    exp_avg_sq: torch.Tensor,
    max_exp_avg_sq: torch.Tensor | None,
    *,
    step: int,
    lr: float,
    beta1: float,
    beta2: float,
    eps: float,
    weight_decay: float,
    amsgrad: bool,
    guard: torch.Tensor | None = None,
) -> None:
    if guard is not None and not bool(guard):
        return  # nan-guard skip
    if weight_decay != 0.0:
        grad = grad.add(param, alpha=weight_decay)
    exp_avg.mul_(beta1).add_(grad, alpha=1.0 - beta1)
    exp_avg_sq.mul_(beta2).addcmul_(grad, grad, value=1.0 - beta2)
    bias_c1 = 1.0 - beta1**step
    bias_c2 = 1.0 - beta2**step
    if amsgrad:
        assert max_exp_avg_sq is not None
        torch.maximum(max_exp_avg_sq, exp_avg_sq, out=max_exp_avg_sq)
        denom = max_exp_avg_sq.sqrt().add_(eps)
    else:
        denom = exp_avg_sq.sqrt().add_(eps)
    step_size = lr * (bias_c2**0.5) / bias_c1
    param.addcdiv_(exp_avg, denom, value=-step_size)


# --------------------------------------------------------------------- adversary
@torch.no_grad()
def inject_(grad: torch.Tensor, mode: str, cyclic: bool = False) -> None:
    """Byzantine fault injection at the send boundary, in place.

    reference err_simulation: rev_grad -> -100*g (replace), constant -> -100 fill,
    random -> passthrough (left TODO upstream; kept as passthrough for parity, with
    'gauss' provided as a real randomising mode).  cyclic=True adds the error to the
    gradient instead of replacing it (model_ops/utils.py:8-20).
    """
    if mode == "rev_grad":
        err = grad * ADVERSARY_
    elif mode == "constant":
        err = torch.full_like(grad, ADVERSARY_)
    elif mode == "random":
        return  # reference passthrough (model_ops/utils.py:21-23 TODO)
    elif mode == "gauss":
        err = torch.randn_like(grad) * grad.abs().mean().clamp(min=1e-12) * 100.0
    elif mode in ("none", ""):
        return
    else:
        raise ValueError(f"unknown err mode {mode!r}")
    if cyclic:
        grad.add_(err)
    else:
        grad.copy_(err)


# --------------------------------------------------------------------- vote
@torch.no_grad()
def rows_equal(x: torch.Tensor, a_idx: torch.Tensor, b_idx: torch.Tensor, atol: float) -> torch.Tensor:
    """Per-pair equality of rows of x over the local shard.

    x: (m, d) fp32; a_idx/b_idx: (k,) int64.  Returns (k,) uint8 — 1 iff
    max|x[a]-x[b]| <= atol over d (atol=0 -> bitwise-identical semantics of the
    reference's np.array_equal vote).
    """
    if x.shape[1] == 0:
        return torch.ones(len(a_idx), dtype=torch.uint8, device=x.device)
    diff = (x[a_idx] - x[b_idx]).abs().amax(dim=1)
    return (diff <= atol).to(torch.uint8)


@torch.no_grad()
def pair_maxdiff(x: torch.Tensor, a_idx: torch.Tensor, b_idx: torch.Tensor) -> torch.Tensor:
    """Per-pair max|x[a]-x[b]| over the local shard.  (k,) fp32."""
    if x.shape[1] == 0:
        return torch.zeros(len(a_idx), dtype=torch.float32, device=x.device)
    return (x[a_idx] - x[b_idx]).abs().amax(dim=1).float()


@torch.no_grad()
def row_absmax(x: torch.Tensor) -> torch.Tensor:
    """Per-row max|x| over the local shard.  (m,) fp32."""
    if x.shape[1] == 0:
        return torch.zeros(x.shape[0], dtype=torch.float32, device=x.device)
    return x.abs().amax(dim=1).float()


@torch.no_grad()
def mean_rows(x: torch.Tensor, idx: torch.Tensor, out: torch.Tensor) -> None:
    """out = mean over selected rows of x.  x: (m, d), idx: (k,), out: (d,)."""
    torch.mean(x[idx], dim=0, out=out)


@torch.no_grad()
def sum_rows(x: torch.Tensor, out: torch.Tensor) -> None:
    torch.sum(x, dim=0, out=out)


# --------------------------------------------------------------------- cyclic code
@torch.no_grad()
def cyclic_encode(grads: torch.Tensor, w_re: torch.Tensor, w_im: torch.Tensor, out: torch.Tensor) -> None:
    """out[0] = sum_k w_re[k]*grads[k];  out[1] = sum_k w_im[k]*grads[k].

    grads: (k, d) fp32 (the 2s+1 sub-batch gradients, band order); w_*: (k,);
    out: (2, d) fp32 planes of the encoded complex gradient.
    """
    torch.sum(grads * w_re[:, None], dim=0, out=out[0])
    torch.sum(grads * w_im[:, None], dim=0, out=out[1])


@torch.no_grad()
def cyclic_project(rows: torch.Tensor, z: torch.Tensor) -> torch.Tensor:
    """Partial per-row projection proj_r = sum_d rows[r, d] * z[d] over the shard.

    rows: (m, d_shard) fp32; z: (d_shard,) fp32 -> (m,) fp32 partials
    (to be summed across ranks).
    """
    if rows.shape[1] == 0:
        return torch.zeros(rows.shape[0], dtype=torch.float32, device=rows.device)
    return rows @ z


@torch.no_grad()
def combine_rows(x: torch.Tensor, rows: torch.Tensor, w: torch.Tensor, out: torch.Tensor) -> None:
    """out = sum_i w[i] * x[rows[i]] — the generic weighted row combination behind
    winner-mean, cyclic encode/recombine and plain summing."""
    if x.shape[1] == 0:
        out.zero_()
        return
    out.copy_((w @ x[rows]).reshape(-1))


@torch.no_grad()
def combine_rows_slice(x: torch.Tensor, rows: torch.Tensor, w: torch.Tensor,
                       out: torch.Tensor, col_off: int) -> None:
    """Column-range variant for the bucketed cyclic encode: identical per-element
    arithmetic to combine_rows restricted to [col_off, col_off + len(out))."""
    n = out.numel()
    if n == 0:
        return
    out.copy_((w @ x[rows][:, col_off : col_off + n]).reshape(-1))


@torch.no_grad()
def cyclic_recombine(r_planes: torch.Tensor, v_re: torch.Tensor, v_im: torch.Tensor, out: torch.Tensor) -> None:
    """out = Re( v @ R ) over the local shard.

    Re(v_i * (re + i*im)) = v_re*re - v_im*im summed over rows i.
    r_planes: (n, 2, d_shard); v_*: (n,); out: (d_shard,).
    """
    re = torch.einsum("nd,n->d", r_planes[:, 0, :], v_re)
    im = torch.einsum("nd,n->d", r_planes[:, 1, :], v_im)
    torch.sub(re, im, out=out)


# --------------------------------------------------------------------- geo-median
@torch.no_grad()
def segment_absmax(x: torch.Tensor, seg: torch.Tensor) -> torch.Tensor:
    """(rows, d), (L+1,) bounds -> (rows, L) per-segment max|x| (segment vote)."""
    L = seg.numel() - 1
    out = torch.zeros(x.shape[0], L, dtype=torch.float32, device=x.device)
    for l in range(L):
        lo, hi = int(seg[l]), int(seg[l + 1])
        if hi > lo:
            out[:, l] = x[:, lo:hi].abs().amax(dim=1)
    return out


def segment_pair_maxdiff(x: torch.Tensor, a_idx: torch.Tensor, b_idx: torch.Tensor,
                         seg: torch.Tensor) -> torch.Tensor:
    """(pairs, L) per-segment max|x[a]-x[b]| (segment-granular tolerance vote)."""
    return segment_absmax(x[a_idx] - x[b_idx], seg)


def segment_sqdist(x: torch.Tensor, z: torch.Tensor, seg: torch.Tensor) -> torch.Tensor:
    """Per-(row, segment) partial squared distance ||x[p, seg_l] - z[seg_l]||^2.

    x: (P, d_shard), z: (d_shard,), seg: (L+1,) int64 local segment bounds.
    Returns (P, L) fp32 partials.
    """
    P = x.shape[0]
    L = len(seg) - 1
    out = torch.zeros(P, L, dtype=torch.float32, device=x.device)
    d2 = (x - z[None, :]).pow(2)
    for l in range(L):
        lo, hi = int(seg[l]), int(seg[l + 1])
        if hi > lo:
            out[:, l] = d2[:, lo:hi].sum(dim=1)
    return out


@torch.no_grad()
def segment_weighted_mean(x: torch.Tensor, w: torch.Tensor, seg: torch.Tensor, out: torch.Tensor) -> None:
    """out[seg_l] = sum_p w[p, l] * x[p, seg_l]  (weights pre-normalised per segment).

    x: (P, d_shard); w: (P, L); seg: (L+1,); out: (d_shard,).
    """
    L = len(seg) - 1
    for l in range(L):
        lo, hi = int(seg[l]), int(seg[l + 1])
        if hi > lo:
            out[lo:hi] = w[:, l] @ x[:, lo:hi]


@torch.no_grad()
def segment_gram(x: torch.Tensor, seg: torch.Tensor) -> torch.Tensor:
    """Per-segment Gram matrices G[l] = X_l @ X_l^T (for Krum's distance matrix).

    x: (P, d_shard); seg: (L+1,).  Returns (L, P, P) fp32 partials.
    """
    P = x.shape[0]
    L = len(seg) - 1
    out = torch.zeros(L, P, P, dtype=torch.float32, device=x.device)
    for l in range(L):
        lo, hi = int(seg[l]), int(seg[l + 1])
        if hi > lo:
            xs = x[:, lo:hi]
            out[l] = xs @ xs.T
    return out


# --------------------------------------------------------------------- dtype cast
@torch.no_grad()
def cast_to_bf16(x: torch.Tensor, out: torch.Tensor) -> None:
    out.copy_(x)


@torch.no_grad()
def cast_from_bf16(x: torch.Tensor, out: torch.Tensor) -> None:
    out.copy_(x)
