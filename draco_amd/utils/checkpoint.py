"""Checkpointing in the reference's file layout (`train_dir/model_step_N`,
baseline_master.py:237-248), improved per SURVEY §5.4: state_dict + optimizer state
(the reference drops momentum) written atomically (tmp + rename); `--checkpoint-step`
resume supported on every rank.

Layout independence: model weights restore through load_state_dict (per-parameter),
and flat optimizer buffers (momentum / Adam moments) are stored PER PARAMETER via the
flat-space views — so a checkpoint written under one flat layout (e.g.
channels_last=False) resumes correctly under another.
"""
from __future__ import annotations

import os

import torch


def _flat_to_params(space, flat: torch.Tensor):
    """Split a flat optimizer buffer into per-parameter tensors (logical layout)."""
    return [
        space._view(flat, o, n, shape).detach().cpu().clone()
        for o, n, shape in zip(space.offsets, space.numels, space.shapes)
    ]


def _params_to_flat(space, tensors, flat: torch.Tensor) -> None:
    with torch.no_grad():
        for t, o, n, shape in zip(tensors, space.offsets, space.numels, space.shapes):
            space._view(flat, o, n, shape).copy_(t.to(flat.device))


def _pack_opt_state(space, opt) -> dict:
    sd = opt.state_dict()
    out = {}
    for k, v in sd.items():
        if isinstance(v, torch.Tensor) and v.numel() == space.d_pad:
            out[k + "__params"] = _flat_to_params(space, v)
        elif isinstance(v, torch.Tensor):
            out[k] = v.cpu()
        else:
            out[k] = v
    return out


def _unpack_opt_state(space, opt, packed: dict) -> None:
    for k, v in packed.items():
        if k.endswith("__params"):
            name = k[: -len("__params")]
            flat = getattr(opt, name, None)
            if flat is not None:
                _params_to_flat(space, v, flat)
        elif k == "first":
            opt._first = v
        elif k == "t":
            opt.t = v


_COMPILE_PREFIX = "_orig_mod."


def _strip_compile_prefix(sd: dict) -> dict:
    """torch.compile wraps the module, prefixing state_dict keys with `_orig_mod.`;
    checkpoints are stored with canonical keys so they are interchangeable across
    compile settings."""
    return {k[len(_COMPILE_PREFIX):] if k.startswith(_COMPILE_PREFIX) else k: v
            for k, v in sd.items()}


def save_checkpoint(path: str, model, space, opt, step: int, cfg) -> None:
    os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
    payload = {
        "step": step,
        "model": {k: v.cpu() for k, v in _strip_compile_prefix(model.state_dict()).items()},
        "optimizer": _pack_opt_state(space, opt) if opt is not None else None,
        "network": cfg.network,
        "dataset": cfg.dataset,
        # the held-out stream is task-dependent; the evaluator must score against
        # the same synthetic task the checkpoint was trained on
        "synthetic_task": getattr(cfg, "synthetic_task", "means"),
    }
    tmp = path + ".tmp"
    torch.save(payload, tmp)
    os.replace(tmp, path)


def load_checkpoint(path: str, model, space, opt) -> int:
    payload = torch.load(path, map_location="cpu", weights_only=False)
    sd = _strip_compile_prefix(payload["model"])  # tolerate old compiled checkpoints
    if any(k.startswith(_COMPILE_PREFIX) for k in model.state_dict()):
        sd = {_COMPILE_PREFIX + k: v for k, v in sd.items()}  # loading INTO a compiled model
    # load_state_dict copies through the flat-space views: layout-independent restore
    model.load_state_dict(sd)
    if opt is not None and payload.get("optimizer") is not None:
        _unpack_opt_state(space, opt, payload["optimizer"])
    return payload["step"]
