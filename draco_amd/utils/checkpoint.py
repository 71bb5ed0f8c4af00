"""Checkpointing in the reference's file layout (`train_dir/model_step_N`,
baseline_master.py:237-248), improved per SURVEY §5.4: state_dict + optimizer state
(the reference drops momentum) written atomically (tmp + rename); `--checkpoint-step`
resume supported on every rank."""
from __future__ import annotations

import os

import torch


def save_checkpoint(path: str, model, space, opt, step: int, cfg) -> None:
    os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
    payload = {
        "step": step,
        "model": {k: v.cpu() for k, v in model.state_dict().items()},
        "flat_param": space.flat_param[: space.d].detach().cpu(),
        "optimizer": _cpu_sd(opt.state_dict()),
        "network": cfg.network,
        "dataset": cfg.dataset,
    }
    tmp = path + ".tmp"
    torch.save(payload, tmp)
    os.replace(tmp, path)


def load_checkpoint(path: str, model, space, opt) -> int:
    payload = torch.load(path, map_location="cpu", weights_only=False)
    model.load_state_dict(payload["model"])
    # model.load_state_dict copies into the flat-space views, so flat_param is
    # already consistent; restore it explicitly anyway for safety
    space.load_flat(payload["flat_param"])
    if opt is not None and payload.get("optimizer") is not None:
        opt.load_state_dict(payload["optimizer"])
    return payload["step"]


def _cpu_sd(sd: dict) -> dict:
    return {k: (v.cpu() if isinstance(v, torch.Tensor) else v) for k, v in sd.items()}
