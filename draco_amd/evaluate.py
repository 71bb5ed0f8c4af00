"""Standalone evaluator: polls the checkpoint directory and reports Prec@1/Prec@5,
mirroring /root/reference/src/distributed_evaluator.py:59-135 (same model_step_N file
layout, same polling loop) on the synthetic held-out stream."""
from __future__ import annotations

import argparse
import glob
import os
import re
import time

import torch
import torch.nn.functional as F

from .data import SyntheticClassification
from .models import build_model


def _accuracy(logits, y, topk=(1, 5)):
    maxk = max(topk)
    _, pred = logits.topk(maxk, 1, True, True)
    correct = pred.eq(y.view(-1, 1).expand_as(pred))
    return [float(correct[:, :k].any(dim=1).float().mean()) for k in topk]


def evaluate_checkpoint(path: str, device: torch.device, batches: int = 10, batch_size: int = 100):
    payload = torch.load(path, map_location="cpu", weights_only=False)
    model = build_model(payload["network"], payload["dataset"]).to(device)
    model.load_state_dict(payload["model"])
    model.eval()
    data = SyntheticClassification(payload["dataset"], device, seed=1234,
                                   task=payload.get("synthetic_task", "means"))
    p1 = p5 = loss = 0.0
    with torch.no_grad():
        for b in range(batches):
            x, y = data.get_batch(2**40 + b * batch_size, batch_size)
            logits = model(x)
            loss += float(F.cross_entropy(logits, y))
            a1, a5 = _accuracy(logits, y)
            p1 += a1
            p5 += a5
    return {"step": payload["step"], "prec1": p1 / batches, "prec5": p5 / batches, "loss": loss / batches}


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--model-dir", type=str, default="output/models/")
    p.add_argument("--eval-freq", type=int, default=50)
    p.add_argument("--eval-batch-size", type=int, default=100)
    p.add_argument("--poll-seconds", type=float, default=10.0)
    p.add_argument("--once", action="store_true", help="evaluate existing checkpoints and exit")
    args = p.parse_args(argv)
    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")

    seen = set()
    while True:
        paths = sorted(
            glob.glob(os.path.join(args.model_dir, "model_step_*")),
            key=lambda s: int(re.search(r"(\d+)$", s).group(1)),
        )
        fresh = [q for q in paths if q not in seen and not q.endswith(".tmp")]
        for q in fresh:
            seen.add(q)
            rec = evaluate_checkpoint(q, device, batch_size=args.eval_batch_size)
            print(
                "Testset Performance: Cur Step:{step} Prec@1: {prec1:.4f} Prec@5: {prec5:.4f} Loss: {loss:.4f}".format(**rec),
                flush=True,
            )
        if args.once:
            break
        time.sleep(args.poll_seconds)


if __name__ == "__main__":
    main()
