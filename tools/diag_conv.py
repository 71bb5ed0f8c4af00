"""Probe: ResNet-18 CIFAR fwd+bwd variants on MI355X (batch 128, bf16 autocast).
Decides whether channels_last / torch.compile are worth wiring into the trainer."""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from draco_amd.models import build_model


def bench(model, x, y, n=20, warmup=5):
    crit = torch.nn.functional.cross_entropy

    def step():
        for p in model.parameters():
            p.grad = None
        with torch.autocast("cuda", dtype=torch.bfloat16):
            loss = crit(model(x), y)
        loss.backward()

    for _ in range(warmup):
        step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        step()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1000


def main():
    torch.backends.cudnn.benchmark = True
    dev = "cuda"
    B = int(os.environ.get("B", 128))
    x = torch.randn(B, 3, 32, 32, device=dev)
    y = torch.randint(0, 10, (B,), device=dev)

    m = build_model("ResNet18", "Cifar10").to(dev)
    print(f"B={B} contiguous: {bench(m, x, y):.2f} ms")

    m2 = build_model("ResNet18", "Cifar10").to(dev).to(memory_format=torch.channels_last)
    x2 = x.to(memory_format=torch.channels_last)
    print(f"B={B} channels_last: {bench(m2, x2, y):.2f} ms")

    try:
        m3 = torch.compile(build_model("ResNet18", "Cifar10").to(dev))
        print(f"B={B} compiled: {bench(m3, x, y, n=10):.2f} ms")
    except Exception as e:
        print("compile failed:", e)


if __name__ == "__main__":
    main()
