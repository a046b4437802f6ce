#!/usr/bin/env python3
"""Precision accuracy benchmark (reference:
models/image-classification/accuracy_benchmark.py with its
resnet18_{fp32,fp16}.txt / resnet18_bfp16.txt trajectories).

Trains ResNet-18 on a fixed synthetic classification task under fp32 /
fp16-amp / bf16-amp and logs per-step loss + top-1 accuracy so precision
regressions are visible as diverging trajectories. Optionally wraps the
model in DDP with the adapcc hook when launched with world > 1.

    python examples/accuracy_benchmark.py --dtypes fp32,bf16 --steps 50
"""

from __future__ import annotations

import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from adapcc_amd.models.resnet import ResNet18


def run_one(dtype: str, steps: int, batch: int, device, out_dir: str,
            log_gns: bool) -> None:
    torch.manual_seed(0)
    model = ResNet18(num_classes=10).to(device)
    opt = torch.optim.SGD(model.parameters(), lr=0.05, momentum=0.9)
    crit = torch.nn.CrossEntropyLoss()

    # fixed learnable synthetic task: class = argmax of 10 random planes
    torch.manual_seed(42)
    planes = torch.randn(10, 3 * 32 * 32, device=device)

    amp_dtype = {"fp32": None, "fp16": torch.float16,
                 "bf16": torch.bfloat16}[dtype]
    path = os.path.join(out_dir, f"resnet18_{dtype}.txt")
    gns_est = None
    if log_gns:
        from adapcc_amd.utils.gns import GNS

        gns_est = GNS()

    with open(path, "w") as f:
        for step in range(steps):
            torch.manual_seed(1000 + step)
            x = torch.randn(batch, 3, 32, 32, device=device)
            y = (x.flatten(1) @ planes.T).argmax(1)
            opt.zero_grad(set_to_none=True)
            t0 = time.perf_counter()
            if amp_dtype is None:
                logits = model(x)
                loss = crit(logits, y)
            else:
                with torch.autocast(device.type, dtype=amp_dtype):
                    logits = model(x)
                    loss = crit(logits, y)
            loss.backward()
            opt.step()
            acc = (logits.argmax(1) == y).float().mean()
            line = (f"{step},{loss.item():.6f},{acc.item():.4f},"
                    f"{1000 * (time.perf_counter() - t0):.2f}")
            f.write(line + "\n")
            if step % 10 == 0:
                print(f"[{dtype}] step {step}: loss {loss.item():.4f} "
                      f"acc {acc.item():.3f}", flush=True)
    print(f"[{dtype}] wrote {path}")


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--dtypes", default="fp32,fp16,bf16")
    p.add_argument("--steps", type=int, default=50)
    p.add_argument("--batch", type=int, default=64)
    p.add_argument("--out_dir", default=".")
    p.add_argument("--gns", action="store_true")
    args = p.parse_args()

    device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    os.makedirs(args.out_dir, exist_ok=True)
    for dtype in args.dtypes.split(","):
        dtype = dtype.strip()
        if dtype == "fp16" and device.type == "cpu":
            print("skipping fp16 on CPU")
            continue
        run_one(dtype, args.steps, args.batch, device, args.out_dir, args.gns)


if __name__ == "__main__":
    main()
