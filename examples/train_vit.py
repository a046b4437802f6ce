#!/usr/bin/env python3
"""ViT DDP on synthetic images (reference: models/vit/train_vit.py)."""

from __future__ import annotations

import argparse
import os
import sys
import time

import torch
import torch.distributed as dist
from torch.nn.parallel import DistributedDataParallel as DDP

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from adapcc_amd import AdapCC, CommArgs
from adapcc_amd.models.vit import ViT, ViTConfig
from adapcc_amd.runtime.hook import AdapccDDPState, adapcc_allreduce_hook


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--batch", type=int, default=32)
    p.add_argument("--tiny", action="store_true")
    args = p.parse_args()

    rank = int(os.environ.get("RANK", 0))
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    use_cuda = torch.cuda.is_available()
    dev_idx = local_rank % max(1, torch.cuda.device_count() or 1)
    device = torch.device(f"cuda:{dev_idx}" if use_cuda else "cpu")
    if use_cuda:
        torch.cuda.set_device(device)
    if world > 1:
        dist.init_process_group("nccl" if use_cuda else "gloo")

    cfg = ViTConfig.tiny() if args.tiny else ViTConfig.base()
    torch.manual_seed(3)
    model = ViT(cfg).to(device)
    AdapCC.init(CommArgs(entry_point=-1), local_rank, rank, world)
    AdapCC.setup()

    if world > 1:
        model = DDP(model, device_ids=[device.index] if use_cuda else None,
                    bucket_cap_mb=100)
        state = AdapccDDPState(AdapCC.communicator)
        model.register_comm_hook(state, adapcc_allreduce_hook)
    else:
        state = None

    opt = torch.optim.AdamW(model.parameters(), lr=1e-3)
    crit = torch.nn.CrossEntropyLoss()
    torch.manual_seed(100 + rank)
    x = torch.randn(args.batch, 3, cfg.image_size, cfg.image_size,
                    device=device)
    y = torch.randint(0, cfg.num_classes, (args.batch,), device=device)

    for step in range(args.steps):
        t0 = time.perf_counter()
        if state is not None:
            state.on_step(step)
        opt.zero_grad(set_to_none=True)
        loss = crit(model(x), y)
        loss.backward()
        opt.step()
        if use_cuda:
            torch.cuda.synchronize()
        if rank == 0:
            print(f"step {step}: loss {loss.item():.4f} "
                  f"({1000 * (time.perf_counter() - t0):.1f} ms)", flush=True)

    AdapCC.clear()
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
