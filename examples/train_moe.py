#!/usr/bin/env python3
"""Expert-parallel MoE training over the adapcc all-to-all (reference:
models/moe/train_moe.py used fastmoe's FMoETransformerMLP + NCCL; here the
token exchange rides the adapcc engine and dense params sync via the DDP
hook).

    python -m torch.distributed.run --nproc-per-node 8 \
        --master-addr 127.0.0.1 examples/train_moe.py
"""

from __future__ import annotations

import argparse
import os
import sys
import time

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from adapcc_amd import AdapCC, CommArgs
from adapcc_amd.models.moe import MoETransformerBlock


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--batch", type=int, default=8)
    p.add_argument("--seq", type=int, default=256)
    p.add_argument("--d_model", type=int, default=1024)
    p.add_argument("--d_hidden", type=int, default=4096)
    p.add_argument("--local_experts", type=int, default=2)
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

    AdapCC.init(CommArgs(entry_point=-1), local_rank, rank, world)
    AdapCC.setup()
    comm = AdapCC.communicator

    torch.manual_seed(7)  # dense params identical across ranks
    model = MoETransformerBlock(
        d_model=args.d_model, n_head=8, d_hidden=args.d_hidden,
        num_local_experts=args.local_experts, comm=comm, world_size=world,
        rank=rank,
    ).to(device)
    # expert weights are rank-local (expert parallel): re-init per rank
    torch.manual_seed(100 + rank)
    for e in model.moe.experts:
        for m in e.modules():
            if isinstance(m, torch.nn.Linear):
                torch.nn.init.normal_(m.weight, std=0.02)
                torch.nn.init.zeros_(m.bias)

    opt = torch.optim.AdamW(model.parameters(), lr=1e-3)
    torch.manual_seed(500 + rank)
    x = torch.randn(args.batch, args.seq, args.d_model, device=device)

    expert_params = {id(p) for e in model.moe.experts for p in e.parameters()}
    for step in range(args.steps):
        t0 = time.perf_counter()
        opt.zero_grad(set_to_none=True)
        y = model(x)
        loss = (y ** 2).mean()
        loss.backward()
        # dense (non-expert) grads sync across ranks; expert grads stay local
        if world > 1:
            for prm in model.parameters():
                if prm.grad is not None and id(prm) not in expert_params:
                    comm.all_reduce(prm.grad, average=True)
            comm.synchronize()
        opt.step()
        if use_cuda:
            torch.cuda.synchronize()
        if rank == 0:
            print(f"step {step}: loss {loss.item():.5f} "
                  f"({1000 * (time.perf_counter() - t0):.1f} ms)", flush=True)

    AdapCC.clear()
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
