#!/usr/bin/env python3
"""GPT-2 DDP training over the adapcc hook (reference:
models/gpt2/train_gpt2_ddp.py, re-targeted at the self-contained GPT-2 on
synthetic tokens — no dataset downloads in this environment).

    python -m torch.distributed.run --nproc-per-node 8 \
        --master-addr 127.0.0.1 examples/train_gpt2_ddp.py --model small
"""

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
from adapcc_amd.models.gpt2 import GPT2, GPT2Config
from adapcc_amd.runtime.hook import AdapccDDPState, adapcc_allreduce_hook


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="small", choices=["tiny", "small", "medium"])
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--batch", type=int, default=8)
    p.add_argument("--seq", type=int, default=512)
    p.add_argument("--lr", type=float, default=3e-4)
    p.add_argument("--entry_point", type=int, default=-1)
    p.add_argument("--relay", action="store_true")
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

    cfg = getattr(GPT2Config, args.model)()
    torch.manual_seed(1)
    model = GPT2(cfg).to(device)
    AdapCC.init(CommArgs(entry_point=args.entry_point, relay=args.relay),
                local_rank, rank, world)
    AdapCC.setup()

    if world > 1:
        model = DDP(model, device_ids=[device.index] if use_cuda else None,
                    bucket_cap_mb=100)
        state = AdapccDDPState(AdapCC.communicator)
        model.register_comm_hook(state, adapcc_allreduce_hook)
    else:
        state = None

    opt = torch.optim.AdamW(model.parameters(), lr=args.lr)
    T = min(args.seq, cfg.n_positions)
    torch.manual_seed(100 + rank)
    data = torch.randint(0, cfg.vocab_size, (args.batch, T + 1), device=device)
    x, y = data[:, :-1], data[:, 1:].contiguous()

    for step in range(args.steps):
        t0 = time.perf_counter()
        if state is not None:
            state.on_step(step)
        opt.zero_grad(set_to_none=True)
        with torch.autocast(device.type, dtype=torch.bfloat16,
                            enabled=use_cuda):
            _, loss = model(x, y)
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
