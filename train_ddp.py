#!/usr/bin/env python3
"""Canonical AdapCC DDP training template (reference: train_ddp.py).

VGG16, synthetic data, DDP(bucket_cap_mb=100) with the adapcc comm hook,
per-step relay update, and on-the-fly topology reconstruction every
``--profile_freq`` steps. Accepts the launcher's 6-flag contract.

    python -m torch.distributed.run --nproc-per-node 8 \
        --master-addr 127.0.0.1 train_ddp.py --entry_point -1
"""

from __future__ import annotations

import argparse
import os
import sys
import time

import os as _os
_os.environ.setdefault("MIOPEN_FIND_MODE", "FAST")

import torch
import torch.distributed as dist
from torch.nn.parallel import DistributedDataParallel as DDP

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from adapcc_amd import AdapCC, CommArgs  # noqa: E402
from adapcc_amd.models.vgg import VGG16  # noqa: E402
from adapcc_amd.runtime.hook import AdapccDDPState, adapcc_allreduce_hook  # noqa: E402


def parse_args():
    p = argparse.ArgumentParser()
    # the 6 forwarded launcher flags (reference launcher.py:54-62)
    p.add_argument("--port", type=int, default=18000)
    p.add_argument("--entry_point", type=int, default=-1)
    p.add_argument("--strategy_file", type=str, default="")
    p.add_argument("--logical_graph", type=str, default="")
    p.add_argument("--parallel_degree", type=int, default=0)
    p.add_argument("--profile_freq", type=int, default=0)
    p.add_argument("--steps", type=int, default=5)
    p.add_argument("--batch", type=int, default=16)
    p.add_argument("--image_size", type=int, default=224)
    p.add_argument("--relay", action="store_true")
    return p.parse_args()


def main():
    args = parse_args()
    rank = int(os.environ.get("RANK", 0))
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    use_cuda = torch.cuda.is_available()
    dev_idx = local_rank % max(1, torch.cuda.device_count() or 1)
    device = torch.device(f"cuda:{dev_idx}" if use_cuda else "cpu")
    if use_cuda:
        torch.cuda.set_device(device)
    if world > 1:
        dist.init_process_group("nccl" if use_cuda else "gloo",
                                rank=rank, world_size=world)

    torch.manual_seed(42)
    model = VGG16(num_classes=100, in_size=args.image_size).to(device)
    AdapCC.init(CommArgs.from_namespace(args), local_rank, rank, world)
    AdapCC.setup()

    if world > 1:
        model = DDP(model, device_ids=[device.index] if use_cuda else None,
                    bucket_cap_mb=100)
        state = AdapccDDPState(AdapCC.communicator)
        model.register_comm_hook(state, adapcc_allreduce_hook)
    else:
        state = None

    opt = torch.optim.SGD(model.parameters(), lr=0.01, momentum=0.9)
    crit = torch.nn.CrossEntropyLoss()
    x = torch.randn(args.batch, 3, args.image_size, args.image_size,
                    device=device)
    y = torch.randint(0, 100, (args.batch,), device=device)

    for step in range(args.steps):
        t0 = time.perf_counter()
        if state is not None:
            state.on_step(step)
        if args.profile_freq and step and step % args.profile_freq == 0:
            AdapCC.reconstruct_topology()
        opt.zero_grad(set_to_none=True)
        loss = crit(model(x), y)
        loss.backward()
        opt.step()
        if use_cuda:
            torch.cuda.synchronize()
        if rank == 0:
            print(f"[Rank {rank}] step {step} loss {loss.item():.4f} "
                  f"time {1000 * (time.perf_counter() - t0):.1f} ms",
                  flush=True)

    AdapCC.clear()
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
