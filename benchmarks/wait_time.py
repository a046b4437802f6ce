#!/usr/bin/env python3
"""Straggler wait-time measurement harness (reference:
units-test/get_wait_time.py + throughput.py).

Runs a DDP training loop with the adapcc hook and records, per step, the
spread of first-bucket arrival times across ranks: ``(max - min) * alpha``
where ``--heter_alpha`` emulates heterogeneity by scaling one rank's
compute with an injected sleep (reference scaled measured waits the same
way, get_wait_time.py:50-62). Writes the reference's CSV shape
(step,wait) and, with --relay, exercises the rent-or-buy coordinator so
straggler steps show a shrunken active set.

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 benchmarks/wait_time.py \
        --steps 20 --heter_alpha 2.7 --out gpurun_out/wait_time.csv
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

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from adapcc_amd import AdapCC, CommArgs
from adapcc_amd.models.vgg import VGG16
from adapcc_amd.runtime.hook import AdapccDDPState, adapcc_allreduce_hook


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--batch", type=int, default=8)
    p.add_argument("--image_size", type=int, default=64)
    p.add_argument("--heter_alpha", type=float, default=1.0,
                   help=">1 injects a sleep on rank world-1 scaled by alpha")
    p.add_argument("--straggle_ms", type=float, default=20.0)
    p.add_argument("--relay", action="store_true")
    p.add_argument("--out", default="wait_time.csv")
    args = p.parse_args()

    rank = int(os.environ.get("RANK", 0))
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    use_cuda = torch.cuda.is_available()
    dev_idx = local_rank % max(1, torch.cuda.device_count() or 1)
    device = torch.device(f"cuda:{dev_idx}" if use_cuda else "cpu")
    if use_cuda:
        torch.cuda.set_device(device)
    backend = "nccl" if (use_cuda and torch.cuda.device_count() >= world) else "gloo"
    dist.init_process_group(backend, rank=rank, world_size=world)

    torch.manual_seed(0)
    model = VGG16(num_classes=10, in_size=args.image_size).to(device)
    AdapCC.init(CommArgs(entry_point=-1, relay=args.relay), local_rank, rank,
                world)
    AdapCC.setup()
    ddp = DDP(model, device_ids=[device.index] if use_cuda else None,
              bucket_cap_mb=25)
    state = AdapccDDPState(AdapCC.communicator)

    first_ts = {}

    def timing_hook(st, bucket):
        if st._first_bucket_of_step:
            first_ts[st.step] = time.perf_counter()
        return adapcc_allreduce_hook(st, bucket)

    ddp.register_comm_hook(state, timing_hook)

    crit = torch.nn.CrossEntropyLoss()
    x = torch.randn(args.batch, 3, args.image_size, args.image_size,
                    device=device)
    y = torch.randint(0, 10, (args.batch,), device=device)
    opt = torch.optim.SGD(ddp.parameters(), lr=0.01)

    is_straggler = (rank == world - 1) and args.heter_alpha > 1.0
    for step in range(args.steps):
        state.on_step(step)
        if is_straggler:
            time.sleep(args.straggle_ms / 1000.0 * (args.heter_alpha - 1.0))
        opt.zero_grad(set_to_none=True)
        loss = crit(ddp(x), y)
        loss.backward()
        opt.step()
    if use_cuda:
        torch.cuda.synchronize()

    # gather first-bucket timestamps; spread = (max - min) * alpha
    gathered = [None] * world
    dist.all_gather_object(gathered, first_ts)
    if rank == 0:
        rows = []
        for step in range(args.steps):
            ts = [g[step] for g in gathered if step in g]
            if len(ts) == world:
                rows.append((step, (max(ts) - min(ts)) * args.heter_alpha))
        os.makedirs(os.path.dirname(args.out) or ".", exist_ok=True)
        with open(args.out, "w") as f:
            for step, wait in rows:
                f.write(f"{step},{wait * 1000:.3f}\n")
        waits = [w for _, w in rows]
        print(f"wait spread ms: min {1000 * min(waits):.2f} "
              f"max {1000 * max(waits):.2f} "
              f"mean {1000 * sum(waits) / len(waits):.2f} -> {args.out}")
        if args.relay:
            stats = AdapCC.communicator.stats()
            n = stats.get("relay_negotiations", 0)
            if n:
                ms = 1000 * stats.get("relay_negotiation_s", 0) / n
                print(f"relay negotiation latency: {ms:.2f} ms/step over "
                      f"{int(n)} steps")

    AdapCC.clear()
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
