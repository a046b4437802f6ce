#!/usr/bin/env python3
"""xGMI link-contention probe (reference: nccl-perf/contention/flow.cu).

The reference measured PCIe-switch sharing by timing a probe flow while
background flows saturated sibling links. The MI355X analog asks whether
xGMI point-to-point links are independent: measure the 0->1 transfer
bandwidth (a) alone and (b) while every other rank pair streams
concurrently. On a true per-link mesh the probe should hold its bandwidth
(degradation ~0); a shared fabric hop or mis-routed pair shows up as a
drop, and a drop > --threshold is reported as contention.

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 benchmarks/contention_probe.py \
        --out gpurun_out/contention.csv

On a single GPU (multi-process) this measures HBM-port contention instead
of xGMI - the harness reports which regime it ran in.
"""

from __future__ import annotations

import argparse
import os
import sys
import time

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def timed_flow(dst, src, elems, device, reps=8):
    t_out = torch.ones(elems, dtype=torch.float32, device=device)
    t_in = torch.empty(elems, dtype=torch.float32, device=device)
    if device.type == "cuda":
        torch.cuda.synchronize()
    start = time.perf_counter()
    for _ in range(reps):
        req = dist.isend(t_out, dst=dst)
        dist.recv(t_in, src=src)
        req.wait()
    if device.type == "cuda":
        torch.cuda.synchronize()
    return elems * 4 * reps / (time.perf_counter() - start) / 1e9  # GB/s


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--mb", type=int, default=64, help="flow size (MB)")
    ap.add_argument("--threshold", type=float, default=0.10,
                    help="fractional drop flagged as contention")
    ap.add_argument("--out", default="")
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    local = int(os.environ.get("LOCAL_RANK", rank))
    use_cuda = torch.cuda.is_available()
    if use_cuda:
        torch.cuda.set_device(local % max(1, torch.cuda.device_count()))
    backend = "nccl" if (use_cuda and torch.cuda.device_count() >= world) \
        else "gloo"
    dist.init_process_group(backend)
    device = torch.device("cuda", torch.cuda.current_device()) if use_cuda \
        else torch.device("cpu")
    elems = args.mb * (1 << 20) // 4
    if world < 2:
        print("need >= 2 ranks")
        return

    # phase 1: probe pair (0,1) alone
    dist.barrier()
    solo = 0.0
    if rank == 0:
        solo = timed_flow(1, 1, elems, device)
    elif rank == 1:
        solo = timed_flow(0, 0, elems, device)
    dist.barrier()

    # phase 2: probe pair (0,1) while pairs (2,3), (4,5), ... stream
    loaded = 0.0
    if rank == 0:
        loaded = timed_flow(1, 1, elems, device)
    elif rank == 1:
        loaded = timed_flow(0, 0, elems, device)
    else:
        peer = rank ^ 1
        if peer < world:
            for _ in range(3):  # keep streaming through the probe window
                timed_flow(peer, peer, elems, device)
    dist.barrier()

    vals = [None] * world
    dist.all_gather_object(vals, (solo, loaded))
    if rank == 0:
        solo0, loaded0 = vals[0]
        drop = 0.0 if solo0 <= 0 else max(0.0, 1 - loaded0 / solo0)
        regime = ("xgmi" if use_cuda and torch.cuda.device_count() >= world
                  else ("hbm-shared" if use_cuda else "cpu-tcp"))
        rows = [
            f"regime,{regime}",
            f"probe_pair,0-1",
            f"solo_GBps,{solo0:.2f}",
            f"loaded_GBps,{loaded0:.2f}",
            f"drop_frac,{drop:.4f}",
            f"contended,{int(drop > args.threshold)}",
        ]
        print("\n".join(rows))
        if drop > args.threshold:
            print(f"CONTENTION: probe lost {drop*100:.1f}% under load "
                  f"(> {args.threshold*100:.0f}%)")
        else:
            print("links independent at this load")
        if args.out:
            os.makedirs(os.path.dirname(args.out) or ".", exist_ok=True)
            with open(args.out, "w") as f:
                f.write("\n".join(rows) + "\n")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
