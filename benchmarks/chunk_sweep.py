#!/usr/bin/env python3
"""Chunk-size / kernel-grid sweep for the native engine (reference:
nccl-perf/tree tree_chunk experiments, report_tree_chunk128.txt).

Times the native allreduce at a fixed payload across chunk sizes and
kernel-grid shapes (ADAPCC_N_GROUPS x ADAPCC_WGS_PER_GROUP), printing a
table like the reference's checked-in chunk reports. The best cell is the
chunk/grid default evidence for PERFORMANCE tuning.

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 benchmarks/chunk_sweep.py \
        --mb 100 --out gpurun_out/chunk_sweep.csv

Note: grid knobs are read at engine construction, so each grid shape
rebuilds the engine; chunk size varies per strategy (cheap).
"""

from __future__ import annotations

import argparse
import os
import sys
import time

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def bench_case(rank, world, payload_elems, chunk_bytes, iters=8, warmup=3):
    from adapcc_amd.runtime.engine import NativeEngine
    from adapcc_amd.strategy.partrees import synthesize_stars

    eng = NativeEngine(rank, world, device=torch.cuda.current_device(),
                       cap_bytes=max(64 << 20, payload_elems * 4))
    eng.bootstrap()
    eng.set_strategy(synthesize_stars(world, chunk_bytes=chunk_bytes))
    t = torch.randn(payload_elems, device="cuda")
    for _ in range(warmup):
        eng.all_reduce(t)
    eng.synchronize()
    dist.barrier()
    start = time.perf_counter()
    for _ in range(iters):
        eng.all_reduce(t)
    eng.synchronize()
    dt = (time.perf_counter() - start) / iters
    dist.barrier()
    del eng
    return dt


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--mb", type=int, default=100)
    ap.add_argument("--chunks", default="256K,512K,1M,2M,4M,8M")
    ap.add_argument("--grids", default="1x8,2x8,4x8,4x16",
                    help="N_GROUPSxWGS_PER_GROUP combos")
    ap.add_argument("--out", default="")
    args = ap.parse_args()

    from allreduce_sweep import parse_sizes

    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    local = int(os.environ.get("LOCAL_RANK", rank))
    torch.cuda.set_device(local % max(1, torch.cuda.device_count()))
    dist.init_process_group("gloo")  # bootstrap channel only

    payload = args.mb * (1 << 20) // 4
    busfac = 2 * (world - 1) / world
    rows = ["grid,chunk_bytes,ms,busbw_GBps"]
    for grid in args.grids.split(","):
        ng, wg = grid.strip().split("x")
        os.environ["ADAPCC_N_GROUPS"] = ng
        os.environ["ADAPCC_WGS_PER_GROUP"] = wg
        for cb in parse_sizes(args.chunks):
            dt = bench_case(rank, world, payload, cb)
            bus = payload * 4 * busfac / dt / 1e9
            if rank == 0:
                rows.append(f"{grid},{cb},{dt*1e3:.3f},{bus:.1f}")
                print(rows[-1], flush=True)
    if rank == 0 and args.out:
        os.makedirs(os.path.dirname(args.out) or ".", exist_ok=True)
        with open(args.out, "w") as f:
            f.write("\n".join(rows) + "\n")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
