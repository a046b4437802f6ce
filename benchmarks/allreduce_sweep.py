#!/usr/bin/env python3
"""All-reduce bus-bandwidth sweep (reference: nccl-perf/benchmark busbw
methodology, PERFORMANCE.md: busbw = algbw * 2(n-1)/n for allreduce).

Sweeps buffer sizes 4 KB..1 GB on N ranks and reports algbw/busbw per size
for the selected transport(s):
  - native: the adapcc xGMI pull-engine
  - pg:     torch.distributed (RCCL on GPU) as the baseline

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 benchmarks/allreduce_sweep.py \
        --transports native,pg --out gpurun_out/busbw.csv
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def parse_sizes(spec: str):
    out = []
    for part in spec.split(","):
        part = part.strip().upper()
        mult = 1
        for suf, m in (("K", 1024), ("M", 1 << 20), ("G", 1 << 30)):
            if part.endswith(suf):
                mult = m
                part = part[:-1]
        out.append(int(float(part) * mult))
    return out


DEFAULT_SIZES = "4K,16K,64K,256K,1M,4M,16M,64M,256M,1G"


def _op_fn_native(eng, t, op, world):
    import torch as _t

    if op == "allreduce":
        return lambda: eng.all_reduce(t)
    if op == "broadcast":
        return lambda: eng.broadcast(t, root=0)
    if op == "reduce":
        return lambda: eng.reduce(t, root=0)
    if op == "allgather":
        out = _t.empty(t.numel() * world, dtype=t.dtype, device=t.device)
        return lambda: eng.all_gather(out, t)
    if op == "alltoall":
        out = _t.empty_like(t)
        return lambda: eng.all_to_all(out, t)
    if op == "reducescatter":
        out = _t.empty(t.numel() // world, dtype=t.dtype, device=t.device)
        return lambda: eng.reduce_scatter(out, t)
    raise ValueError(op)


def _op_fn_pg(t, op, world):
    import torch as _t

    if op == "allreduce":
        return lambda: dist.all_reduce(t)
    if op == "broadcast":
        return lambda: dist.broadcast(t, src=0)
    if op == "reduce":
        return lambda: dist.reduce(t, dst=0)
    if op == "allgather":
        out = _t.empty(t.numel() * world, dtype=t.dtype, device=t.device)
        return lambda: dist.all_gather_into_tensor(out, t)
    if op == "alltoall":
        out = _t.empty_like(t)
        return lambda: dist.all_to_all_single(out, t)
    if op == "reducescatter":
        out = _t.empty(t.numel() // world, dtype=t.dtype, device=t.device)
        return lambda: dist.reduce_scatter_tensor(out, t)
    raise ValueError(op)


# busbw factors per PERFORMANCE.md: allreduce 2(n-1)/n; AG/RS/alltoall
# (n-1)/n; broadcast/reduce 1
def bus_factor(op, world):
    if world <= 1:
        return 1.0
    if op == "allreduce":
        return 2 * (world - 1) / world
    if op in ("allgather", "reducescatter", "alltoall"):
        return (world - 1) / world
    return 1.0


def bench_transport(make_allreduce, sizes, iters, warmup, device, world):
    rows = []
    for size in sizes:
        n = size // 4
        t = torch.rand(n, device=device)
        fn = make_allreduce(t)
        for _ in range(warmup):
            fn()
        if device.type == "cuda":
            torch.cuda.synchronize()
        if world > 1:
            dist.barrier()
        t0 = time.perf_counter()
        for _ in range(iters):
            fn()
        if device.type == "cuda":
            torch.cuda.synchronize()
        if world > 1:
            dist.barrier()
        dt = (time.perf_counter() - t0) / iters
        # max over ranks
        if world > 1:
            m = torch.tensor([dt], dtype=torch.float64)
            dist.all_reduce(m, op=dist.ReduceOp.MAX)
            dt = float(m)
        algbw = size / dt / 1e9
        rows.append((size, dt * 1e6, algbw))
    return rows


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--sizes", default=DEFAULT_SIZES)
    p.add_argument("--iters", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--transports", default="native,pg")
    p.add_argument("--op", default="allreduce",
                   choices=["allreduce", "allgather", "alltoall",
                            "reducescatter", "broadcast", "reduce"])
    p.add_argument("--chunk_bytes", type=int, default=1 << 20)
    p.add_argument("--out", default="")
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
        backend = "nccl" if (use_cuda and torch.cuda.device_count() >= world) else "gloo"
        dist.init_process_group(backend)

    sizes = parse_sizes(args.sizes)
    results = {}

    for transport in args.transports.split(","):
        transport = transport.strip()
        if transport == "native":
            if not use_cuda:
                continue
            from adapcc_amd.runtime.engine import NativeEngine
            from adapcc_amd.strategy.partrees import synthesize_stars

            eng = NativeEngine(rank, world, device=device.index,
                               cap_bytes=max(sizes) + (1 << 20))
            eng.bootstrap()
            strat = synthesize_stars(world, chunk_bytes=args.chunk_bytes)
            eng.set_strategy(strat)

            def make(t, eng=eng, op=args.op, world=world):
                return _op_fn_native(eng, t, op, world)
        elif transport == "pg":
            if world == 1:
                continue

            def make(t, op=args.op, world=world):
                return _op_fn_pg(t, op, world)
        else:
            raise SystemExit(f"unknown transport {transport}")

        rows = bench_transport(make, sizes, args.iters, args.warmup, device,
                               world)
        bf = bus_factor(args.op, world)
        results[transport] = [(sz, us, ab, ab * bf) for sz, us, ab in rows]
        if transport == "native":
            eng.synchronize()

    if rank == 0:
        lines = ["transport,bytes,us,algbw_GBps,busbw_GBps"]
        for tr, rows in results.items():
            print(f"\n== {args.op} {tr} (n={world}) ==")
            print(f"{'bytes':>12} {'time(us)':>12} {'algbw GB/s':>12} "
                  f"{'busbw GB/s':>12}")
            for size, us, algbw, busbw in rows:
                print(f"{size:>12} {us:>12.1f} {algbw:>12.2f} {busbw:>12.2f}")
                lines.append(f"{tr},{size},{us:.2f},{algbw:.3f},{busbw:.3f}")
        if args.out:
            os.makedirs(os.path.dirname(args.out) or ".", exist_ok=True)
            with open(args.out, "w") as f:
                f.write("\n".join(lines) + "\n")
        summary = {
            "metric": f"{args.op}_busbw_GBps",
            "n_gpus": world,
            "results": {tr: [[s, round(b, 2)] for s, _, _, b in rows]
                        for tr, rows in results.items()},
        }
        print(json.dumps(summary))

    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
