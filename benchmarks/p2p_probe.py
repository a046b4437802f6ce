#!/usr/bin/env python3
"""xGMI link probe: pairwise latency/bandwidth matrix (reference:
csrc/profile.cu probes; MI355X has 7 point-to-point links per GPU).

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 benchmarks/p2p_probe.py \
        --out gpurun_out/topo_profile.csv
"""

from __future__ import annotations

import argparse
import os
import sys

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from adapcc_amd.topology.detect import detect_node_topology, local_peer_matrix
from adapcc_amd.topology.formats import dump_profile
from adapcc_amd.topology.profile import profile_links


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--out", default="topo_profile.csv")
    p.add_argument("--bw_mb", type=int, default=64)
    args = p.parse_args()

    rank = int(os.environ.get("RANK", 0))
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    use_cuda = torch.cuda.is_available()
    if use_cuda:
        torch.cuda.set_device(local_rank % max(1, torch.cuda.device_count()))
    if world > 1:
        backend = "nccl" if (use_cuda and torch.cuda.device_count() >= world) else "gloo"
        dist.init_process_group(backend)

    graph = detect_node_topology(rank, local_rank, world)
    prof = profile_links(rank, world, graph,
                         bw_elems=args.bw_mb * (1 << 20) // 4,
                         concurrent=False)  # unloaded per-link numbers
    if rank == 0:
        print("peer-access matrix:", local_peer_matrix())
        if prof.bandwidth:
            print(f"{'src':>4} {'dst':>4} {'lat us':>10} {'bw GB/s':>10}")
            for (s, d), bw in sorted(prof.bandwidth.items()):
                lat = prof.latency.get((s, d), float('nan'))
                print(f"{s:>4} {d:>4} {lat:>10.2f} {bw:>10.2f}")
        os.makedirs(os.path.dirname(args.out) or ".", exist_ok=True)
        dump_profile(prof, args.out)
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
