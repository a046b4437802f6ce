"""Process launcher (reference: launcher.py + launch_script.sh).

The reference wrapped mpirun with UCX flags; MI355X-native launch is one
process per GPU via torch.distributed.run (RCCL rendezvous), keeping the
reference's 6-flag contract forwarded to the exec file:

    python -m adapcc_amd.launcher --exec_file train.py \
        --hosts 127.0.0.1:8 --port 18000 --entry_point -1 \
        --strategy_file s.xml --logical_graph g.xml \
        --parallel_degree 8 --profile_freq 500

Also writes and distributes the rank->ip table (reference launcher.py:64-83).
"""

from __future__ import annotations

import argparse
import os
import shlex
import subprocess
import sys
from typing import List, Tuple

from .dispatcher import Dispatcher
from .topology.formats import dump_ip_table


def parse_hosts(spec: str) -> List[Tuple[str, int]]:
    """"ip1:4,ip2:4" -> [(ip1, 4), (ip2, 4)]"""
    out = []
    for part in spec.split(","):
        part = part.strip()
        if not part:
            continue
        if ":" in part:
            ip, n = part.rsplit(":", 1)
            out.append((ip, int(n)))
        else:
            out.append((part, 1))
    return out


def build_ip_table(hosts: List[Tuple[str, int]]) -> List[str]:
    table = []
    for ip, slots in hosts:
        table.extend([ip] * slots)
    return table


def main(argv=None) -> int:
    p = argparse.ArgumentParser(description="adapcc_amd launcher")
    p.add_argument("--exec_file", required=True)
    p.add_argument("--hosts", default="127.0.0.1:8",
                   help="ip:slots[,ip:slots...]")
    p.add_argument("--master_port", type=int, default=29500)
    # the 6 forwarded flags (reference launcher.py:54-62)
    p.add_argument("--port", type=int, default=18000)
    p.add_argument("--entry_point", type=int, default=-1)
    p.add_argument("--strategy_file", default="")
    p.add_argument("--logical_graph", default="")
    p.add_argument("--parallel_degree", type=int, default=0)
    p.add_argument("--profile_freq", type=int, default=0)
    p.add_argument("--ip_table", default="topology/ip_table.txt")
    p.add_argument("--dry_run", action="store_true")
    p.add_argument("extra", nargs=argparse.REMAINDER,
                   help="extra args forwarded to exec_file")
    args = p.parse_args(argv)

    hosts = parse_hosts(args.hosts)
    table = build_ip_table(hosts)
    os.makedirs(os.path.dirname(args.ip_table) or ".", exist_ok=True)
    dump_ip_table(table, args.ip_table)
    Dispatcher([h for h, _ in hosts]).dispatch_ip_table(args.ip_table)

    if len(hosts) > 1:
        print("multi-node launch: run this command on every node with "
              "--node_rank set; single-node is the supported fast path",
              file=sys.stderr)
    ip0, nproc = hosts[0]

    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--nnodes=1", f"--nproc-per-node={nproc}",
        "--master-addr", "127.0.0.1" if ip0 in ("localhost",) else ip0,
        "--master-port", str(args.master_port),
        args.exec_file,
        "--port", str(args.port),
        "--entry_point", str(args.entry_point),
        "--strategy_file", args.strategy_file,
        "--logical_graph", args.logical_graph,
        "--parallel_degree", str(args.parallel_degree),
        "--profile_freq", str(args.profile_freq),
    ] + [a for a in args.extra if a != "--"]

    print("[adapcc launcher]", " ".join(shlex.quote(c) for c in cmd))
    if args.dry_run:
        return 0
    return subprocess.call(cmd)


if __name__ == "__main__":
    sys.exit(main())
