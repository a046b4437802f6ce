"""Topology detection (reference: csrc/detect.cu).

The reference infers NUMA/NIC/PCIe-switch affinity with loopback and
contention micro-benchmarks because its clusters hang GPUs off PCIe switches.
On MI355X a node is a fully connected xGMI mesh, so detection reduces to:

1. node membership: group ranks by hostname (reference used a DJB2 hostname
   hash over MPI_Allgather, init.cu:21-51; here an object all_gather)
2. the peer-access matrix (hipDeviceCanAccessPeer via torch) to confirm the
   full mesh — any missing link demotes the pair to host-staged transport
3. per-link bandwidth (left to the profile module, probing over RCCL)

Emits the same logical-graph XML schema as the reference so existing files
interoperate.
"""

from __future__ import annotations

import socket
from typing import List, Optional, Tuple

import torch
import torch.distributed as dist

from .formats import LogicalGraph, Nic, Server


def local_peer_matrix() -> List[List[bool]]:
    """Peer-access matrix among visible devices (single process view)."""
    n = torch.cuda.device_count() if torch.cuda.is_available() else 0
    mat = [[False] * n for _ in range(n)]
    for i in range(n):
        for j in range(n):
            if i == j:
                mat[i][j] = True
            else:
                mat[i][j] = bool(torch.cuda.can_device_access_peer(i, j))
    return mat


def detect_node_topology(
    rank: int, local_rank: int, world_size: int, group=None
) -> LogicalGraph:
    """Build the cluster logical graph by grouping ranks by host."""
    host = socket.gethostname()
    try:
        ip = socket.gethostbyname(host)
    except OSError:
        ip = "127.0.0.1"
    info: List[Optional[Tuple[str, str, int]]] = [None] * world_size
    if world_size > 1 and dist.is_initialized():
        dist.all_gather_object(info, (host, ip, rank), group=group)
    else:
        info = [(host, ip, rank)]

    hosts: dict = {}
    for item in info:
        h, hip, r = item
        hosts.setdefault(h, (hip, []))[1].append(r)

    graph = LogicalGraph()
    for sid, (h, (hip, ranks)) in enumerate(sorted(hosts.items())):
        graph.servers.append(
            Server(server_id=sid, ip=hip, nics=[Nic(nic_id=sid, gpus=sorted(ranks))])
        )
    return graph
