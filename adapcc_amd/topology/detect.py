"""Topology detection (reference: csrc/detect.cu).

The reference infers NUMA/NIC/PCIe-switch affinity with loopback and
contention micro-benchmarks because its clusters hang GPUs off PCIe
switches (detect.cu:209-427: measure, don't assume). On MI355X a node is a
fully connected xGMI mesh, so detection measures:

1. node membership: group ranks by hostname (reference used a DJB2 hostname
   hash over MPI_Allgather, init.cu:21-51; here an object all_gather)
2. the peer-access matrix (hipDeviceCanAccessPeer via torch) — a missing
   link is recorded unhealthy and the pair is demoted to host-staged
   transport
3. a per-link bandwidth micro-probe (short timed p2p transfers in shifted
   rounds) whose results are recorded on the graph's ``links`` map; links
   probing below ``ADAPCC_LINK_HEALTH_FRACTION`` (default 0.5) of the
   median link are marked unhealthy so the synthesizer deweights trees
   that traverse them (detect -> graph -> weights -> plan).

Fault injection for tests/ops: ``ADAPCC_LINK_BW_OVERRIDE="src-dst:GBps,…"``
forces probe results, letting a degraded link be simulated end-to-end.

Emits the reference's logical-graph XML schema plus ``<link>`` elements;
reference files without them still load.
"""

from __future__ import annotations

import os
import socket
from typing import Dict, List, Optional, Tuple

import torch
import torch.distributed as dist

from .formats import Link, LogicalGraph, Nic, Server


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


def _bw_overrides() -> Dict[Tuple[int, int], float]:
    """Parse ADAPCC_LINK_BW_OVERRIDE="0-1:20,1-0:20" (GB/s)."""
    spec = os.environ.get("ADAPCC_LINK_BW_OVERRIDE", "")
    out: Dict[Tuple[int, int], float] = {}
    for part in spec.split(","):
        part = part.strip()
        if not part:
            continue
        try:
            pair, bw = part.split(":")
            a, b = pair.split("-")
            out[(int(a), int(b))] = float(bw)
        except ValueError:
            raise ValueError(
                f"bad ADAPCC_LINK_BW_OVERRIDE entry {part!r} "
                "(want 'src-dst:GBps')")
    return out


def probe_links(rank: int, world_size: int, group=None,
                bw_elems: int = 2 * 1024 * 1024,
                overrides: Optional[Dict[Tuple[int, int], float]] = None,
                ) -> Dict[Tuple[int, int], float]:
    """Short per-link bandwidth micro-probe (8 MB fp32, shifted rounds so
    every rank probes concurrently). Cheaper and earlier than the profile
    module's full matrix; feeds link-health classification at detect time.
    """
    out: Dict[Tuple[int, int], float] = dict(overrides or {})
    if world_size <= 1 or not dist.is_initialized():
        return out
    from .profile import _timed_round, _device

    device = _device()
    for k in range(1, world_size):
        dst = (rank + k) % world_size
        src = (rank - k) % world_size
        dt = _timed_round(dst, src, bw_elems, device, group)
        out.setdefault((rank, dst), bw_elems * 4 / dt / 1e9)

    gathered: List[Optional[dict]] = [None] * world_size
    dist.all_gather_object(gathered, out, group=group)
    merged: Dict[Tuple[int, int], float] = {}
    for d in gathered:
        if d:
            merged.update(d)
    merged.update(overrides or {})
    return merged


def detect_node_topology(
    rank: int, local_rank: int, world_size: int, group=None,
    probe_bandwidth: bool = True,
    bw_overrides: Optional[Dict[Tuple[int, int], float]] = None,
) -> LogicalGraph:
    """Build the cluster logical graph: hostname grouping, peer-access
    matrix, and (on GPU multi-rank runs or with injected overrides) the
    per-link bandwidth probe with health classification."""
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

    # peer-access matrix for this node's local devices (reference detect
    # task analog; round-1 left local_peer_matrix un-wired)
    peer = local_peer_matrix()
    me = graph.server_of(rank) if graph.servers else None
    if me is not None and peer:
        local = me.gpus()
        n = min(len(local), len(peer))
        for i in range(n):
            for j in range(n):
                if i != j:
                    ri, rj = local[i], local[j]
                    graph.links[(ri, rj)] = Link(
                        src=ri, dst=rj, peer_access=peer[i][j],
                        healthy=peer[i][j])

    overrides = dict(_bw_overrides())
    if bw_overrides:
        overrides.update(bw_overrides)
    do_probe = probe_bandwidth and world_size > 1 and dist.is_initialized() \
        and torch.cuda.is_available()
    bws: Dict[Tuple[int, int], float] = {}
    if do_probe:
        bws = probe_links(rank, world_size, group=group, overrides=overrides)
    elif overrides:
        bws = overrides
    if bws:
        for (a, b), bw in bws.items():
            ln = graph.links.setdefault((a, b), Link(src=a, dst=b))
            ln.bw_gbps = bw
        frac = float(os.environ.get("ADAPCC_LINK_HEALTH_FRACTION", "0.5"))
        vals = sorted(bw for bw in bws.values())
        med = vals[len(vals) // 2]
        for ln in graph.links.values():
            if ln.bw_gbps is not None and ln.bw_gbps < frac * med:
                ln.healthy = False
    return graph
