"""Pairwise link profiling (reference: csrc/profile.cu).

Probes latency (small tensor) and bandwidth (large tensor) between rank
pairs with timed torch.distributed point-to-point transfers — RCCL p2p over
xGMI on GPU, TCP on gloo (functional but only indicative on CPU).

Two schedules:

- ``concurrent=True`` (default): N-1 shifted rounds like the reference's
  packed ring rounds (profile.cu:119-158) — in round k every rank sends to
  ``(rank+k) % N`` and receives from ``(rank-k) % N`` simultaneously, so a
  full N-rank profile costs O(N) round trips instead of O(N^2). This is
  what ``reconstruct_topology`` uses mid-training (at 8 ranks it finishes
  in well under a second).
- ``concurrent=False``: each directed pair probed in isolation with
  barriers between — slower but gives unloaded per-link numbers; used by
  the offline p2p probe harness.

Results feed the synthesizer's bandwidth-delay ordering and are dumped in
the reference's CSV schema (src,dst,type,value).
"""

from __future__ import annotations

import time
from typing import Optional

import torch
import torch.distributed as dist

from .formats import LogicalGraph, ProfileMatrices

_LAT_ELEMS = 64            # reference: 64 floats (profile.cu:173)
_BW_ELEMS = 8 * 1024 * 1024  # 32 MB fp32 (reference used 20 Mi floats intra)
_REPS = 5


def _device() -> torch.device:
    if torch.cuda.is_available():
        return torch.device("cuda", torch.cuda.current_device())
    return torch.device("cpu")


def _timed_transfer(send_to: Optional[int], recv_from: Optional[int],
                    elems: int, device, group) -> float:
    """Timed one-way transfers with a closing ack so sender-side buffering
    cannot under-count: the sender's clock stops only after the receiver
    confirms the last payload."""
    t = torch.ones(elems, dtype=torch.float32, device=device)
    ack = torch.zeros(1, dtype=torch.float32, device=device)
    if device.type == "cuda":
        torch.cuda.synchronize()
    start = time.perf_counter()
    for _ in range(_REPS):
        if send_to is not None:
            dist.send(t, dst=send_to, group=group)
        if recv_from is not None:
            dist.recv(t, src=recv_from, group=group)
    if send_to is not None:
        dist.recv(ack, src=send_to, group=group)
    if recv_from is not None:
        dist.send(ack, dst=recv_from, group=group)
    if device.type == "cuda":
        torch.cuda.synchronize()
    return (time.perf_counter() - start) / _REPS


def _timed_round(dst: int, src: int, elems: int, device, group) -> float:
    """One shifted round: this rank sends to ``dst`` while receiving from
    ``src``, all ranks concurrently. Returns seconds per rep for this
    rank's outgoing transfer (clock stops when both directions drain)."""
    t_out = torch.ones(elems, dtype=torch.float32, device=device)
    t_in = torch.empty(elems, dtype=torch.float32, device=device)
    if device.type == "cuda":
        torch.cuda.synchronize()
    dist.barrier(group=group)
    start = time.perf_counter()
    for _ in range(_REPS):
        req = dist.isend(t_out, dst=dst, group=group)
        dist.recv(t_in, src=src, group=group)
        req.wait()
    if device.type == "cuda":
        torch.cuda.synchronize()
    return (time.perf_counter() - start) / _REPS


def profile_links(
    rank: int,
    world_size: int,
    graph: Optional[LogicalGraph] = None,
    group=None,
    bw_elems: int = _BW_ELEMS,
    concurrent: bool = True,
) -> ProfileMatrices:
    prof = ProfileMatrices()
    if world_size <= 1 or not dist.is_initialized():
        return prof

    device = _device()
    if concurrent:
        # N-1 shifted rounds, all ranks probing at once
        # (reference profile.cu:119-158 task packing).
        for k in range(1, world_size):
            dst = (rank + k) % world_size
            src = (rank - k) % world_size
            for elems, kind in ((_LAT_ELEMS, "latency"),
                                (bw_elems, "bandwidth")):
                dt = _timed_round(dst, src, elems, device, group)
                if kind == "latency":
                    prof.latency[(rank, dst)] = dt * 1e6  # us
                else:
                    prof.bandwidth[(rank, dst)] = (elems * 4) / dt / 1e9
    else:
        # Probe each directed pair in isolation (sequential,
        # barrier-separated) for unloaded per-link numbers.
        for src in range(world_size):
            for dst in range(world_size):
                if src == dst:
                    continue
                for elems, kind in ((_LAT_ELEMS, "latency"),
                                    (bw_elems, "bandwidth")):
                    if rank == src:
                        dt = _timed_transfer(dst, None, elems, device, group)
                    elif rank == dst:
                        dt = _timed_transfer(None, src, elems, device, group)
                    else:
                        dt = None
                    if dt is not None and rank == src:
                        if kind == "latency":
                            prof.latency[(src, dst)] = dt * 1e6  # us
                        else:
                            prof.bandwidth[(src, dst)] = \
                                (elems * 4) / dt / 1e9  # GB/s
                dist.barrier(group=group)

    # gather everyone's measurements on every rank
    all_profiles = [None] * world_size
    dist.all_gather_object(all_profiles, prof, group=group)
    merged = ProfileMatrices()
    for p in all_profiles:
        merged.merge(p)
    return merged
