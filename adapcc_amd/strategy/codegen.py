"""Broadcast-schedule exploration over arbitrary arc graphs (reference:
gurobi/code-gen/cvxpy-broadcast-multi-round.py — a CVXPY/networkx LP over
arc files like all-to-all.txt / 8-node-hgx.txt / ring-only.txt; a research
aside, not wired into the runtime).

This analog uses scipy.linprog:

- ``max_flow(arcs, s, t)``: LP max-flow on the capacity graph.
- ``broadcast_rate_bound(arcs, root)``: the classic cut bound — a
  broadcast from ``root`` cannot beat min over receivers of
  max-flow(root -> r) (edge version of Edmonds' theorem; tight for
  fractional multi-round schedules).
- ``binomial_rounds(world, root)``: the log2 multi-round schedule the
  reference's experiments compare against.

Arc-file format (reference's): one ``src dst capacity`` per line.
"""

from __future__ import annotations

from typing import List, Sequence, Tuple

import numpy as np
from scipy.optimize import linprog

Arc = Tuple[int, int, float]


def parse_arc_file(path_or_text: str) -> List[Arc]:
    import os

    text = (open(path_or_text).read()
            if os.path.exists(path_or_text) else path_or_text)
    arcs = []
    for line in text.splitlines():
        line = line.split("#")[0].strip()
        if not line:
            continue
        s, d, c = line.split()
        arcs.append((int(s), int(d), float(c)))
    return arcs


def fully_connected_arcs(world: int, cap: float = 1.0) -> List[Arc]:
    return [(s, d, cap) for s in range(world) for d in range(world) if s != d]


def ring_arcs(world: int, cap: float = 1.0) -> List[Arc]:
    out = []
    for r in range(world):
        out.append((r, (r + 1) % world, cap))
        out.append(((r + 1) % world, r, cap))
    return out


def max_flow(arcs: Sequence[Arc], src: int, dst: int) -> float:
    """LP max-flow: maximize net outflow of src s.t. conservation and
    capacity."""
    if src == dst:
        return float("inf")
    nodes = sorted({n for a in arcs for n in a[:2]})
    idx = {n: i for i, n in enumerate(nodes)}
    E = len(arcs)
    # variables: flow on each arc in [0, cap]; objective: maximize
    # sum(out of src) - sum(into src)
    c = np.zeros(E)
    for e, (s, d, cap) in enumerate(arcs):
        if s == src:
            c[e] -= 1.0
        if d == src:
            c[e] += 1.0
    # conservation at every node except src/dst
    rows = []
    rhs = []
    for n in nodes:
        if n in (src, dst):
            continue
        row = np.zeros(E)
        for e, (s, d, cap) in enumerate(arcs):
            if s == n:
                row[e] += 1.0
            if d == n:
                row[e] -= 1.0
        rows.append(row)
        rhs.append(0.0)
    bounds = [(0.0, cap) for (_, _, cap) in arcs]
    res = linprog(c, A_eq=np.array(rows) if rows else None,
                  b_eq=np.array(rhs) if rows else None, bounds=bounds,
                  method="highs")
    if not res.success:
        raise RuntimeError(f"max_flow LP failed: {res.message}")
    return -res.fun


def broadcast_rate_bound(arcs: Sequence[Arc], root: int) -> float:
    nodes = sorted({n for a in arcs for n in a[:2]})
    return min(max_flow(arcs, root, r) for r in nodes if r != root)


def binomial_rounds(world: int, root: int = 0) -> List[List[Tuple[int, int]]]:
    """Multi-round binomial broadcast: round k doubles the holder set."""
    have = [root]
    rounds: List[List[Tuple[int, int]]] = []
    rest = [r for r in range(world) if r != root]
    while rest:
        sends = []
        for h in list(have):
            if not rest:
                break
            nxt = rest.pop(0)
            sends.append((h, nxt))
            have.append(nxt)
        rounds.append(sends)
    return rounds
