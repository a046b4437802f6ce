"""Strategy synthesis driver (reference: gurobi/synthesizer.py:44-62).

Policies:
- ``par-trees`` (default): profile-driven heuristic forest (partrees.py) —
  star forest on a single fully-connected xGMI node, BDP-sorted binary
  trees with intra-node chains across nodes.
- ``stars`` / ``chains``: force the named single-node shape.
- ``milp``: exact optimization (milp.py: candidate-portfolio evaluation +
  a scipy/HiGHS MILP over a structural tree pool). Falls back to
  ``par-trees`` on failure.

All policies see the DETECTED link facts: the logical graph's per-link
bandwidth probes and health verdicts (topology/detect.py) are merged into
the profile matrices before synthesis, and a degraded link deweights the
trees that traverse it (slice weights / BDP ordering).
"""

from __future__ import annotations

from typing import List, Optional

from ..topology.formats import (
    LogicalGraph,
    ProfileMatrices,
    Strategy,
    dump_strategy,
    single_node_graph,
)
from .partrees import ParTrees, synthesize_chains, synthesize_stars


def _merge_link_facts(
    graph: LogicalGraph, profile: Optional[ProfileMatrices]
) -> Optional[ProfileMatrices]:
    """Fold detect-time link probes/health into the profile matrices: a
    link's effective bandwidth is the minimum of the detect probe and the
    profile measurement, and an unhealthy link is floored near zero so no
    policy routes a tree trunk over it."""
    if not getattr(graph, "links", None):
        return profile
    merged = ProfileMatrices()
    if profile is not None:
        merged.merge(profile)
    for (a, b), ln in graph.links.items():
        if not ln.healthy and not ln.peer_access:
            merged.bandwidth[(a, b)] = 0.01
            continue
        if ln.bw_gbps is not None:
            prev = merged.bandwidth.get((a, b))
            merged.bandwidth[(a, b)] = (ln.bw_gbps if prev is None
                                        else min(prev, ln.bw_gbps))
        if not ln.healthy:
            # degraded but usable: halve again so weights shift away
            merged.bandwidth[(a, b)] = merged.bandwidth.get((a, b), 10.0) * 0.5
    return merged


class Synthesizer:
    def __init__(
        self,
        policy: str = "par-trees",
        parallel_degree: int = 2,
        chunk_bytes: int = 4 * 1024 * 1024,
    ) -> None:
        self.policy = policy
        self.parallel_degree = parallel_degree
        self.chunk_bytes = chunk_bytes

    def generate_strategy(
        self,
        graph: Optional[LogicalGraph] = None,
        profile: Optional[ProfileMatrices] = None,
        world_size: Optional[int] = None,
        ips: Optional[List[str]] = None,
        out_path: Optional[str] = None,
    ) -> Strategy:
        if graph is None:
            if world_size is None:
                raise ValueError("need graph or world_size")
            graph = single_node_graph(world_size, ips[0] if ips else "127.0.0.1")
        world = len(graph.ranks())
        profile = _merge_link_facts(graph, profile)

        if self.policy == "stars":
            strat = synthesize_stars(world, ips=ips, chunk_bytes=self.chunk_bytes)
        elif self.policy == "chains":
            strat = synthesize_chains(
                world, ips=ips, num_trees=max(2, self.parallel_degree),
                chunk_bytes=self.chunk_bytes,
            )
        elif self.policy == "milp":
            strat = self._milp(graph, profile)
        else:  # par-trees
            strat = ParTrees(
                graph, profile, parallel_degree=self.parallel_degree,
                chunk_bytes=self.chunk_bytes,
            ).optimize()

        # degraded-link deweighting for the heuristic single-node policies
        # (the milp policy already weights; multi-node par-trees orders by
        # BDP from the same merged profile)
        if (strat.slice_weights is None and profile is not None
                and profile.bandwidth and len(graph.servers) <= 1):
            from .milp import MilpSolver

            MilpSolver(graph, profile,
                       chunk_bytes=self.chunk_bytes)._set_slice_weights(strat)

        strat.validate(world)
        if out_path:
            dump_strategy(strat, out_path)
        return strat

    def _milp(
        self, graph: LogicalGraph, profile: Optional[ProfileMatrices]
    ) -> Strategy:
        try:
            from .milp import MilpSolver

            return MilpSolver(
                graph, profile, parallel_degree=self.parallel_degree,
                chunk_bytes=self.chunk_bytes,
            ).optimize()
        except Exception:
            return ParTrees(
                graph, profile, parallel_degree=self.parallel_degree,
                chunk_bytes=self.chunk_bytes,
            ).optimize()
