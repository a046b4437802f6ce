"""Strategy synthesis driver (reference: gurobi/synthesizer.py:44-62).

Policies:
- ``par-trees`` (default): profile-driven heuristic forest (partrees.py) —
  star forest on a single fully-connected xGMI node, BDP-sorted binary
  trees with intra-node chains across nodes.
- ``stars`` / ``chains``: force the named single-node shape.
- ``milp``: LP/MILP tree+chunk optimization (milp.py; uses gurobipy when
  importable, otherwise a scipy fallback). Falls back to ``par-trees`` if
  neither backend is available.
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
