"""Optimization-based strategy synthesis (reference: gurobi/solver.py).

The reference formulated tree routing + chunking as a Gurobi MILP (binary
routing vars per (edge, tree, flow), root selection, per-tree data split,
link-load accumulation, completion-time recursion, pipelined objective
min T_max + num_chunks * T_bottleneck — solver.py:11-211).

Here the same decision — which forest, which per-tree split, what chunk
size — is made two ways and the better one wins under the measured link
model:

1. a fixed candidate portfolio (star / chain / binary forests at several
   parallel degrees), exact-evaluated — fast, optimal on a homogeneous
   fully connected xGMI node;
2. a real MILP (``solve_milp``, scipy/HiGHS — gurobi is not available in
   this environment and scipy's branch-and-bound covers this model class):
   binary tree selection over a structural pool plus continuous per-tree
   split with a min-max link-completion objective, which adapts the forest
   and the split to degraded links the fixed portfolio cannot express
   (e.g. one slow GPU pulling several links down).

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
        cands = self.candidates()
        exact = self.solve_milp()
        if exact is not None:
            cands.append(exact)
        for cand in cands:
            for cb in chunk_grid:
                cost = self._weighted_cost(cand, cb)
                if cost < best_cost:
                    best_cost = cost
                    cand.chunk_bytes = cb
                    best = cand
        assert best is not None
        if best.slice_weights is None:
            self._set_slice_weights(best)
        return best

    def _weighted_cost(self, strategy: Strategy, chunk_bytes: int) -> float:
        """Cost model honoring per-tree slice weights (the MILP candidate
        carries non-uniform splits; the uniform `evaluate` would misprice
        it)."""
        T = strategy.num_trees
        weights = strategy.slice_weights or [1.0] * T
        total_w = sum(weights)
        load: Dict[Tuple[int, int], float] = {}
        depth_max = 0
        n_chunks_max = 1
        for i, tree in enumerate(strategy.trees):
            slice_bytes = self.payload * weights[i] / total_w
            n_chunks = max(1, math.ceil(slice_bytes / chunk_bytes))
            n_chunks_max = max(n_chunks_max, n_chunks)
            cb = slice_bytes / n_chunks
            edges: List[Tuple[int, int]] = []
            depth = self._edges(tree, edges)
            depth_max = max(depth_max, depth)
            for (src, dst) in edges:
                load[(src, dst)] = load.get((src, dst), 0.0) + cb

        def link_time(src: int, dst: int, bytes_on_link: float) -> float:
            return bytes_on_link / self._bw(src, dst) / 1000.0 + \
                self._lat(src, dst)

        bottleneck = max(
            link_time(s, d, b) for (s, d), b in load.items()
        ) if load else 0.0
        return 2 * (depth_max * bottleneck + (n_chunks_max - 1) * bottleneck)

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

    # -- exact MILP over a candidate tree pool ----------------------------

    def _tree_pool(self) -> List[TreeNode]:
        """Candidate tree pool: stars rooted at every rank, rotated chains
        and rotated binary trees — the same structural families as the
        reference's routing space (gurobi/solver.py builds trees from
        per-edge routing vars; here each pool entry is one realizable
        tree and the MILP picks the forest and the per-tree split)."""
        w = self.world
        pool: List[TreeNode] = []
        for root in range(w):
            star = TreeNode(rank=root)
            for r in range(w):
                if r != root:
                    star.children.append(TreeNode(rank=r))
            pool.append(star)
        for t in range(w):
            order = [(t + i) % w for i in range(w)]
            chain = TreeNode(rank=order[0])
            cur = chain
            for r in order[1:]:
                nxt = TreeNode(rank=r)
                cur.children.append(nxt)
                cur = nxt
            pool.append(chain)
            nodes = [TreeNode(rank=r) for r in order]
            for i in range(1, w):
                nodes[(i - 1) // 2].children.append(nodes[i])
            pool.append(nodes[0])
        return pool

    def solve_milp(self, max_trees: int = 8) -> Optional[Strategy]:
        """Real MILP (scipy/HiGHS) replacing the reference's Gurobi model
        (gurobi/solver.py:11-211): binary tree selection y_t over the pool,
        continuous per-tree split s_t (the reference's s_m), min-max link
        completion time with per-link loads from both pipelined phases, and
        a per-tree latency penalty standing in for the chunk/flag overhead
        term (the reference's num_chunks * T_bottleneck coupling).

            min  T + eps * sum_t y_t
            s.t. T >= sum_t s_t * (B/bw_e) * uses(t, e)     for all links e
                 sum_t s_t = 1;  0 <= s_t <= y_t;  sum_t y_t <= max_trees

        Returns None when scipy's MILP is unavailable or infeasible.
        """
        try:
            import numpy as np
            from scipy.optimize import LinearConstraint, milp
        except ImportError:
            return None
        if self.world < 2:
            return None
        pool = self._tree_pool()
        n = len(pool)
        # per-tree directed-link usage for reduce (child->parent) plus
        # broadcast (parent->child)
        uses: List[Dict[Tuple[int, int], int]] = []
        depths: List[int] = []
        for tree in pool:
            edges: List[Tuple[int, int]] = []
            depth = self._edges(tree, edges)
            u: Dict[Tuple[int, int], int] = {}
            for (c, par) in edges:
                u[(c, par)] = u.get((c, par), 0) + 1   # reduce
                u[(par, c)] = u.get((par, c), 0) + 1   # bcast
            uses.append(u)
            depths.append(depth)
        links = sorted({e for u in uses for e in u})
        B = self.payload

        # variables: [s_0..s_{n-1}, y_0..y_{n-1}, T]
        nv = 2 * n + 1
        cost = np.zeros(nv)
        cost[2 * n] = 1.0                         # T (us)
        # per-tree flag/launch overhead (us): small enough never to beat
        # a real bandwidth difference, large enough to break ties toward
        # fewer trees
        cost[n:2 * n] = 1.0

        A, lb, ub = [], [], []
        # T >= sum_t s_t * B * uses / bw_e  ->  sum - T <= 0
        for (a, b) in links:
            row = np.zeros(nv)
            for t, u in enumerate(uses):
                if (a, b) in u:
                    row[t] = u[(a, b)] * B / self._bw(a, b) / 1000.0  # us
            row[2 * n] = -1.0
            A.append(row)
            lb.append(-np.inf)
            ub.append(0.0)
        # per-tree completion: T >= s_t * B/bw_bottleneck + y_t * depth-fill
        # (the chunk-pipeline fill of a depth-d tree: d chunk-times; the
        # reference's h_jf completion recursion collapses to this on a
        # per-tree basis)
        for t, tree in enumerate(pool):
            bw_bot = self._tree_bottleneck_bw(tree)
            fill = depths[t] * (self.chunk_bytes / bw_bot / 1000.0 +
                                _DEFAULT_LAT)
            row = np.zeros(nv)
            row[t] = B / bw_bot / 1000.0
            row[n + t] = fill
            row[2 * n] = -1.0
            A.append(row)
            lb.append(-np.inf)
            ub.append(0.0)
        # sum s = 1
        row = np.zeros(nv)
        row[:n] = 1.0
        A.append(row)
        lb.append(1.0)
        ub.append(1.0)
        # s_t - y_t <= 0
        for t in range(n):
            row = np.zeros(nv)
            row[t] = 1.0
            row[n + t] = -1.0
            A.append(row)
            lb.append(-np.inf)
            ub.append(0.0)
        # sum y <= max_trees
        row = np.zeros(nv)
        row[n:2 * n] = 1.0
        A.append(row)
        lb.append(0.0)
        ub.append(float(min(max_trees, 16)))

        integrality = np.zeros(nv)
        integrality[n:2 * n] = 1
        bounds_lb = np.zeros(nv)
        bounds_ub = np.ones(nv)
        bounds_ub[2 * n] = np.inf
        from scipy.optimize import Bounds

        res = milp(c=cost,
                   constraints=LinearConstraint(np.array(A), lb, ub),
                   integrality=integrality,
                   bounds=Bounds(bounds_lb, bounds_ub))
        if not res.success:
            return None
        s = res.x[:n]
        chosen = [t for t in range(n) if s[t] > 1e-6]
        if not chosen:
            return None
        import copy

        trees = [copy.deepcopy(pool[t]) for t in chosen]
        weights = [s[t] for t in chosen]
        mean = sum(weights) / len(weights)
        weights = [w / mean for w in weights]
        strat = Strategy(trees=trees, chunk_bytes=self.chunk_bytes)
        spread = (max(weights) - min(weights)) / max(weights)
        strat.slice_weights = weights if spread > 0.02 else None
        return strat
