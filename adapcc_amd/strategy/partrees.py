"""ParTrees: heuristic strategy synthesis for the xGMI mesh.

Reference equivalent: gurobi/trees.py (BDP-sorted inter-node binary trees with
intra-node chains, rotated roots, fixed 4 MiB chunks). That design targets
NVLink-switch + NIC clusters. The MI355X-native heuristic is different because
a single MI355X node is a *fully connected point-to-point* xGMI mesh: every
GPU has one direct ~153 GB/s link to each of the 7 others, and aggregate
bandwidth comes from keeping all 7 links of every GPU busy simultaneously.

Single-node policy (the headline case): emit ``world_size`` parallel
**star** trees, tree ``t`` rooted at rank ``t`` with every other rank a direct
child. Tree ``t`` owns slice ``t`` of the tensor, so the reduce phase is a
direct reduce-scatter (each GPU concurrently sends 7 distinct slices over its
7 distinct links) and the broadcast phase is a direct all-gather. Each
directed link carries exactly one flow per phase -> busbw ceiling
(n-1) * link_bw ~= 1.07 TB/s at 8 GPUs, vs link_bw (~153 GB/s) for a ring.

Multi-node policy: reference-style forest — one node-level binary tree per
parallel tree, node order sorted by bandwidth-delay product (reference:
trees.py:133), rotated per tree for root diversity (trees.py:137), each
node expanded into an intra-node chain of its local GPUs (trees.py:45-65).
"""

from __future__ import annotations

from typing import List, Optional

from ..topology.formats import (
    LogicalGraph,
    ProfileMatrices,
    Strategy,
    TreeNode,
)

DEFAULT_CHUNK_BYTES = 4 * 1024 * 1024


def synthesize_stars(
    world_size: int,
    ips: Optional[List[str]] = None,
    num_trees: Optional[int] = None,
    chunk_bytes: int = DEFAULT_CHUNK_BYTES,
) -> Strategy:
    """Parallel star forest for a fully connected intra-node mesh."""
    if ips is not None and not isinstance(ips, (list, tuple)):
        raise TypeError("ips must be a list of addresses")
    if ips is None:
        ips = ["127.0.0.1"] * world_size
    if num_trees is None:
        num_trees = world_size if world_size > 1 else 1
    trees = []
    for t in range(num_trees):
        root_rank = t % world_size
        root = TreeNode(rank=root_rank, ip=ips[root_rank])
        for r in range(world_size):
            if r != root_rank:
                root.children.append(TreeNode(rank=r, ip=ips[r]))
        trees.append(root)
    return Strategy(trees=trees, chunk_bytes=chunk_bytes)


def synthesize_chains(
    world_size: int,
    ips: Optional[List[str]] = None,
    num_trees: int = 2,
    chunk_bytes: int = DEFAULT_CHUNK_BYTES,
) -> Strategy:
    """Parallel pipelined chains (rotated). Mostly useful as a baseline and
    for bandwidth-asymmetric meshes; the star forest dominates on xGMI."""
    if ips is not None and not isinstance(ips, (list, tuple)):
        raise TypeError("ips must be a list of addresses")
    if ips is None:
        ips = ["127.0.0.1"] * world_size
    trees = []
    for t in range(num_trees):
        order = [(t + i) % world_size for i in range(world_size)]
        root = TreeNode(rank=order[0], ip=ips[order[0]])
        cur = root
        for r in order[1:]:
            nxt = TreeNode(rank=r, ip=ips[r])
            cur.children.append(nxt)
            cur = nxt
        trees.append(root)
    return Strategy(trees=trees, chunk_bytes=chunk_bytes)


def _bdp(
    prof: ProfileMatrices, a: int, b: int, default_bw: float, default_lat: float
) -> float:
    bw = prof.bandwidth.get((a, b), prof.bandwidth.get((b, a), default_bw))
    lat = prof.latency.get((a, b), prof.latency.get((b, a), default_lat))
    return bw * lat


class ParTrees:
    """Profile-driven forest synthesis (reference: gurobi/trees.py)."""

    def __init__(
        self,
        graph: LogicalGraph,
        profile: Optional[ProfileMatrices] = None,
        parallel_degree: int = 2,
        chunk_bytes: int = DEFAULT_CHUNK_BYTES,
    ) -> None:
        self.graph = graph
        self.profile = profile or ProfileMatrices()
        self.parallel_degree = max(1, parallel_degree)
        self.chunk_bytes = chunk_bytes

    def optimize(self) -> Strategy:
        servers = self.graph.servers
        if len(servers) <= 1:
            gpus = servers[0].gpus() if servers else []
            world = len(gpus)
            if gpus and sorted(gpus) != list(range(world)):
                # synthesize_stars names ranks 0..world-1; a hand-written
                # graph with non-zero-based ranks would silently get the
                # wrong rank names (advisor finding, round 1).
                raise ValueError(
                    "single-server logical graph must use global ranks "
                    f"0..{world - 1}, got {sorted(gpus)}")
            ips = {r: servers[0].ip for r in gpus} if servers else {}
            # Fully-connected xGMI: one star per rank keeps every directed
            # link carrying exactly one flow per phase.
            n_trees = max(1, world)
            strat = synthesize_stars(
                world, ips=[ips.get(r, "127.0.0.1") for r in range(world)],
                num_trees=n_trees, chunk_bytes=self._pick_chunk(world),
            )
            return strat
        return self._multi_node()

    def _pick_chunk(self, world: int) -> int:
        return self.chunk_bytes

    def _multi_node(self) -> Strategy:
        servers = list(self.graph.servers)
        # representative rank per server = its lowest local rank
        reps = [min(s.gpus()) for s in servers]
        # order servers by bandwidth-delay product of their rep's links
        default_bw, default_lat = 10.0, 100.0

        def score(i: int) -> float:
            tot = 0.0
            for j in range(len(servers)):
                if i != j:
                    tot += _bdp(self.profile, reps[i], reps[j], default_bw, default_lat)
            return tot

        order = sorted(range(len(servers)), key=score, reverse=True)
        trees: List[TreeNode] = []
        for t in range(self.parallel_degree):
            rotated = order[t % len(order):] + order[: t % len(order)]
            node_tree = self._binary_tree(rotated)
            trees.append(self._expand(node_tree, servers))
        return Strategy(trees=trees, chunk_bytes=self.chunk_bytes)

    def _binary_tree(self, order: List[int]) -> "_SNode":
        nodes = [_SNode(i) for i in order]
        for i, n in enumerate(nodes):
            left, right = 2 * i + 1, 2 * i + 2
            if left < len(nodes):
                n.children.append(nodes[left])
            if right < len(nodes):
                n.children.append(nodes[right])
        return nodes[0]

    def _expand(self, snode: "_SNode", servers) -> TreeNode:
        """Expand a server-level tree node into an intra-node chain of its
        GPUs (reference: trees.py:45-65 chain policy); child servers hang off
        the chain head."""
        server = servers[snode.idx]
        gpus = server.gpus()
        head = TreeNode(rank=gpus[0], ip=server.ip)
        cur = head
        for g in gpus[1:]:
            nxt = TreeNode(rank=g, ip=server.ip)
            cur.children.append(nxt)
            cur = nxt
        for child in snode.children:
            head.children.append(self._expand(child, servers))
        return head


class _SNode:
    def __init__(self, idx: int) -> None:
        self.idx = idx
        self.children: List["_SNode"] = []
