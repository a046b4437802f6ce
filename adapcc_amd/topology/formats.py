"""Strategy / topology artifact formats.

File-compatible with the reference's XML schemas so existing strategy and
logical-graph files work unchanged:

- strategy XML:     ``<trees><root id= ip=><gpu id= ip=>...</gpu></root></trees>``
  (reference: strategy/strategy_test.xml, parsed by csrc/allreduce.cu:52-104 treeDFS)
- logical graph:    ``<graph><server id= ip=><nic id=><gpu id=/></nic></server></graph>``
  (reference: topology/logical_graph_2n.xml)
- ip table:         one IP per line, line i = rank i's host
  (reference: topology/ip_table.txt, launcher.py:64-83)
- profile CSV:      ``src,dst,type,value`` with type in {latency, bandwidth}
  (reference: csrc/profile.cu:336-357)

Implemented with stdlib ElementTree (the reference used tinyxml2 + xmltodict).
"""

from __future__ import annotations

import os
import xml.etree.ElementTree as ET
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple


# ---------------------------------------------------------------------------
# Strategy (forest of trees)
# ---------------------------------------------------------------------------


@dataclass
class TreeNode:
    """One GPU in a communication tree. ``children`` are the node's
    precedents for reduce (data flows child -> parent) and its broadcast
    fan-out for the down phase."""

    rank: int
    ip: str = "127.0.0.1"
    children: List["TreeNode"] = field(default_factory=list)

    def all_ranks(self) -> List[int]:
        out = [self.rank]
        for c in self.children:
            out.extend(c.all_ranks())
        return out


@dataclass
class Strategy:
    """A forest of parallel trees; tree t owns slice t of every tensor.

    ``chunk_bytes`` is the default pipelining granule chosen by the
    synthesizer (reference: trees.py:118 default_chunk = 4 MiB)."""

    trees: List[TreeNode]
    chunk_bytes: int = 4 * 1024 * 1024
    # optional per-tree slice fractions (heterogeneous links): tree t owns
    # weight_t / sum(weights) of every tensor; None = equal slices
    slice_weights: Optional[List[float]] = None

    @property
    def num_trees(self) -> int:
        return len(self.trees)

    def ranks(self) -> List[int]:
        return sorted(set(self.trees[0].all_ranks())) if self.trees else []

    def validate(self, world_size: Optional[int] = None) -> None:
        if not self.trees:
            raise ValueError("strategy has no trees")
        base = sorted(self.trees[0].all_ranks())
        if len(base) != len(set(base)):
            raise ValueError("duplicate rank inside a tree")
        for i, t in enumerate(self.trees):
            r = sorted(t.all_ranks())
            if r != base:
                raise ValueError(f"tree {i} covers ranks {r} != tree 0 ranks {base}")
            if len(r) != len(set(r)):
                raise ValueError(f"duplicate rank inside tree {i}")
        if world_size is not None and base != list(range(world_size)):
            raise ValueError(
                f"strategy ranks {base} do not cover world size {world_size}"
            )
        if self.slice_weights is not None:
            if len(self.slice_weights) != len(self.trees):
                raise ValueError("slice_weights length != num_trees")
            if any(w <= 0 for w in self.slice_weights):
                raise ValueError("slice_weights must be positive")

    # -- roles -------------------------------------------------------------

    def roles(self, rank: int) -> List["TreeRole"]:
        """Per-tree role of ``rank``: its parent, children, and the index of
        this rank among its parent's children (the reference's siblingIdx,
        allreduce.cu:77-84, used to offset into the parent's staging buffer)."""
        out = []
        for tid, tree in enumerate(self.trees):
            role = _find_role(tree, rank, parent=None, sibling_idx=0)
            if role is None:
                raise ValueError(f"rank {rank} not in tree {tid}")
            out.append(role)
        return out


@dataclass
class TreeRole:
    rank: int
    root: int
    parent: Optional[int]  # None at root
    children: List[int]
    sibling_idx: int  # index among parent's children

    @property
    def is_root(self) -> bool:
        return self.parent is None


def _find_role(
    node: TreeNode, rank: int, parent: Optional[int], sibling_idx: int, root: Optional[int] = None
) -> Optional[TreeRole]:
    if root is None:
        root = node.rank
    if node.rank == rank:
        return TreeRole(
            rank=rank,
            root=root,
            parent=parent,
            children=[c.rank for c in node.children],
            sibling_idx=sibling_idx,
        )
    for i, c in enumerate(node.children):
        r = _find_role(c, rank, parent=node.rank, sibling_idx=i, root=root)
        if r is not None:
            return r
    return None


def _tree_to_xml(node: TreeNode, tag: str) -> ET.Element:
    el = ET.Element(tag, {"id": str(node.rank), "ip": node.ip})
    for c in node.children:
        el.append(_tree_to_xml(c, "gpu"))
    return el


def _tree_from_xml(el: ET.Element) -> TreeNode:
    node = TreeNode(rank=int(el.get("id")), ip=el.get("ip", "127.0.0.1"))
    for c in el:
        if c.tag == "gpu":
            node.children.append(_tree_from_xml(c))
    return node


def dump_strategy(strategy: Strategy, path: str) -> None:
    root = ET.Element("trees")
    if strategy.chunk_bytes:
        root.set("chunk_bytes", str(strategy.chunk_bytes))
    for i, t in enumerate(strategy.trees):
        el = _tree_to_xml(t, "root")
        if strategy.slice_weights is not None:
            el.set("weight", f"{strategy.slice_weights[i]:.6f}")
        root.append(el)
    _indent(root)
    data = ET.tostring(root, encoding="unicode", xml_declaration=False)
    with open(path, "w") as f:
        f.write('<?xml version="1.0" encoding="utf-8"?>\n')
        f.write(data)
        f.write("\n")


def load_strategy(path_or_text: str) -> Strategy:
    if os.path.exists(path_or_text):
        tree = ET.parse(path_or_text)
        root = tree.getroot()
    else:
        root = ET.fromstring(path_or_text)
    if root.tag != "trees":
        raise ValueError(f"expected <trees> root, got <{root.tag}>")
    chunk = int(root.get("chunk_bytes", 4 * 1024 * 1024))
    tree_els = [el for el in root if el.tag == "root"]
    trees = [_tree_from_xml(el) for el in tree_els]
    weights = None
    if any(el.get("weight") is not None for el in tree_els):
        weights = [float(el.get("weight", 1.0)) for el in tree_els]
    return Strategy(trees=trees, chunk_bytes=chunk, slice_weights=weights)


# ---------------------------------------------------------------------------
# Logical graph
# ---------------------------------------------------------------------------


@dataclass
class Link:
    """One directed GPU-GPU link as detected (xGMI on a node): peer access,
    the detect-time bandwidth micro-probe result, and a health verdict
    (False when the link probes well below its peers — the synthesizer
    deweights trees that traverse unhealthy links)."""

    src: int
    dst: int
    peer_access: bool = True
    bw_gbps: Optional[float] = None
    healthy: bool = True


@dataclass
class LogicalGraph:
    """Cluster layout: server -> nic -> gpus (global ranks), plus the
    detected link structure (an MI355X extension of the reference schema;
    files without <link> elements load with an empty link map)."""

    servers: List["Server"] = field(default_factory=list)
    version: str = "mi355x"
    links: Dict[Tuple[int, int], "Link"] = field(default_factory=dict)

    def ranks(self) -> List[int]:
        out: List[int] = []
        for s in self.servers:
            for n in s.nics:
                out.extend(n.gpus)
        return sorted(out)

    def server_of(self, rank: int) -> "Server":
        for s in self.servers:
            for n in s.nics:
                if rank in n.gpus:
                    return s
        raise KeyError(rank)


@dataclass
class Nic:
    nic_id: int
    gpus: List[int] = field(default_factory=list)


@dataclass
class Server:
    server_id: int
    ip: str
    nics: List[Nic] = field(default_factory=list)

    def gpus(self) -> List[int]:
        out: List[int] = []
        for n in self.nics:
            out.extend(n.gpus)
        return sorted(out)


def dump_logical_graph(graph: LogicalGraph, path: str) -> None:
    root = ET.Element("graph", {"version": graph.version})
    for s in graph.servers:
        sel = ET.SubElement(root, "server", {"id": str(s.server_id), "ip": s.ip})
        for n in s.nics:
            nel = ET.SubElement(sel, "nic", {"id": str(n.nic_id)})
            for g in n.gpus:
                ET.SubElement(nel, "gpu", {"id": str(g)})
    for (src, dst), ln in sorted(graph.links.items()):
        attrs = {"src": str(src), "dst": str(dst),
                 "peer": "1" if ln.peer_access else "0",
                 "healthy": "1" if ln.healthy else "0"}
        if ln.bw_gbps is not None:
            attrs["bw"] = f"{ln.bw_gbps:.2f}"
        ET.SubElement(root, "link", attrs)
    _indent(root)
    with open(path, "w") as f:
        f.write(ET.tostring(root, encoding="unicode"))
        f.write("\n")


def load_logical_graph(path: str) -> LogicalGraph:
    root = ET.parse(path).getroot()
    graph = LogicalGraph(version=root.get("version", ""))
    for sel in root.findall("server"):
        server = Server(server_id=int(sel.get("id")), ip=sel.get("ip", "127.0.0.1"))
        for nel in sel.findall("nic"):
            nic = Nic(nic_id=int(nel.get("id")))
            for gel in nel.findall("gpu"):
                nic.gpus.append(int(gel.get("id")))
            server.nics.append(nic)
        graph.servers.append(server)
    for lel in root.findall("link"):
        src, dst = int(lel.get("src")), int(lel.get("dst"))
        bw = lel.get("bw")
        graph.links[(src, dst)] = Link(
            src=src, dst=dst,
            peer_access=lel.get("peer", "1") == "1",
            bw_gbps=float(bw) if bw is not None else None,
            healthy=lel.get("healthy", "1") == "1",
        )
    return graph


def single_node_graph(world_size: int, ip: str = "127.0.0.1") -> LogicalGraph:
    """The degenerate single-server graph for one 8x MI355X node."""
    return LogicalGraph(
        servers=[Server(server_id=0, ip=ip, nics=[Nic(nic_id=0, gpus=list(range(world_size)))])]
    )


# ---------------------------------------------------------------------------
# IP table
# ---------------------------------------------------------------------------


def dump_ip_table(ips: List[str], path: str) -> None:
    with open(path, "w") as f:
        for ip in ips:
            f.write(ip + "\n")


def load_ip_table(path: str) -> List[str]:
    with open(path) as f:
        return [line.strip() for line in f if line.strip()]


# ---------------------------------------------------------------------------
# Profile matrices (latency / bandwidth)
# ---------------------------------------------------------------------------


@dataclass
class ProfileMatrices:
    """Pairwise link measurements. latency in microseconds, bandwidth in
    GB/s; keys are (src_rank, dst_rank)."""

    latency: Dict[Tuple[int, int], float] = field(default_factory=dict)
    bandwidth: Dict[Tuple[int, int], float] = field(default_factory=dict)

    def merge(self, other: "ProfileMatrices") -> None:
        self.latency.update(other.latency)
        self.bandwidth.update(other.bandwidth)


def dump_profile(prof: ProfileMatrices, path: str) -> None:
    with open(path, "w") as f:
        for (s, d), v in sorted(prof.latency.items()):
            f.write(f"{s},{d},latency,{v:.6f}\n")
        for (s, d), v in sorted(prof.bandwidth.items()):
            f.write(f"{s},{d},bandwidth,{v:.6f}\n")


def load_profile(path: str) -> ProfileMatrices:
    prof = ProfileMatrices()
    with open(path) as f:
        for line in f:
            line = line.strip()
            if not line:
                continue
            s, d, kind, v = line.split(",")
            key = (int(s), int(d))
            if kind == "latency":
                prof.latency[key] = float(v)
            elif kind == "bandwidth":
                prof.bandwidth[key] = float(v)
            else:
                raise ValueError(f"unknown profile row type {kind!r}")
    return prof


# ---------------------------------------------------------------------------
# helpers
# ---------------------------------------------------------------------------


def _indent(elem: ET.Element, level: int = 0) -> None:
    pad = "\n" + level * "    "
    if len(elem):
        if not elem.text or not elem.text.strip():
            elem.text = pad + "    "
        for child in elem:
            _indent(child, level + 1)
        if not child.tail or not child.tail.strip():
            child.tail = pad
    if level and (not elem.tail or not elem.tail.strip()):
        elem.tail = pad
