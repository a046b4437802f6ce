"""Topology artifacts: formats, detection, link profiling."""

from .formats import LogicalGraph, ProfileMatrices, Strategy, TreeNode

__all__ = ["LogicalGraph", "ProfileMatrices", "Strategy", "TreeNode"]
