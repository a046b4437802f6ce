"""Detect-time link probing, health classification, and the end-to-end
degraded-link path: detect -> graph XML -> merged profile -> re-weighted
strategy (round-1 VERDICT item 5)."""

import os

import pytest

from adapcc_amd.strategy.synthesizer import Synthesizer, _merge_link_facts
from adapcc_amd.topology.detect import _bw_overrides, detect_node_topology
from adapcc_amd.topology.formats import (
    Link,
    LogicalGraph,
    dump_logical_graph,
    load_logical_graph,
    single_node_graph,
)


def _fake_graph_with_links(world=4, slow=None, bw=150.0, slow_bw=20.0):
    g = single_node_graph(world)
    for a in range(world):
        for b in range(world):
            if a == b:
                continue
            val = slow_bw if slow and (a, b) in slow else bw
            g.links[(a, b)] = Link(src=a, dst=b, bw_gbps=val,
                                   healthy=val >= 0.5 * bw)
    return g


def test_bw_override_parsing(monkeypatch):
    monkeypatch.setenv("ADAPCC_LINK_BW_OVERRIDE", "0-1:20, 1-0:22.5")
    assert _bw_overrides() == {(0, 1): 20.0, (1, 0): 22.5}
    monkeypatch.setenv("ADAPCC_LINK_BW_OVERRIDE", "garbage")
    with pytest.raises(ValueError):
        _bw_overrides()


def test_detect_records_injected_link_bandwidths():
    """Without torch.distributed, injected overrides still land on the
    graph with health classification (fault-injection path)."""
    g = detect_node_topology(
        0, 0, 1, probe_bandwidth=False,
        bw_overrides={(0, 1): 150.0, (1, 0): 150.0, (0, 2): 150.0,
                      (2, 0): 150.0, (1, 2): 20.0, (2, 1): 150.0})
    assert g.links[(1, 2)].bw_gbps == 20.0
    assert not g.links[(1, 2)].healthy
    assert g.links[(0, 1)].healthy


def test_link_graph_xml_roundtrip(tmp_path):
    g = _fake_graph_with_links(4, slow={(0, 1)})
    path = str(tmp_path / "graph.xml")
    dump_logical_graph(g, path)
    g2 = load_logical_graph(path)
    assert len(g2.links) == 12
    assert g2.links[(0, 1)].bw_gbps == pytest.approx(20.0)
    assert not g2.links[(0, 1)].healthy
    assert g2.links[(2, 3)].healthy


def test_reference_graph_without_links_still_loads(tmp_path):
    g = single_node_graph(4)
    path = str(tmp_path / "graph.xml")
    dump_logical_graph(g, path)
    g2 = load_logical_graph(path)
    assert g2.links == {}


def test_merge_link_facts_degrades_bandwidth():
    g = _fake_graph_with_links(4, slow={(0, 1), (1, 0)})
    prof = _merge_link_facts(g, None)
    # unhealthy link: probe bw then halved again
    assert prof.bandwidth[(0, 1)] == pytest.approx(10.0)
    assert prof.bandwidth[(2, 3)] == pytest.approx(150.0)


def test_degraded_link_reweights_strategy_end_to_end():
    """The acceptance test from the round-1 verdict: an injected slow link
    must produce a visibly re-weighted strategy through the full
    detect-graph-synthesize path."""
    g = _fake_graph_with_links(4, slow={(0, 1), (1, 0)})
    strat = Synthesizer(policy="par-trees").generate_strategy(graph=g)
    strat.validate(4)
    assert strat.slice_weights is not None, \
        "degraded link must produce non-uniform slice weights"
    w = strat.slice_weights
    # star trees rooted at 0 and 1 traverse the slow link
    assert max(w[0], w[1]) < min(w[2], w[3]), w

    # healthy mesh: no reweighting
    g2 = _fake_graph_with_links(4)
    strat2 = Synthesizer(policy="par-trees").generate_strategy(graph=g2)
    assert strat2.slice_weights is None


def test_merge_link_facts_floors_unreachable_links():
    """No peer access + unhealthy => bandwidth floored near zero so no
    policy routes a trunk over the dead link."""
    g = single_node_graph(2)
    g.links[(0, 1)] = Link(src=0, dst=1, peer_access=False, healthy=False)
    g.links[(1, 0)] = Link(src=1, dst=0, bw_gbps=150.0, healthy=True)
    prof = _merge_link_facts(g, None)
    assert prof.bandwidth[(0, 1)] == pytest.approx(0.01)
    assert prof.bandwidth[(1, 0)] == pytest.approx(150.0)


def test_merge_keeps_minimum_of_probe_and_profile():
    from adapcc_amd.topology.formats import ProfileMatrices

    g = single_node_graph(2)
    g.links[(0, 1)] = Link(src=0, dst=1, bw_gbps=100.0, healthy=True)
    prof0 = ProfileMatrices()
    prof0.bandwidth[(0, 1)] = 80.0   # profile saw it slower
    merged = _merge_link_facts(g, prof0)
    assert merged.bandwidth[(0, 1)] == pytest.approx(80.0)
