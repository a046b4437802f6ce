import math

from adapcc_amd.strategy.codegen import (
    binomial_rounds,
    broadcast_rate_bound,
    fully_connected_arcs,
    max_flow,
    parse_arc_file,
    ring_arcs,
)


def test_max_flow_fully_connected():
    # 4 nodes, unit caps: max flow 0->3 = direct 1 + two 2-hop paths = 3
    arcs = fully_connected_arcs(4)
    assert math.isclose(max_flow(arcs, 0, 3), 3.0, rel_tol=1e-6)


def test_broadcast_bound_mesh_vs_ring():
    # fully connected 8-node mesh: bound (n-1)*cap = 7; ring: 2 (both dirs)
    assert math.isclose(broadcast_rate_bound(fully_connected_arcs(8), 0),
                        7.0, rel_tol=1e-6)
    assert math.isclose(broadcast_rate_bound(ring_arcs(8), 0), 2.0,
                        rel_tol=1e-6)


def test_binomial_rounds():
    rounds = binomial_rounds(8, root=0)
    assert len(rounds) == 3  # log2(8)
    have = {0}
    for sends in rounds:
        for (s, d) in sends:
            assert s in have
            have.add(d)
    assert have == set(range(8))


def test_parse_arc_file():
    arcs = parse_arc_file("0 1 100.0\n1 0 100.0\n# comment\n0 2 50\n")
    assert arcs == [(0, 1, 100.0), (1, 0, 100.0), (0, 2, 50.0)]
