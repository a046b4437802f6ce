import os

import pytest

from adapcc_amd.topology.formats import (
    LogicalGraph,
    Nic,
    ProfileMatrices,
    Server,
    Strategy,
    TreeNode,
    dump_ip_table,
    dump_logical_graph,
    dump_profile,
    dump_strategy,
    load_ip_table,
    load_logical_graph,
    load_profile,
    load_strategy,
    single_node_graph,
)


def star(root, ranks, ip="127.0.0.1"):
    node = TreeNode(rank=root, ip=ip)
    node.children = [TreeNode(rank=r, ip=ip) for r in ranks if r != root]
    return node


def test_strategy_roundtrip(tmp_path):
    strat = Strategy(trees=[star(t, range(4)) for t in range(4)], chunk_bytes=1 << 20)
    path = str(tmp_path / "s.xml")
    dump_strategy(strat, path)
    loaded = load_strategy(path)
    assert loaded.num_trees == 4
    assert loaded.chunk_bytes == 1 << 20
    loaded.validate(4)
    for t in range(4):
        assert loaded.trees[t].rank == t
        assert sorted(c.rank for c in loaded.trees[t].children) == [
            r for r in range(4) if r != t
        ]


def test_reference_strategy_schema_parses(tmp_path):
    # the reference's nested-gpu schema (strategy/strategy_test.xml shape)
    xml = """<?xml version="1.0" encoding="utf-8"?>
<trees>
    <root id="0" ip="10.0.0.1">
        <gpu id="1" ip="10.0.0.1">
            <gpu id="2" ip="10.0.0.1"/>
            <gpu id="3" ip="10.0.0.1"/>
        </gpu>
    </root>
    <root id="3" ip="10.0.0.1">
        <gpu id="2" ip="10.0.0.1">
            <gpu id="1" ip="10.0.0.1"/>
            <gpu id="0" ip="10.0.0.1"/>
        </gpu>
    </root>
</trees>"""
    p = tmp_path / "ref.xml"
    p.write_text(xml)
    s = load_strategy(str(p))
    assert s.num_trees == 2
    s.validate(4)
    roles = s.roles(2)
    assert roles[0].parent == 1 and roles[0].children == []
    assert roles[1].parent == 3 and sorted(roles[1].children) == [0, 1]


def test_strategy_validation_catches_mismatch():
    s = Strategy(trees=[star(0, range(4)), star(1, range(3))])
    with pytest.raises(ValueError):
        s.validate(4)


def test_logical_graph_roundtrip(tmp_path):
    g = single_node_graph(8, ip="10.1.2.3")
    path = str(tmp_path / "g.xml")
    dump_logical_graph(g, path)
    g2 = load_logical_graph(path)
    assert g2.ranks() == list(range(8))
    assert g2.servers[0].ip == "10.1.2.3"


def test_ip_table_roundtrip(tmp_path):
    ips = ["10.0.0.1"] * 4 + ["10.0.0.2"] * 4
    p = str(tmp_path / "ip_table.txt")
    dump_ip_table(ips, p)
    assert load_ip_table(p) == ips


def test_profile_roundtrip(tmp_path):
    prof = ProfileMatrices(
        latency={(0, 1): 12.5, (1, 0): 13.0},
        bandwidth={(0, 1): 150.0},
    )
    p = str(tmp_path / "prof.csv")
    dump_profile(prof, p)
    prof2 = load_profile(p)
    assert prof2.latency[(0, 1)] == pytest.approx(12.5)
    assert prof2.bandwidth[(0, 1)] == pytest.approx(150.0)


def test_shipped_strategy_files():
    import glob
    import os

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    files = glob.glob(os.path.join(repo, "strategy", "*.xml"))
    assert files, "shipped strategy examples missing"
    for f in files:
        s = load_strategy(f)
        s.validate(len(s.ranks()))


def test_strategy_ignores_unknown_attrs(tmp_path):
    xml = ("<trees chunk_bytes=\'1024\' future_attr=\'x\'>"
           "<root id=\'0\' ip=\'a\' extra=\'1\'><gpu id=\'1\' ip=\'a\'/></root>"
           "<root id=\'1\' ip=\'a\'><gpu id=\'0\' ip=\'a\'/></root></trees>")
    s = load_strategy(xml)
    s.validate(2)
    assert s.chunk_bytes == 1024
