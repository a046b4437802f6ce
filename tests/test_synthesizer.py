from adapcc_amd.strategy.partrees import ParTrees, synthesize_chains, synthesize_stars
from adapcc_amd.strategy.synthesizer import Synthesizer
from adapcc_amd.topology.formats import (
    LogicalGraph,
    Nic,
    ProfileMatrices,
    Server,
    single_node_graph,
)


def test_star_forest_link_disjoint():
    """On the fully connected xGMI mesh, every directed link must carry
    exactly one reduce flow: child->root edges of tree t are (i, t) for all
    i != t, so across the 8 trees each ordered pair appears exactly once."""
    s = synthesize_stars(8)
    s.validate(8)
    edges = set()
    for t, tree in enumerate(s.trees):
        assert tree.rank == t
        for c in tree.children:
            assert not c.children
            edge = (c.rank, tree.rank)
            assert edge not in edges
            edges.add(edge)
    assert len(edges) == 8 * 7


def test_chains():
    s = synthesize_chains(4, num_trees=2)
    s.validate(4)
    # each tree is a path
    for tree in s.trees:
        node, depth = tree, 1
        while node.children:
            assert len(node.children) == 1
            node = node.children[0]
            depth += 1
        assert depth == 4


def test_partrees_single_node_is_stars():
    strat = ParTrees(single_node_graph(8)).optimize()
    strat.validate(8)
    assert strat.num_trees == 8
    for t, tree in enumerate(strat.trees):
        assert tree.rank == t
        assert all(not c.children for c in tree.children)


def test_partrees_multi_node():
    g = LogicalGraph(servers=[
        Server(0, "10.0.0.1", [Nic(0, [0, 1, 2, 3])]),
        Server(1, "10.0.0.2", [Nic(1, [4, 5, 6, 7])]),
    ])
    prof = ProfileMatrices(bandwidth={(0, 4): 10.0}, latency={(0, 4): 50.0})
    strat = ParTrees(g, prof, parallel_degree=2).optimize()
    strat.validate(8)
    assert strat.num_trees == 2
    # roots rotate across trees
    assert strat.trees[0].rank != strat.trees[1].rank


def test_synthesizer_policies():
    for policy in ("par-trees", "stars", "chains", "milp"):
        syn = Synthesizer(policy=policy, parallel_degree=2)
        strat = syn.generate_strategy(world_size=4)
        strat.validate(4)


def test_synthesizer_writes_file(tmp_path):
    syn = Synthesizer(policy="stars")
    out = str(tmp_path / "strategy.xml")
    syn.generate_strategy(world_size=4, out_path=out)
    from adapcc_amd.topology.formats import load_strategy

    load_strategy(out).validate(4)
