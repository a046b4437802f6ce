from adapcc_amd.strategy.milp import MilpSolver
from adapcc_amd.strategy.synthesizer import Synthesizer
from adapcc_amd.topology.formats import ProfileMatrices, single_node_graph


def uniform_profile(world, bw=150.0, lat=10.0):
    prof = ProfileMatrices()
    for s in range(world):
        for d in range(world):
            if s != d:
                prof.bandwidth[(s, d)] = bw
                prof.latency[(s, d)] = lat
    return prof


def test_milp_prefers_stars_on_uniform_mesh():
    """On a homogeneous fully connected mesh the star forest (1-hop,
    link-disjoint) dominates chains and binary trees."""
    g = single_node_graph(8)
    solver = MilpSolver(g, uniform_profile(8))
    strat = solver.optimize()
    strat.validate(8)
    assert strat.num_trees == 8
    for t, tree in enumerate(strat.trees):
        assert all(not c.children for c in tree.children), "expected stars"


def test_milp_cost_model_orders_candidates():
    g = single_node_graph(8)
    solver = MilpSolver(g, uniform_profile(8))
    from adapcc_amd.strategy.partrees import synthesize_chains, synthesize_stars

    stars = solver.evaluate(synthesize_stars(8), 1 << 20)
    chain1 = solver.evaluate(synthesize_chains(8, num_trees=1), 1 << 20)
    assert stars < chain1  # one chain serializes the whole payload on 1 link


def test_milp_chunk_choice_balances_pipeline():
    g = single_node_graph(4)
    solver = MilpSolver(g, uniform_profile(4), payload_bytes=64 << 20)
    strat = solver.optimize()
    assert 256 << 10 <= strat.chunk_bytes <= 4 << 20


def test_milp_via_synthesizer_policy():
    syn = Synthesizer(policy="milp")
    strat = syn.generate_strategy(world_size=4)
    strat.validate(4)


def test_milp_degraded_link_avoids_stars_bottleneck():
    """If one directed link is 10x slower, the portfolio should still pick
    a valid strategy and cost must reflect the slow link."""
    prof = uniform_profile(4)
    prof.bandwidth[(0, 1)] = 15.0
    g = single_node_graph(4)
    solver = MilpSolver(g, prof)
    from adapcc_amd.strategy.partrees import synthesize_stars

    fast = MilpSolver(g, uniform_profile(4)).evaluate(synthesize_stars(4), 1 << 20)
    slow = solver.evaluate(synthesize_stars(4), 1 << 20)
    assert slow > fast
    strat = solver.optimize()
    strat.validate(4)


def test_milp_degraded_link_sets_slice_weights():
    """A slow 0<->1 link shrinks the slices of the trees that traverse it
    (trees rooted at 0 and 1), leaving trees 2/3 larger (reference
    solver.py's s_m per-tree data split)."""
    prof = uniform_profile(4)
    prof.bandwidth[(0, 1)] = 15.0
    prof.bandwidth[(1, 0)] = 15.0
    g = single_node_graph(4)
    strat = MilpSolver(g, prof).optimize()
    strat.validate(4)
    assert strat.slice_weights is not None
    w = strat.slice_weights
    # trees crossing the slow link carry less; equivalent fast trees may
    # split degenerately (MILP alternate optima), so compare groups
    assert max(w[0], w[1]) < min(w[2:]), w


def test_milp_homogeneous_keeps_equal_slices():
    g = single_node_graph(4)
    strat = MilpSolver(g, uniform_profile(4)).optimize()
    assert strat.slice_weights is None


def test_solve_milp_runs_and_is_valid():
    """The scipy MILP returns a valid strategy on a uniform mesh."""
    g = single_node_graph(8)
    solver = MilpSolver(g, uniform_profile(8))
    strat = solver.solve_milp()
    assert strat is not None
    strat.validate(8)
    assert 1 <= strat.num_trees <= 8


def test_solve_milp_beats_portfolio_on_heterogeneous_mesh():
    """One GPU with several degraded links: the MILP's mixed forest +
    split must price at or below every fixed-portfolio candidate
    (VERDICT round-1 item 6 acceptance)."""
    world = 8
    prof = uniform_profile(world)
    # GPU 3 has three degraded outgoing/incoming links
    for peer in (0, 1, 2):
        prof.bandwidth[(3, peer)] = 25.0
        prof.bandwidth[(peer, 3)] = 25.0
    g = single_node_graph(world)
    solver = MilpSolver(g, prof)
    exact = solver.solve_milp()
    assert exact is not None
    exact.validate(world)
    best_fixed = float("inf")
    for cand in solver.candidates():
        solver._set_slice_weights(cand)
        for cb in (1 << 20, 4 << 20):
            best_fixed = min(best_fixed, solver._weighted_cost(cand, cb))
    exact_cost = min(solver._weighted_cost(exact, 1 << 20),
                     solver._weighted_cost(exact, 4 << 20))
    assert exact_cost <= best_fixed * 1.001
    # and optimize() must end up no worse than the fixed portfolio
    chosen = solver.optimize()
    chosen.validate(world)
    assert min(solver._weighted_cost(chosen, cb)
               for cb in (256 << 10, 512 << 10, 1 << 20, 2 << 20, 4 << 20)) \
        <= best_fixed * 1.001


def test_solve_milp_shifts_split_away_from_slow_links():
    """With one very slow link, the per-tree split must not be uniform."""
    world = 4
    prof = uniform_profile(world)
    prof.bandwidth[(0, 1)] = 10.0
    prof.bandwidth[(1, 0)] = 10.0
    g = single_node_graph(world)
    solver = MilpSolver(g, prof)
    strat = solver.solve_milp()
    assert strat is not None
    strat.validate(world)
    # trees whose edges cross the 0<->1 link should carry less data
    assert strat.slice_weights is not None
