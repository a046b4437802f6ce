"""Multi-rank protocol simulation over the native engine's unit plans.

Drives the exact C++ plan/relay logic (adapcc_amd._core.compute_plan /
compute_primitive_plan — the same code the Engine runs on GPU) through a
discrete-event simulation with numpy buffers: units execute only when their
flag waits are satisfied, in worst-case serialized order, so completion
proves both numerical correctness and deadlock-freedom of the flag protocol
for every primitive / strategy / active-set combination below.
"""

import numpy as np
import pytest

core = pytest.importorskip("adapcc_amd._core")

SEND, ACC, RESULT = 0, 1, 2


def stars(n):
    return [[-1 if r == t else t for r in range(n)] for t in range(n)]


def chains(n, ntrees=2):
    out = []
    for t in range(ntrees):
        order = [(t + i) % n for i in range(n)]
        parents = [0] * n
        parents[order[0]] = -1
        for i in range(1, n):
            parents[order[i]] = order[i - 1]
        out.append(parents)
    return out


def binary_trees(n, ntrees=2):
    out = []
    for t in range(ntrees):
        order = [(t + i) % n for i in range(n)]
        parents = [0] * n
        parents[order[0]] = -1
        for i in range(1, n):
            parents[order[i]] = order[(i - 1) // 2]
        out.append(parents)
    return out


class Sim:
    """Execute per-rank unit plans against numpy buffers."""

    def __init__(self, world, plans, user_in, out_elems, scale=1.0):
        self.world = world
        self.plans = plans
        self.user = user_in
        total = max(len(u) for u in user_in)
        self.out = [np.full(out_elems, np.nan, dtype=np.float32)
                    for _ in range(world)]
        self.send = [np.zeros(total, dtype=np.float32) for _ in range(world)]
        self.acc = [np.zeros(total, dtype=np.float32) for _ in range(world)]
        self.result = [np.zeros(total, dtype=np.float32) for _ in range(world)]
        self.ready = set()
        self.bcast = set()
        self.scale = scale
        self.queues = []
        for r in range(world):
            p = plans[r]
            self.queues.append({
                "red": list(p["copy"]) + list(p["reduce"]),
                "red_kind": ["copy"] * len(p["copy"]) + ["reduce"] * len(p["reduce"]),
                "red_i": 0,
                "bc": list(p["bcast"]),
                "bc_i": 0,
            })

    def buf(self, kind, r):
        return {SEND: self.send, ACC: self.acc, RESULT: self.result}[kind][r]

    def try_red(self, r):
        q = self.queues[r]
        i = q["red_i"]
        if i >= len(q["red"]):
            return False
        u, kind = q["red"][i], q["red_kind"][i]
        t, c = u["tree"], u["chunk"]
        if kind == "copy":
            off, cnt = u["offset"], u["count"]
            self.send[r][off:off + cnt] = self.user[r][off:off + cnt]
            for peer in u["notify_to"]:
                if u["flag_space"] == 0:
                    self.ready.add((peer, r, t, c))
                else:
                    self.bcast.add((peer, t, c))
        else:
            off, cnt = u["offset"], u["count"]
            for (sr, sk) in u["srcs"]:
                if sr != r and (r, sr, t, c) not in self.ready:
                    return False
            pieces = [self.buf(sk, sr)[off:off + cnt] for (sr, sk) in u["srcs"]]
            if u["include_self"]:
                pieces.append(self.send[r][off:off + cnt])
            assert pieces, "reduce unit with no sources"
            self.acc[r][off:off + cnt] = np.sum(pieces, axis=0)
            if u["notify"]:
                self.ready.add((u["consumer"], r, t, c))
            if u["is_root"]:
                for k in u["publish_to"]:
                    self.bcast.add((k, t, c))
        q["red_i"] += 1
        return True

    def try_bc(self, r):
        q = self.queues[r]
        i = q["bc_i"]
        if i >= len(q["bc"]):
            return False
        u = q["bc"][i]
        t, c = u["tree"], u["chunk"]
        if u["parent"] >= 0 or len(u["publish_to"]) or True:
            # self-pulls (parent == -1) still wait on the bcast flag: the
            # producing unit pushes to self
            if (r, t, c) not in self.bcast:
                return False
        soff, doff, cnt = u["src_offset"], u["dst_offset"], u["count"]
        src_rank = r if u["parent"] < 0 else u["parent"]
        src = self.buf(u["parent_kind"], src_rank)[soff:soff + cnt]
        self.out[r][doff:doff + cnt] = src * self.scale
        if u["forward"]:
            self.result[r][soff:soff + cnt] = src
            for k in u["publish_to"]:
                self.bcast.add((k, t, c))
        q["bc_i"] += 1
        return True

    def run(self):
        while True:
            progress = False
            for r in range(self.world):
                while self.try_red(r):
                    progress = True
                while self.try_bc(r):
                    progress = True
            if all(q["red_i"] == len(q["red"]) and q["bc_i"] == len(q["bc"])
                   for q in self.queues):
                return
            if not progress:
                raise AssertionError("protocol deadlock: no runnable unit")


def rand_inputs(world, n, seed=0):
    rng = np.random.default_rng(seed)
    return [rng.standard_normal(n).astype(np.float32) for _ in range(world)]


# ---------------------------------------------------------------------------
# allreduce
# ---------------------------------------------------------------------------


def sim_allreduce(parents, total, active=None, chunk_bytes=256, average=False):
    world = len(parents[0])
    act = active if active is not None else list(range(world))
    plans = [core.compute_plan(parents, r, total, 4, chunk_bytes, act)
             for r in range(world)]
    user = rand_inputs(world, total)
    scale = 1.0 / len(act) if average else 1.0
    sim = Sim(world, plans, user, total, scale)
    sim.run()
    expect = np.sum([user[r] for r in act], axis=0) * scale
    for r in range(world):
        np.testing.assert_allclose(sim.out[r], expect, rtol=1e-5, atol=1e-5)


@pytest.mark.parametrize("world", [2, 3, 4, 8])
def test_stars_allreduce(world):
    sim_allreduce(stars(world), total=1000)


@pytest.mark.parametrize("world", [2, 4, 8])
def test_chains_allreduce(world):
    sim_allreduce(chains(world, ntrees=2), total=777)


@pytest.mark.parametrize("world", [4, 8])
def test_binary_trees_allreduce(world):
    sim_allreduce(binary_trees(world, ntrees=3), total=513)


def test_single_tree():
    sim_allreduce([[1, -1, 1, 1]], total=300)


def test_average():
    sim_allreduce(stars(4), total=256, average=True)


def test_tiny_tensor_smaller_than_slices():
    sim_allreduce(stars(8), total=10)


@pytest.mark.parametrize("inactive", [[0], [3], [1, 2]])
def test_relay_star_inactive(inactive):
    world = 4
    active = [r for r in range(world) if r not in inactive]
    sim_allreduce(stars(world), total=512, active=active)


def test_relay_chain_passthrough():
    sim_allreduce([[-1, 0, 1, 2]], total=512, active=[0, 2, 3])


def test_relay_inactive_aggregator():
    sim_allreduce([[-1, 0, 0, 1, 1, 2, 2, 3]], total=640,
                  active=[1, 2, 3, 4, 5, 6, 7])


def test_relay_only_one_active():
    sim_allreduce(stars(4), total=128, active=[2])


def test_plan_deterministic_across_ranks():
    parents = stars(4)
    plans = [core.compute_plan(parents, r, 5000, 4, 256) for r in range(4)]
    grids = [
        sorted((u["tree"], u["chunk"], u["src_offset"], u["count"])
               for u in p["bcast"])
        for p in plans
    ]
    assert all(g == grids[0] for g in grids)


def test_malformed_strategy_raises():
    with pytest.raises(Exception):
        core.compute_plan([[0, 1, 2, 3]], 0, 100, 4, 256)
    with pytest.raises(Exception):
        core.compute_plan([[-1, -1, 0, 0]], 0, 100, 4, 256)


# ---------------------------------------------------------------------------
# other primitives
# ---------------------------------------------------------------------------


def prim_plans(prim, world, elems, chunk_bytes=256, root=0, active=()):
    return [
        core.compute_primitive_plan(prim, world, r, elems, 4, chunk_bytes,
                                    root=root, active=list(active))
        for r in range(world)
    ]


@pytest.mark.parametrize("world,root", [(2, 0), (4, 2), (8, 7)])
def test_reduce(world, root):
    total = 900
    plans = prim_plans("reduce", world, total, root=root)
    user = rand_inputs(world, total)
    sim = Sim(world, plans, user, total)
    sim.run()
    expect = np.sum(user, axis=0)
    np.testing.assert_allclose(sim.out[root], expect, rtol=1e-5, atol=1e-5)
    for r in range(world):
        if r != root:
            assert np.isnan(sim.out[r]).all()  # only root receives


def test_reduce_with_relay():
    world, root = 4, 1
    active = [0, 2, 3]
    plans = prim_plans("reduce", world, 500, root=root, active=active)
    user = rand_inputs(world, 500)
    sim = Sim(world, plans, user, 500)
    sim.run()
    expect = np.sum([user[r] for r in active], axis=0)
    np.testing.assert_allclose(sim.out[root], expect, rtol=1e-5, atol=1e-5)


@pytest.mark.parametrize("world,root", [(2, 1), (4, 0), (8, 3)])
def test_broadcast(world, root):
    total = 700
    plans = prim_plans("broadcast", world, total, root=root)
    user = rand_inputs(world, total)
    sim = Sim(world, plans, user, total)
    sim.run()
    for r in range(world):
        if r != root:
            np.testing.assert_allclose(sim.out[r], user[root], rtol=1e-6,
                                       atol=1e-6)


@pytest.mark.parametrize("world", [2, 4, 8])
def test_allgather(world):
    L = 300
    plans = prim_plans("allgather", world, L)
    user = rand_inputs(world, L)
    sim = Sim(world, plans, user, world * L)
    sim.run()
    expect = np.concatenate(user)
    for r in range(world):
        np.testing.assert_allclose(sim.out[r], expect, rtol=1e-6, atol=1e-6)


@pytest.mark.parametrize("world", [2, 4, 8])
def test_reducescatter(world):
    L = 256
    plans = prim_plans("reducescatter", world, L)
    user = rand_inputs(world, world * L)
    sim = Sim(world, plans, user, L)
    sim.run()
    total = np.sum(user, axis=0)
    for r in range(world):
        np.testing.assert_allclose(sim.out[r], total[r * L:(r + 1) * L],
                                   rtol=1e-5, atol=1e-5)


@pytest.mark.parametrize("world", [2, 4, 8])
def test_alltoall(world):
    L = 200
    plans = prim_plans("alltoall", world, L)
    user = rand_inputs(world, world * L)
    sim = Sim(world, plans, user, world * L)
    sim.run()
    for r in range(world):
        expect = np.concatenate([user[s][r * L:(r + 1) * L]
                                 for s in range(world)])
        np.testing.assert_allclose(sim.out[r], expect, rtol=1e-6, atol=1e-6)


# ---------------------------------------------------------------------------
# additional edge coverage
# ---------------------------------------------------------------------------


def test_allreduce_exact_capacity_chunks():
    # slice == exactly kMaxChunkSlots chunks boundary: chunk auto-growth
    world = 2
    total = 2 * 64 * 600  # forces > 512 chunks at chunk_bytes=256 -> growth
    sim_allreduce(stars(world), total=total, chunk_bytes=256)


def test_allgather_multi_chunk():
    world = 4
    L = 3000
    plans = prim_plans("allgather", world, L, chunk_bytes=1024)
    user = rand_inputs(world, L)
    sim = Sim(world, plans, user, world * L)
    sim.run()
    expect = np.concatenate(user)
    for r in range(world):
        np.testing.assert_allclose(sim.out[r], expect, rtol=1e-6, atol=1e-6)


def test_alltoall_multi_chunk_slots():
    world = 4
    L = 2000
    plans = prim_plans("alltoall", world, L, chunk_bytes=512)
    user = rand_inputs(world, world * L)
    sim = Sim(world, plans, user, world * L)
    sim.run()
    for r in range(world):
        expect = np.concatenate([user[s][r * L:(r + 1) * L]
                                 for s in range(world)])
        np.testing.assert_allclose(sim.out[r], expect, rtol=1e-6, atol=1e-6)


def test_reducescatter_relay_subset():
    world = 4
    L = 256
    active = [0, 3]
    plans = prim_plans("reducescatter", world, L, active=active)
    user = rand_inputs(world, world * L)
    sim = Sim(world, plans, user, L)
    sim.run()
    total = np.sum([user[r] for r in active], axis=0)
    for r in range(world):
        np.testing.assert_allclose(sim.out[r], total[r * L:(r + 1) * L],
                                   rtol=1e-5, atol=1e-5)


def test_single_element_tensor():
    sim_allreduce(stars(4), total=1)


def test_sixteen_ranks():
    sim_allreduce(stars(16), total=2048)


def test_weighted_slices_allreduce():
    """Heterogeneity adaptation: non-uniform per-tree slice weights still
    produce exact results (boundaries aligned, all elements covered)."""
    world = 4
    parents = stars(world)
    weights = [4.0, 1.0, 2.0, 0.5]
    total = 5000
    plans = [core.compute_plan(parents, r, total, 4, 256, [], weights)
             for r in range(world)]
    user = rand_inputs(world, total)
    sim = Sim(world, plans, user, total)
    sim.run()
    expect = np.sum(user, axis=0)
    for r in range(world):
        np.testing.assert_allclose(sim.out[r], expect, rtol=1e-5, atol=1e-5)
    # slice sizes actually differ (weight 4 tree ~8x weight 0.5 tree)
    per_tree = {}
    for u in plans[0]["bcast"]:
        per_tree[u["tree"]] = per_tree.get(u["tree"], 0) + u["count"]
    assert per_tree[0] > per_tree[3] * 3, per_tree


def test_weighted_slices_extreme_and_tiny():
    world = 3
    weights = [100.0, 1.0, 1.0]
    for total in (7, 64, 700):
        plans = [core.compute_plan(stars(world), r, total, 4, 128, [], weights)
                 for r in range(world)]
        user = rand_inputs(world, total)
        sim = Sim(world, plans, user, total)
        sim.run()
        expect = np.sum(user, axis=0)
        for r in range(world):
            np.testing.assert_allclose(sim.out[r], expect, rtol=1e-5,
                                       atol=1e-5)


def test_weighted_reduce_to_root():
    world, root = 4, 2
    weights = [3.0, 1.0, 0.25, 1.5]
    total = 1200
    plans = [
        core.compute_primitive_plan("reduce", world, r, total, 4, 256,
                                    root=root, parents=stars(world),
                                    slice_weights=weights)
        for r in range(world)
    ]
    user = rand_inputs(world, total)
    sim = Sim(world, plans, user, total)
    sim.run()
    np.testing.assert_allclose(sim.out[root], np.sum(user, axis=0),
                               rtol=1e-5, atol=1e-5)


def test_partial_mask_excluded_ranks_never_aggregate_or_forward():
    """Re-rooting invariant (round-2): with a partial active set, an
    excluded rank's plan has no copy-in and no reduce units, and no
    ACTIVE rank's units name an excluded rank as a source, consumer, or
    publish target (so a wedged excluded straggler sits on nobody's
    critical path); excluded ranks may only appear as broadcast leaves
    pulling results."""
    import adapcc_amd._core as core

    world = 8
    # deep chains are the adversarial shape: the old behavior forwarded
    # broadcasts through excluded intermediates
    chains = []
    for t in range(3):
        order = [(t + i) % world for i in range(world)]
        par = [-1] * world
        for i in range(1, world):
            par[order[i]] = order[i - 1]
        chains.append(par)
    active = [0, 2, 5]
    excluded = [r for r in range(world) if r not in active]
    for rank in range(world):
        plan = core.compute_plan(chains, rank, 4096, 4, 1024, active)
        if rank in excluded:
            assert plan["copy"] == [], rank
            assert plan["reduce"] == [], rank
            # broadcast pulls must come from ACTIVE parents only
            for bu in plan["bcast"]:
                assert bu["parent"] in active, (rank, bu)
                assert not bu["forward"], (rank, bu)
        else:
            for ru in plan["reduce"]:
                for (src, _kind) in ru["srcs"]:
                    assert src in active, (rank, ru)
                assert all(k in range(world) for k in ru["publish_to"])
            for cu in plan["copy"]:
                for n in cu["notify_to"]:
                    assert n in active, (rank, cu)
