"""Multi-rank protocol simulation over the native engine's unit plans.

Drives the exact C++ plan/relay logic (adapcc_amd._core.compute_plan — the
same code path Engine::get_plan uses on GPU) through a discrete-event
simulation with numpy buffers: units execute only when their flag waits are
satisfied, in worst-case serialized order (one group), so completion proves
both numerical correctness and deadlock-freedom of the flag protocol for
the given strategy/active-set.
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


def simulate(parents, total, active=None, chunk_bytes=256, average=False):
    """Returns (out_per_rank, n_iterations). Raises on deadlock."""
    world = len(parents[0])
    if active is None:
        active = list(range(world))
    plans = [
        core.compute_plan(parents, r, total, 4, chunk_bytes, active)
        for r in range(world)
    ]
    rng = np.random.default_rng(0)
    user = [rng.standard_normal(total).astype(np.float32) for _ in range(world)]
    out = [np.zeros(total, dtype=np.float32) for _ in range(world)]
    send = [np.zeros(total, dtype=np.float32) for _ in range(world)]
    acc = [np.zeros(total, dtype=np.float32) for _ in range(world)]
    result = [np.zeros(total, dtype=np.float32) for _ in range(world)]
    ready = set()   # (dst_rank, src_rank, tree, chunk)
    bcast = set()   # (dst_rank, tree, chunk)

    scale = 1.0 / len(active) if average else 1.0

    def buf(kind, r):
        return {SEND: send, ACC: acc, RESULT: result}[kind][r]

    # per-rank sequential queues: copy -> reduce stream; bcast stream
    queues = []
    for r in range(world):
        p = plans[r]
        queues.append({
            "red": list(p["copy"]) + list(p["reduce"]),  # same stream order
            "red_kind": ["copy"] * len(p["copy"]) + ["reduce"] * len(p["reduce"]),
            "red_i": 0,
            "bc": list(p["bcast"]),
            "bc_i": 0,
        })

    def try_red(r):
        q = queues[r]
        i = q["red_i"]
        if i >= len(q["red"]):
            return False
        u, kind = q["red"][i], q["red_kind"][i]
        t, c = u["tree"], u["chunk"]
        off, cnt = u["offset"], u["count"]
        if kind == "copy":
            send[r][off:off + cnt] = user[r][off:off + cnt]
            if u["notify"]:
                ready.add((u["consumer"], r, t, c))
        else:
            for (sr, sk) in u["srcs"]:
                if sr != r and (r, sr, t, c) not in ready:
                    return False
            pieces = [buf(sk, sr)[off:off + cnt] for (sr, sk) in u["srcs"]]
            if u["include_self"]:
                pieces.append(send[r][off:off + cnt])
            assert pieces, "reduce unit with no sources"
            acc[r][off:off + cnt] = np.sum(pieces, axis=0)
            if u["notify"]:
                ready.add((u["consumer"], r, t, c))
            if u["is_root"]:
                for k in u["publish_to"]:
                    bcast.add((k, t, c))
        q["red_i"] += 1
        return True

    def try_bc(r):
        q = queues[r]
        i = q["bc_i"]
        if i >= len(q["bc"]):
            return False
        u = q["bc"][i]
        t, c = u["tree"], u["chunk"]
        off, cnt = u["offset"], u["count"]
        if (r, t, c) not in bcast:
            return False
        src_rank = r if u["parent"] < 0 else u["parent"]
        src = buf(u["parent_kind"], src_rank)[off:off + cnt]
        out[r][off:off + cnt] = src * scale
        if u["forward"]:
            result[r][off:off + cnt] = src
            for k in u["publish_to"]:
                bcast.add((k, t, c))
        q["bc_i"] += 1
        return True

    iters = 0
    while True:
        progress = False
        for r in range(world):
            while try_red(r):
                progress = True
            while try_bc(r):
                progress = True
        iters += 1
        done = all(
            q["red_i"] == len(q["red"]) and q["bc_i"] == len(q["bc"])
            for q in queues
        )
        if done:
            break
        if not progress:
            raise AssertionError("protocol deadlock: no runnable unit")

    expect = np.sum([user[r] for r in active], axis=0) * scale
    for r in range(world):
        np.testing.assert_allclose(out[r], expect, rtol=1e-5, atol=1e-5)
    return out, iters


@pytest.mark.parametrize("world", [2, 3, 4, 8])
def test_stars_allreduce(world):
    simulate(stars(world), total=1000)


@pytest.mark.parametrize("world", [2, 4, 8])
def test_chains_allreduce(world):
    simulate(chains(world, ntrees=2), total=777)


@pytest.mark.parametrize("world", [4, 8])
def test_binary_trees_allreduce(world):
    simulate(binary_trees(world, ntrees=3), total=513)


def test_single_tree():
    simulate([[1, -1, 1, 1]], total=300)


def test_average():
    simulate(stars(4), total=256, average=True)


def test_tiny_tensor_smaller_than_slices():
    # 10 elements, 8 trees: most trees get empty slices
    simulate(stars(8), total=10)


@pytest.mark.parametrize("inactive", [[0], [3], [1, 2]])
def test_relay_star_inactive(inactive):
    world = 4
    active = [r for r in range(world) if r not in inactive]
    simulate(stars(world), total=512, active=active)


def test_relay_chain_passthrough():
    # chain 0<-1<-2<-3 with 1 inactive: reducer 0 must pull 2's subtree
    # result directly (passthrough skip), totals still correct
    world = 4
    parents = [[-1, 0, 1, 2]]
    simulate(parents, total=512, active=[0, 2, 3])


def test_relay_inactive_aggregator():
    # binary tree root 0 inactive with two subtrees: still aggregates
    parents = [[-1, 0, 0, 1, 1, 2, 2, 3]]
    simulate(parents, total=640, active=[1, 2, 3, 4, 5, 6, 7])


def test_relay_only_one_active():
    simulate(stars(4), total=128, active=[2])


def test_plan_deterministic_across_ranks():
    parents = stars(4)
    # all ranks must agree on the chunk grid
    plans = [core.compute_plan(parents, r, 5000, 4, 256) for r in range(4)]
    grids = [
        sorted((u["tree"], u["chunk"], u["offset"], u["count"]) for u in p["bcast"])
        for p in plans
    ]
    assert all(g == grids[0] for g in grids)


def test_malformed_strategy_raises():
    with pytest.raises(Exception):
        core.compute_plan([[0, 1, 2, 3]], 0, 100, 4, 256)  # cycle, no root
    with pytest.raises(Exception):
        core.compute_plan([[-1, -1, 0, 0]], 0, 100, 4, 256)  # two roots
