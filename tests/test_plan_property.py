"""Property-based fuzz of the unit planner: random valid forests, random
active sets, random sizes — the simulator must complete (deadlock-freedom)
and produce exact allreduce results for every case."""

import random

import numpy as np
import pytest

core = pytest.importorskip("adapcc_amd._core")

from test_plan_sim import Sim, rand_inputs  # noqa: E402


def random_forest(rng: random.Random, world: int, ntrees: int):
    """Uniform random recursive trees over `world` ranks per tree."""
    parents = []
    for _ in range(ntrees):
        order = list(range(world))
        rng.shuffle(order)
        p = [0] * world
        p[order[0]] = -1
        for i in range(1, world):
            p[order[i]] = order[rng.randrange(i)]  # parent among earlier
        parents.append(p)
    return parents


@pytest.mark.parametrize("seed", range(24))
def test_random_forests_allreduce(seed):
    rng = random.Random(seed)
    world = rng.choice([2, 3, 4, 5, 8, 11, 16])
    ntrees = rng.randint(1, min(8, world))
    parents = random_forest(rng, world, ntrees)
    total = rng.choice([1, 7, 64, 513, 5000, 70_000])
    chunk_bytes = rng.choice([128, 256, 4096, 1 << 20])
    n_active = rng.randint(1, world)
    active = sorted(rng.sample(range(world), n_active))
    average = rng.random() < 0.5
    weights = ([rng.uniform(0.1, 5.0) for _ in range(ntrees)]
               if rng.random() < 0.4 else [])

    plans = [
        core.compute_plan(parents, r, total, 4, chunk_bytes, active, weights)
        for r in range(world)
    ]
    user = rand_inputs(world, total, seed=seed)
    scale = 1.0 / len(active) if average else 1.0
    sim = Sim(world, plans, user, total, scale)
    sim.run()
    expect = np.sum([user[r] for r in active], axis=0) * scale
    for r in range(world):
        np.testing.assert_allclose(sim.out[r], expect, rtol=1e-5, atol=1e-5)


@pytest.mark.parametrize("seed", range(10))
def test_random_deep_chains(seed):
    """Degenerate deep trees (max forwarding depth) with random relays."""
    rng = random.Random(1000 + seed)
    world = rng.choice([6, 10, 16])
    order = list(range(world))
    rng.shuffle(order)
    p = [0] * world
    p[order[0]] = -1
    for i in range(1, world):
        p[order[i]] = order[i - 1]
    active = sorted(rng.sample(range(world), rng.randint(1, world)))
    total = rng.choice([100, 4096])
    plans = [core.compute_plan([p], r, total, 4, 512, active)
             for r in range(world)]
    user = rand_inputs(world, total, seed=seed)
    sim = Sim(world, plans, user, total, 1.0)
    sim.run()
    expect = np.sum([user[r] for r in active], axis=0)
    for r in range(world):
        np.testing.assert_allclose(sim.out[r], expect, rtol=1e-5, atol=1e-5)
