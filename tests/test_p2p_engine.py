"""P2P tree-transport engine tests over gloo (CPU): the same unit plans as
the native engine executed with torch.distributed point-to-point."""

import os

import pytest
import torch

from util_mp import run_mp


def _p2p_case(rank, world, strategy_kind, count, active, average):
    os.environ["ADAPCC_TRANSPORT"] = "p2p"
    import torch.distributed as dist

    from adapcc_amd.runtime.p2p_engine import P2PTreeEngine
    from adapcc_amd.strategy.partrees import synthesize_chains, synthesize_stars

    eng = P2PTreeEngine(rank, world)
    eng.bootstrap()
    strat = (synthesize_stars(world) if strategy_kind == "stars"
             else synthesize_chains(world, num_trees=2))
    strat.chunk_bytes = 4096
    eng.set_strategy(strat)

    torch.manual_seed(100 + rank)
    t = torch.randn(count)
    g = [torch.zeros_like(t) for _ in range(world)]
    dist.all_gather(g, t)
    act = active or list(range(world))
    expect = torch.stack([g[r] for r in act]).sum(0)
    if average:
        expect = expect / len(act)

    eng.all_reduce(t, active=active or None, average=average)
    torch.testing.assert_close(t, expect, rtol=1e-5, atol=1e-5)
    return True


@pytest.mark.parametrize("world,kind", [(2, "stars"), (4, "stars"),
                                        (4, "chains")])
def test_p2p_allreduce(world, kind):
    assert all(run_mp(_p2p_case, world, backend="gloo",
                      args=(kind, 10_000, [], False), timeout=180))


def test_p2p_relay_and_average():
    assert all(run_mp(_p2p_case, 4, backend="gloo",
                      args=("stars", 5_000, [0, 2, 3], True), timeout=180))


def _p2p_facade(rank, world):
    os.environ["ADAPCC_TRANSPORT"] = "p2p"
    from adapcc_amd import AdapCC, CommArgs

    AdapCC.init(CommArgs(entry_point=-1), rank, rank, world)
    AdapCC.setup()
    assert AdapCC.communicator.effective_transport == "p2p"
    t = torch.full((512,), float(rank + 1))
    AdapCC.allreduce(t)
    expect = float(sum(range(1, world + 1)))
    assert torch.allclose(t, torch.full_like(t, expect))
    AdapCC.clear()
    return True


def test_p2p_facade():
    assert all(run_mp(_p2p_facade, 2, backend="gloo", timeout=180))
