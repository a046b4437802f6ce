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


def _p2p_prims(rank, world):
    os.environ["ADAPCC_TRANSPORT"] = "p2p"
    import torch.distributed as dist

    from adapcc_amd.runtime.p2p_engine import P2PTreeEngine
    from adapcc_amd.strategy.partrees import synthesize_stars

    eng = P2PTreeEngine(rank, world)
    eng.bootstrap()
    strat = synthesize_stars(world)
    strat.chunk_bytes = 2048
    eng.set_strategy(strat)
    L = 3000

    def gath(t):
        g = [torch.zeros_like(t) for _ in range(world)]
        dist.all_gather(g, t)
        return g

    # reduce to root
    torch.manual_seed(1 + rank)
    t = torch.randn(L)
    g = gath(t)
    eng.reduce(t, root=world - 1)
    if rank == world - 1:
        torch.testing.assert_close(t, torch.stack(g).sum(0), rtol=1e-5,
                                   atol=1e-5)

    # broadcast
    torch.manual_seed(2 + rank)
    t = torch.randn(L)
    g = gath(t)
    eng.broadcast(t, root=0)
    torch.testing.assert_close(t, g[0])

    # allgather
    torch.manual_seed(3 + rank)
    t = torch.randn(L)
    g = gath(t)
    out = torch.zeros(world * L)
    eng.all_gather(out, t)
    torch.testing.assert_close(out, torch.cat(g))

    # reduce_scatter
    torch.manual_seed(4 + rank)
    t = torch.randn(world * L)
    g = gath(t)
    out = torch.zeros(L)
    eng.reduce_scatter(out, t)
    total = torch.stack(g).sum(0)
    torch.testing.assert_close(out, total[rank * L:(rank + 1) * L],
                               rtol=1e-5, atol=1e-5)

    # alltoall
    torch.manual_seed(5 + rank)
    t = torch.randn(world * L)
    g = gath(t)
    out = torch.zeros(world * L)
    eng.all_to_all(out, t)
    expect = torch.cat([g[s][rank * L:(rank + 1) * L] for s in range(world)])
    torch.testing.assert_close(out, expect)
    return True


@pytest.mark.parametrize("world", [2, 3, 4])
def test_p2p_other_primitives(world):
    assert all(run_mp(_p2p_prims, world, backend="gloo", timeout=240))
