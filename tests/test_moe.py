"""Expert-parallel MoE tests (CPU/gloo; GPU path covered by -m gpu MP
tests of the underlying all_to_all)."""

import os

import pytest
import torch

from util_mp import run_mp

from adapcc_amd.models.moe import MoEMLP, MoETransformerBlock


def test_moe_single_rank_forward_backward():
    torch.manual_seed(0)
    m = MoEMLP(d_model=32, d_hidden=64, num_local_experts=4, world_size=1)
    x = torch.randn(6, 10, 32, requires_grad=True)
    y = m(x)
    assert y.shape == x.shape
    y.sum().backward()
    assert x.grad is not None
    assert m.gate.weight.grad is not None
    assert m.experts[0].w1.weight.grad is not None


def test_moe_capacity_drops_overflow():
    torch.manual_seed(0)
    m = MoEMLP(d_model=8, d_hidden=16, num_local_experts=1, world_size=1,
               capacity_factor=0.25)
    # force all tokens to expert 0: overflow tokens get zero output
    with torch.no_grad():
        m.gate.weight.zero_()
        m.gate.weight[0, 0] = 10.0
    x = torch.randn(1, 16, 8)
    x[..., 0] = 5.0
    y = m(x)
    # capacity = ceil(16 * 0.25) = 4 -> exactly 4 tokens non-zero
    nonzero_tokens = (y.abs().sum(-1) > 1e-6).sum()
    assert nonzero_tokens == 4, nonzero_tokens


def _moe_ep_matches_local(rank, world):
    os.environ["ADAPCC_TRANSPORT"] = "pg"
    import torch.distributed as dist

    from adapcc_amd import AdapCC, CommArgs

    AdapCC.init(CommArgs(entry_point=-1), rank, rank, world)
    AdapCC.setup()
    comm = AdapCC.communicator

    d, local_e = 16, 2
    torch.manual_seed(123 + rank)  # per-rank expert weights differ
    dist_moe = MoEMLP(d_model=d, d_hidden=32, num_local_experts=local_e,
                      comm=comm, world_size=world, rank=rank,
                      capacity_factor=4.0)
    # same gate everywhere (DDP would sync it)
    torch.manual_seed(55)
    with torch.no_grad():
        gate_w = torch.randn_like(dist_moe.gate.weight)
        dist_moe.gate.weight.copy_(gate_w)

    # build the equivalent single-process MoE holding ALL experts in
    # rank-major order with identical weights
    states = [None] * world
    dist.all_gather_object(
        states, {k: v.clone() for k, v in dist_moe.experts.state_dict().items()})
    local = MoEMLP(d_model=d, d_hidden=32, num_local_experts=world * local_e,
                   world_size=1, capacity_factor=4.0)
    with torch.no_grad():
        local.gate.weight.copy_(gate_w)
        for r in range(world):
            for i in range(local_e):
                gi = r * local_e + i
                local.experts[gi].w1.weight.copy_(states[r][f"{i}.w1.weight"])
                local.experts[gi].w1.bias.copy_(states[r][f"{i}.w1.bias"])
                local.experts[gi].w2.weight.copy_(states[r][f"{i}.w2.weight"])
                local.experts[gi].w2.bias.copy_(states[r][f"{i}.w2.bias"])

    torch.manual_seed(777 + rank)  # per-rank batch
    x = torch.randn(4, 8, d)
    y_dist = dist_moe(x)
    y_local = local(x)
    # NOTE: capacity in the EP run is computed over the LOCAL token count
    # on both paths (same T), and per-expert overflow order is token order,
    # identical in both implementations only when each rank's tokens fill
    # experts independently -- with capacity_factor=4 nothing overflows.
    assert torch.allclose(y_dist, y_local, atol=1e-5), (
        (y_dist - y_local).abs().max())
    y_dist.sum().backward()
    AdapCC.clear()
    return True


def test_moe_ep_matches_local_world2():
    assert all(run_mp(_moe_ep_matches_local, 2, backend="gloo", timeout=180))


def test_moe_transformer_block():
    torch.manual_seed(0)
    blk = MoETransformerBlock(d_model=32, n_head=4, d_hidden=64,
                              num_local_experts=2, world_size=1)
    x = torch.randn(2, 12, 32)
    y = blk(x)
    assert y.shape == x.shape
    y.mean().backward()
