"""Expert-parallel MoE over the adapcc all-to-all (reference:
models/moe/train_moe.py used fastmoe's FMoETransformerMLP(num_expert=10,
d_model=1024, d_hidden=4096, top_k=1) with its NCCL all-to-all; here the
token dispatch/combine rides the adapcc engine so EP has no external
dependency — SURVEY.md §2.4 called this out as the gap to close).

Capacity-based top-1 routing (GShard-style): per expert, up to
``capacity = ceil(tokens/num_experts * capacity_factor)`` tokens are
dispatched, the rest fall through on the residual path. Equal-size buffers
make the exchange a single equal-split all-to-all in each direction.
"""

from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops.fused import FusedLayerNorm


class _AllToAll(torch.autograd.Function):
    """Differentiable equal-split all-to-all; the backward pass is the
    reverse all-to-all of the gradients."""

    @staticmethod
    def forward(ctx, comm, x: torch.Tensor) -> torch.Tensor:
        ctx.comm = comm
        if comm is None or comm.world_size == 1:
            return x.clone()
        out = torch.empty_like(x)
        comm.all_to_all(out.view(-1), x.contiguous().view(-1))
        return out

    @staticmethod
    def backward(ctx, grad: torch.Tensor):
        comm = ctx.comm
        if comm is None or comm.world_size == 1:
            return None, grad
        out = torch.empty_like(grad)
        comm.all_to_all(out.view(-1), grad.contiguous().view(-1))
        return None, out


class Expert(nn.Module):
    def __init__(self, d_model: int, d_hidden: int):
        super().__init__()
        self.w1 = nn.Linear(d_model, d_hidden)
        self.w2 = nn.Linear(d_hidden, d_model)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.w2(F.gelu(self.w1(x)))


class MoEMLP(nn.Module):
    """Distributed MoE MLP: ``num_local_experts`` experts per rank,
    ``world_size * num_local_experts`` experts total, top-1 gate."""

    def __init__(
        self,
        d_model: int = 1024,
        d_hidden: int = 4096,
        num_local_experts: int = 2,
        comm=None,
        world_size: int = 1,
        rank: int = 0,
        capacity_factor: float = 1.25,
    ):
        super().__init__()
        self.comm = comm
        self.world_size = world_size
        self.rank = rank
        self.num_local = num_local_experts
        self.num_experts = world_size * num_local_experts
        self.capacity_factor = capacity_factor
        self.gate = nn.Linear(d_model, self.num_experts, bias=False)
        self.experts = nn.ModuleList(
            Expert(d_model, d_hidden) for _ in range(num_local_experts)
        )
        self.d_model = d_model

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        orig_shape = x.shape
        x = x.reshape(-1, self.d_model)
        T = x.shape[0]
        E = self.num_experts
        cap = max(1, math.ceil(T / E * self.capacity_factor))

        logits = self.gate(x)
        probs = logits.softmax(dim=-1)
        gate_p, expert_idx = probs.max(dim=-1)  # top-1

        # position of each token within its expert's queue; drop overflow
        onehot = F.one_hot(expert_idx, E)
        pos = (onehot.cumsum(dim=0) - 1).gather(1, expert_idx[:, None]).squeeze(1)
        keep = pos < cap

        # dispatch buffer [E, cap, d]
        dispatch = x.new_zeros(E * cap, self.d_model)
        slot = expert_idx * cap + pos
        kept_slot = slot[keep]
        dispatch.index_copy_(0, kept_slot, x[keep])

        # exchange: rank r sends [experts of rank d] to d.
        # layout [world, local_E*cap, d] -> all_to_all -> received tokens for
        # MY experts from every source rank: [world, local_E, cap, d]
        dispatch = dispatch.view(self.world_size, self.num_local * cap,
                                 self.d_model)
        recv = _AllToAll.apply(self.comm, dispatch)
        recv = recv.view(self.world_size, self.num_local, cap, self.d_model)

        # run local experts on [world*cap] tokens each
        outs = []
        for i, expert in enumerate(self.experts):
            outs.append(expert(recv[:, i].reshape(-1, self.d_model)))
        out = torch.stack(outs, dim=1)  # [world*cap? ...]
        out = out.view(self.world_size, cap, self.num_local, self.d_model)
        out = out.permute(0, 2, 1, 3).contiguous()  # [world, local, cap, d]

        # return to sources
        back = _AllToAll.apply(
            self.comm, out.view(self.world_size, self.num_local * cap,
                                self.d_model))
        back = back.reshape(E * cap, self.d_model)

        # combine: kept tokens weighted by gate prob; dropped -> residual 0
        y = x.new_zeros(T, self.d_model)
        y[keep] = back.index_select(0, kept_slot) * gate_p[keep, None]
        return y.view(orig_shape)


class MoETransformerBlock(nn.Module):
    """Pre-norm block with MoE MLP (the reference's FMoETransformerMLP
    stand-in for workload tests)."""

    def __init__(self, d_model: int = 1024, n_head: int = 8, **moe_kw):
        super().__init__()
        self.ln1 = FusedLayerNorm(d_model)
        self.attn = nn.MultiheadAttention(d_model, n_head, batch_first=True)
        self.ln2 = FusedLayerNorm(d_model)
        self.moe = MoEMLP(d_model=d_model, **moe_kw)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        h = self.ln1(x)
        a, _ = self.attn(h, h, h, need_weights=False)
        x = x + a
        x = x + self.moe(self.ln2(x))
        return x
