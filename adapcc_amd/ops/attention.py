"""Hand-written CDNA4 flash attention (csrc/attn.hip) with autograd.

Supported fast path: bf16, head_dim 64, causal, seq len a multiple of 128,
no dropout. Anything else falls back to torch SDPA. Tensors are [B, H, S, D]
with any batch/head/row strides (rows must be contiguous and 16-B aligned),
so the usual ``.view(B,T,H,hd).transpose(1,2)`` projection views work
without a copy.
"""

from __future__ import annotations

import math

import torch

from .fused import _core

__all__ = ["flash_attention", "fa_supported"]


def _strides3(t: torch.Tensor):
    return [t.stride(0), t.stride(1), t.stride(2)]


def _row_ok(t: torch.Tensor) -> bool:
    if t.stride(3) != 1:
        return False
    if t.data_ptr() % 16 != 0:
        return False
    # every row/plane base must stay 16-B aligned (bf16: 8 elements)
    return all(s % 8 == 0 for s in _strides3(t))


def fa_supported(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                 causal: bool, dropout_p: float) -> bool:
    if not (causal and dropout_p == 0.0):
        return False
    if not (q.is_cuda and q.dtype == torch.bfloat16):
        return False
    if q.dim() != 4 or q.shape != k.shape or q.shape != v.shape:
        return False
    B, H, S, D = q.shape
    if D != 64 or S % 128 != 0 or S < 128:
        return False
    if not (_row_ok(q) and _row_ok(k) and _row_ok(v)):
        return False
    return bool(_core())


class _FlashAttnFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, scale):
        c = _core()
        B, H, S, D = q.shape
        o = torch.empty((B, H, S, D), dtype=q.dtype, device=q.device)
        lse = torch.empty((B, H, S), dtype=torch.float32, device=q.device)
        strides = _strides3(q) + _strides3(k) + _strides3(v) + _strides3(o)
        stream = torch.cuda.current_stream(q.device).cuda_stream
        c.fa_fwd(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(),
                 lse.data_ptr(), B, H, S, strides, scale, stream)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.scale = scale
        return o

    @staticmethod
    def backward(ctx, dout):
        c = _core()
        q, k, v, o, lse = ctx.saved_tensors
        B, H, S, D = q.shape
        do = dout.contiguous()
        delta = (do.float() * o.float()).sum(-1)  # [B,H,S] fp32
        dq = torch.empty((B, H, S, D), dtype=q.dtype, device=q.device)
        dk = torch.empty_like(dq)
        dv = torch.empty_like(dq)
        strides = (_strides3(q) + _strides3(k) + _strides3(v) +
                   _strides3(do) + _strides3(dq) + _strides3(dk) +
                   _strides3(dv))
        stream = torch.cuda.current_stream(q.device).cuda_stream
        c.fa_bwd(q.data_ptr(), k.data_ptr(), v.data_ptr(), do.data_ptr(),
                 lse.data_ptr(), delta.data_ptr(), dq.data_ptr(),
                 dk.data_ptr(), dv.data_ptr(), B, H, S, strides, ctx.scale,
                 stream)
        return dq, dk, dv, None


def flash_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                    causal: bool = True, dropout_p: float = 0.0,
                    scale: float = None) -> torch.Tensor:
    """SDPA drop-in for [B,H,S,D] inputs; runs the hand-written CDNA4
    kernels when the shape qualifies, torch SDPA otherwise."""
    if fa_supported(q, k, v, causal, dropout_p):
        s = scale if scale is not None else 1.0 / math.sqrt(q.shape[-1])
        return _FlashAttnFn.apply(q, k, v, s)
    return torch.nn.functional.scaled_dot_product_attention(
        q, k, v, is_causal=causal, dropout_p=dropout_p, scale=scale)
