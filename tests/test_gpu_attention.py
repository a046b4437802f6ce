"""Numerics for the hand-written CDNA4 flash attention (csrc/attn.hip)
against a plain PyTorch fp32 reference."""

import math

import pytest
import torch

pytestmark = pytest.mark.gpu


def ref_attention_f32(q, k, v, scale):
    """fp32 causal attention reference (explicit, no SDPA)."""
    qf, kf, vf = q.float(), k.float(), v.float()
    s = torch.matmul(qf, kf.transpose(-1, -2)) * scale
    S = q.shape[-2]
    mask = torch.ones(S, S, dtype=torch.bool, device=q.device).triu(1)
    s = s.masked_fill(mask, float("-inf"))
    p = torch.softmax(s, dim=-1)
    return torch.matmul(p, vf)


def test_mfma_layout_probe():
    """Verify the documented v_mfma_f32_32x32x16_bf16 fragment layouts."""
    from adapcc_amd.ops.fused import _core

    c = _core()
    torch.manual_seed(0)
    # asymmetric A and B so operand/output transposes can't pass (guide G9)
    a = torch.randn(32, 16, device="cuda").to(torch.bfloat16)
    b = torch.randn(16, 32, device="cuda").to(torch.bfloat16)
    out = torch.zeros(32, 32, dtype=torch.float32, device="cuda")
    stream = torch.cuda.current_stream().cuda_stream
    c.mfma_probe(a.data_ptr(), b.data_ptr(), out.data_ptr(), stream)
    torch.cuda.synchronize()
    ref = a.float() @ b.float()
    torch.testing.assert_close(out, ref, rtol=2e-2, atol=2e-2)


def test_tr_read_probe():
    """ds_read_b64_tr_b16 fragment reads off the swizzled row image."""
    from adapcc_amd.ops.fused import _core

    c = _core()
    stream = torch.cuda.current_stream().cuda_stream
    for kb in (0, 1):
        for db in (0, 1):
            out = torch.zeros(64 * 8, dtype=torch.float32, device="cuda")
            c.tr_probe(out.data_ptr(), kb, db, stream)
            torch.cuda.synchronize()
            o = out.view(64, 8)
            l = torch.arange(64, device="cuda")
            e = torch.arange(8, device="cuda")
            k = kb * 16 + 8 * (l >> 5).unsqueeze(1) + e.unsqueeze(0)
            d = (db * 32 + (l & 31)).unsqueeze(1).expand(64, 8)
            want = (k * 64 + d).float().to(torch.bfloat16).float()
            torch.testing.assert_close(o, want, rtol=0, atol=0)


def test_pack_frag_probe():
    """cvt_pk + permlane32_swap operand redistribution."""
    from adapcc_amd.ops.fused import _core

    c = _core()
    stream = torch.cuda.current_stream().cuda_stream
    out = torch.zeros(64 * 8, dtype=torch.float32, device="cuda")
    c.pack_probe(out.data_ptr(), stream)
    torch.cuda.synchronize()
    o = out.view(64, 8).cpu()
    crow = lambda r, hi: (r & 3) + 8 * (r >> 2) + 4 * hi
    for l in range(64):
        hi = l >> 5
        for e in range(8):
            row = 8 * hi + e
            src = None
            for h2 in range(2):
                for r in range(8):
                    if crow(r, h2) == row:
                        src = (r, h2)
            r, h2 = src
            exp = float((l & 31) + 32 * h2) * 100 + r
            expb = torch.tensor(exp).to(torch.bfloat16).float().item()
            assert o[l][e].item() == expb, (l, e, o[l][e].item(), expb)


@pytest.mark.parametrize("B,H,S", [(1, 1, 128), (2, 3, 256), (2, 12, 1024)])
def test_fa_fwd_bwd_matches_reference(B, H, S):
    from adapcc_amd.ops.attention import _FlashAttnFn, fa_supported

    torch.manual_seed(0)
    D = 64
    q = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    k = torch.randn_like(q, requires_grad=True)
    v = torch.randn_like(q, requires_grad=True)
    assert fa_supported(q, k, v, True, 0.0)
    scale = 1.0 / math.sqrt(D)

    o = _FlashAttnFn.apply(q, k, v, scale)
    ref = ref_attention_f32(q, k, v, scale)
    torch.testing.assert_close(o.float(), ref, rtol=2e-2, atol=2e-2)

    g = torch.randn_like(o)
    o.backward(g)
    qr = q.detach().float().requires_grad_(True)
    kr = k.detach().float().requires_grad_(True)
    vr = v.detach().float().requires_grad_(True)
    rr = ref_attention_f32(qr, kr, vr, scale)
    rr.backward(g.float())
    torch.testing.assert_close(q.grad.float(), qr.grad, rtol=5e-2, atol=5e-2)
    torch.testing.assert_close(k.grad.float(), kr.grad, rtol=5e-2, atol=5e-2)
    torch.testing.assert_close(v.grad.float(), vr.grad, rtol=5e-2, atol=5e-2)


def test_fa_strided_views_match_contiguous():
    """The projection-view layout ([B,T,H,hd].transpose(1,2)) must give the
    same result as contiguous [B,H,S,D] inputs."""
    from adapcc_amd.ops.attention import flash_attention

    torch.manual_seed(1)
    B, H, S, D = 2, 4, 256, 64
    qkv = torch.randn(B, S, 3 * H * D, device="cuda", dtype=torch.bfloat16)
    q, k, v = qkv.split(H * D, dim=2)
    qv = q.view(B, S, H, D).transpose(1, 2)
    kv = k.view(B, S, H, D).transpose(1, 2)
    vv = v.view(B, S, H, D).transpose(1, 2)
    out_view = flash_attention(qv, kv, vv, causal=True)
    out_cont = flash_attention(qv.contiguous(), kv.contiguous(),
                               vv.contiguous(), causal=True)
    torch.testing.assert_close(out_view, out_cont, rtol=0, atol=0)


def test_fa_extreme_values_stable():
    """Large-magnitude scores exercise the online-softmax rescale path."""
    from adapcc_amd.ops.attention import _FlashAttnFn

    torch.manual_seed(2)
    B, H, S, D = 1, 2, 256, 64
    q = (torch.randn(B, H, S, D, device="cuda") * 4).to(torch.bfloat16)
    k = (torch.randn(B, H, S, D, device="cuda") * 4).to(torch.bfloat16)
    # spike one key so later tiles force a max jump (guide rule 26)
    k[:, :, 200, :] *= 8
    v = torch.randn(B, H, S, D, device="cuda").to(torch.bfloat16)
    scale = 1.0 / math.sqrt(D)
    o = _FlashAttnFn.apply(q.requires_grad_(True), k.requires_grad_(True),
                           v.requires_grad_(True), scale)
    assert torch.isfinite(o.float()).all()
    ref = ref_attention_f32(q, k, v, scale)
    torch.testing.assert_close(o.float(), ref, rtol=3e-2, atol=3e-2)


def test_fa_fallback_non_qualifying():
    """Odd shapes route to SDPA, still correct."""
    from adapcc_amd.ops.attention import flash_attention

    q = torch.randn(1, 2, 197, 64, device="cuda", dtype=torch.bfloat16)
    k, v = torch.randn_like(q), torch.randn_like(q)
    out = flash_attention(q, k, v, causal=False)
    ref = torch.nn.functional.scaled_dot_product_attention(q, k, v)
    torch.testing.assert_close(out, ref, rtol=1e-2, atol=1e-2)
