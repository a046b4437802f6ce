"""Numerics + perf sanity for the fused CDNA4 LayerNorm vs plain PyTorch
fp32 references."""

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.mark.parametrize("dtype,tol", [
    (torch.float32, 1e-5),
    (torch.bfloat16, 3e-2),
    (torch.float16, 1e-2),
])
@pytest.mark.parametrize("rows,cols", [(128, 768), (1000, 768), (64, 128),
                                       (333, 1024), (64, 3072), (31, 4096)])
def test_fused_ln_matches_reference(dtype, tol, rows, cols):
    from adapcc_amd.ops.fused import FusedLayerNorm, ln_fusable

    assert ln_fusable(cols, dtype)
    torch.manual_seed(0)
    ln = FusedLayerNorm(cols).to("cuda", dtype)
    with torch.no_grad():
        ln.weight.normal_(1.0, 0.2)
        ln.bias.normal_(0.0, 0.2)
    ref = torch.nn.LayerNorm(cols).to("cuda", torch.float32)
    with torch.no_grad():
        ref.weight.copy_(ln.weight.float())
        ref.bias.copy_(ln.bias.float())

    x = torch.randn(rows, cols, device="cuda", dtype=dtype, requires_grad=True)
    xr = x.detach().float().clone().requires_grad_(True)

    y = ln(x)
    yr = ref(xr)
    torch.testing.assert_close(y.float(), yr, rtol=tol, atol=tol)

    g = torch.randn_like(y)
    y.backward(g)
    yr.backward(g.float())
    torch.testing.assert_close(x.grad.float(), xr.grad, rtol=tol,
                               atol=tol * 5)
    torch.testing.assert_close(ln.weight.grad.float(), ref.weight.grad,
                               rtol=tol, atol=tol * max(1.0, rows ** 0.5))
    torch.testing.assert_close(ln.bias.grad.float(), ref.bias.grad,
                               rtol=tol, atol=tol * max(1.0, rows ** 0.5))


def test_fused_ln_cpu_fallback():
    from adapcc_amd.ops.fused import FusedLayerNorm

    ln = FusedLayerNorm(64)
    x = torch.randn(4, 64, requires_grad=True)
    y = ln(x)
    y.sum().backward()
    assert x.grad is not None


def test_fused_ln_bench():
    """Wall-clock sanity: fused LN fwd+bwd should beat stock LayerNorm on
    the GPT-2 shape (not asserted hard — printed for the profile log)."""
    import time

    from adapcc_amd.ops.fused import FusedLayerNorm

    rows, cols = 65536, 768
    x = torch.randn(rows, cols, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    g = torch.randn(rows, cols, device="cuda", dtype=torch.bfloat16)

    def bench(mod):
        for _ in range(3):
            y = mod(x)
            y.backward(g)
            x.grad = None
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(10):
            y = mod(x)
            y.backward(g)
            x.grad = None
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / 10 * 1000

    fused = FusedLayerNorm(cols).to("cuda", torch.bfloat16)
    stock = torch.nn.LayerNorm(cols).to("cuda", torch.bfloat16)
    t_fused = bench(fused)
    t_stock = bench(stock)
    print(f"\nLN fwd+bwd {rows}x{cols} bf16: fused {t_fused:.3f} ms vs "
          f"stock {t_stock:.3f} ms ({t_stock / t_fused:.2f}x)")
    assert t_fused < t_stock * 1.5  # never catastrophically slower
