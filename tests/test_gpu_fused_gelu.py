"""Numerics for the fused tanh-GeLU (csrc/gelu.hip) vs torch fp32."""

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.mark.parametrize("dtype,tol", [
    (torch.float32, 2e-5),
    (torch.bfloat16, 2e-2),
])
@pytest.mark.parametrize("n", [128, 131072 * 3072 // 64, 999999])
def test_fused_gelu_matches_reference(dtype, tol, n):
    from adapcc_amd.ops.fused import fused_gelu

    torch.manual_seed(0)
    x = (torch.randn(n, device="cuda", dtype=dtype) * 3).requires_grad_(True)
    y = fused_gelu(x)
    xr = x.detach().float().requires_grad_(True)
    yr = torch.nn.functional.gelu(xr, approximate="tanh")
    torch.testing.assert_close(y.float(), yr, rtol=tol, atol=tol)
    g = torch.randn_like(y)
    y.backward(g)
    yr.backward(g.float())
    torch.testing.assert_close(x.grad.float(), xr.grad, rtol=tol, atol=tol)
