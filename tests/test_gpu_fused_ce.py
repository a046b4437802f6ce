"""Numerics for the fused cross-entropy (csrc/ce.hip) vs plain PyTorch
fp32 cross_entropy."""

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.mark.parametrize("dtype,tol", [
    (torch.float32, 1e-5),
    (torch.bfloat16, 2e-2),
    (torch.float16, 1e-2),
])
@pytest.mark.parametrize("rows,cols", [
    (128, 50257),   # GPT-2 vocab, odd width (misaligned bf16 rows)
    (64, 1024),
    (37, 4099),     # odd everything
    (256, 32000),
])
def test_fused_ce_matches_reference(dtype, tol, rows, cols):
    from adapcc_amd.ops.fused import fused_cross_entropy

    torch.manual_seed(0)
    logits = (torch.randn(rows, cols, device="cuda", dtype=dtype) * 3.0
              ).requires_grad_(True)
    targets = torch.randint(0, cols, (rows,), device="cuda")

    loss = fused_cross_entropy(logits, targets)
    ref_logits = logits.detach().float().clone().requires_grad_(True)
    ref = torch.nn.functional.cross_entropy(ref_logits, targets)
    torch.testing.assert_close(loss, ref, rtol=tol, atol=tol)

    g = torch.tensor(1.7, device="cuda")
    loss.backward(g)
    ref.backward(g)
    torch.testing.assert_close(logits.grad.float(), ref_logits.grad,
                               rtol=tol, atol=tol * 0.1)


def test_fused_ce_ignore_index():
    from adapcc_amd.ops.fused import fused_cross_entropy

    torch.manual_seed(1)
    rows, cols = 64, 5003
    logits = torch.randn(rows, cols, device="cuda",
                         dtype=torch.float32).requires_grad_(True)
    targets = torch.randint(0, cols, (rows,), device="cuda")
    targets[::4] = -100  # a quarter ignored

    loss = fused_cross_entropy(logits, targets, ignore_index=-100)
    ref_logits = logits.detach().clone().requires_grad_(True)
    ref = torch.nn.functional.cross_entropy(ref_logits, targets,
                                            ignore_index=-100)
    torch.testing.assert_close(loss, ref, rtol=1e-5, atol=1e-5)

    loss.backward()
    ref.backward()
    torch.testing.assert_close(logits.grad, ref_logits.grad,
                               rtol=1e-5, atol=1e-6)
    # ignored rows produce exactly zero gradient
    assert logits.grad[::4].abs().max().item() == 0.0


def test_fused_ce_native_kernel_is_used():
    """The GPU path must run the HIP kernel, not silently fall back."""
    from adapcc_amd.ops import fused

    assert fused._core(), "native _core.so missing on a GPU box"
    called = {}
    orig = fused._FusedCrossEntropyFn.forward

    def spy(ctx, *a, **k):
        called["yes"] = True
        return orig(ctx, *a, **k)

    try:
        fused._FusedCrossEntropyFn.forward = staticmethod(spy)
        logits = torch.randn(8, 1000, device="cuda", dtype=torch.bfloat16,
                             requires_grad=True)
        targets = torch.randint(0, 1000, (8,), device="cuda")
        fused.fused_cross_entropy(logits, targets)
    finally:
        fused._FusedCrossEntropyFn.forward = orig
    assert called.get("yes")
