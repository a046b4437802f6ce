"""Numerics tests for the native HIP kernels against plain PyTorch fp32
references (single GPU)."""

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def core():
    import adapcc_amd._core as c

    return c


def _stream():
    return torch.cuda.current_stream().cuda_stream


@pytest.mark.parametrize("nsrc", [1, 2, 4, 8])
@pytest.mark.parametrize("count", [64, 4096, 1_000_003])
def test_local_reduce_sum_f32(core, nsrc, count):
    srcs = [torch.randn(count, device="cuda") for _ in range(nsrc)]
    dst = torch.empty(count, device="cuda")
    core.local_reduce(dst.data_ptr(), [s.data_ptr() for s in srcs], count,
                      core.DTYPE_F32, core.OP_SUM, 1.0, _stream())
    ref = torch.stack(srcs).sum(0)
    torch.testing.assert_close(dst, ref, rtol=1e-6, atol=1e-5)


@pytest.mark.parametrize("dtype,code", [
    (torch.bfloat16, "DTYPE_BF16"),
    (torch.float16, "DTYPE_F16"),
])
def test_local_reduce_sum_half(core, dtype, code):
    count = 123_457
    srcs = [torch.randn(count, device="cuda", dtype=dtype) for _ in range(5)]
    dst = torch.empty(count, device="cuda", dtype=dtype)
    core.local_reduce(dst.data_ptr(), [s.data_ptr() for s in srcs], count,
                      getattr(core, code), core.OP_SUM, 1.0, _stream())
    # fp32 accumulate in kernel; reference: accumulate fp32 then cast
    ref = torch.stack([s.float() for s in srcs]).sum(0).to(dtype)
    torch.testing.assert_close(dst, ref, rtol=1e-2, atol=1e-2)


def test_local_reduce_max(core):
    count = 77_777
    srcs = [torch.randn(count, device="cuda") for _ in range(3)]
    dst = torch.empty(count, device="cuda")
    core.local_reduce(dst.data_ptr(), [s.data_ptr() for s in srcs], count,
                      core.DTYPE_F32, core.OP_MAX, 1.0, _stream())
    ref = torch.stack(srcs).max(0).values
    torch.testing.assert_close(dst, ref)


def test_local_reduce_scale(core):
    count = 999
    srcs = [torch.randn(count, device="cuda") for _ in range(4)]
    dst = torch.empty(count, device="cuda")
    core.local_reduce(dst.data_ptr(), [s.data_ptr() for s in srcs], count,
                      core.DTYPE_F32, core.OP_SUM, 0.25, _stream())
    ref = torch.stack(srcs).sum(0) / 4
    torch.testing.assert_close(dst, ref, rtol=1e-6, atol=1e-5)


def test_engine_single_gpu_lifecycle(core):
    """World-1 engine: allocation, handle export, trivial allreduce."""
    from adapcc_amd.runtime.engine import NativeEngine
    from adapcc_amd.strategy.partrees import synthesize_stars

    eng = NativeEngine(0, 1, device=0, cap_bytes=64 << 20)
    h = eng._eng.ipc_handle()
    assert len(h) == 64
    eng.bootstrap()
    eng.set_strategy(synthesize_stars(1))
    t = torch.randn(1000, device="cuda")
    ref = t.clone()
    eng.all_reduce(t)
    eng.synchronize()
    torch.testing.assert_close(t, ref)
