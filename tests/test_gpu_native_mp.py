"""Native-engine validation with N PROCESSES sharing one GPU.

Each rank is a real process with its own HIP context; buffers are wired via
real hipIpc handle exchange (dmabuf mode) over a gloo bootstrap — exactly
the production path, except every rank maps to device 0, so peer pulls stay
on-die instead of crossing xGMI. The GPU timeslices the processes' queues,
so spinning kernels make cross-process progress just as on 8 GPUs.

(A single-process emulation with connect_local cannot work: HIP multiplexes
 all of one process's streams onto <=4 hardware queues and a spinning kernel
 blocks everything mapped behind it.)
"""

import os

import pytest
import torch

from util_mp import run_mp

pytestmark = pytest.mark.gpu


def _world_case(rank, world, count, active, average, dtype_str, rounds):
    os.environ["ADAPCC_TIMEOUT_MS"] = "20000"
    import torch
    import torch.distributed as dist

    torch.cuda.set_device(0)
    from adapcc_amd.runtime.engine import NativeEngine
    from adapcc_amd.strategy.partrees import synthesize_stars

    dtype = getattr(torch, dtype_str)
    eng = NativeEngine(rank, world, device=0, cap_bytes=64 << 20)
    eng.bootstrap()
    eng.set_strategy(synthesize_stars(world))

    for rnd in range(rounds):
        n = count // (rnd + 1) + rnd * 13
        torch.manual_seed(1000 * rnd + rank)
        t = torch.randn(n, device="cuda", dtype=dtype)
        # reference via gloo on CPU
        cpu = t.float().cpu()
        gathered = [torch.zeros_like(cpu) for _ in range(world)]
        dist.all_gather(gathered, cpu)
        act = active if active else list(range(world))
        expect = torch.stack([gathered[r] for r in act]).sum(0)
        if average:
            expect = expect / len(act)

        eng.all_reduce(t, active=active or None, average=average)
        eng.synchronize()
        got = t.float().cpu()
        tol = 1e-4 if dtype_str == "float32" else 0.1
        torch.testing.assert_close(got, expect, rtol=tol, atol=tol)
    return True


@pytest.mark.parametrize("world", [2, 4])
def test_native_allreduce_f32(world):
    assert all(run_mp(_world_case, world, backend="gloo",
                      args=(1_000_000, [], False, "float32", 3),
                      timeout=300))


def test_native_allreduce_8rank():
    assert all(run_mp(_world_case, 8, backend="gloo",
                      args=(300_000, [], False, "float32", 2),
                      timeout=300))


def test_native_allreduce_bf16():
    assert all(run_mp(_world_case, 4, backend="gloo",
                      args=(500_000, [], False, "bfloat16", 2),
                      timeout=300))


def test_native_relay_subset():
    assert all(run_mp(_world_case, 4, backend="gloo",
                      args=(200_000, [0, 2, 3], True, "float32", 2),
                      timeout=300))


def test_native_average():
    assert all(run_mp(_world_case, 2, backend="gloo",
                      args=(123_457, [], True, "float32", 2),
                      timeout=300))
