"""Emulated multi-rank validation of the native pull-engine on ONE GPU.

N Engine instances live in one process on device 0 and are wired with
connect_local (raw addresses instead of hipIpc), so the complete flag
protocol — copy-in notify, pull-reduce, publish, forward, end barrier,
sequence epochs — runs on real CDNA4 hardware with every memory-ordering
path except the xGMI hop itself. This is the strongest single-GPU proxy for
the 8-GPU run.
"""

import os

import pytest
import torch

pytestmark = pytest.mark.gpu

os.environ.setdefault("ADAPCC_N_GROUPS", "2")
os.environ.setdefault("ADAPCC_WGS_PER_GROUP", "2")


@pytest.fixture(scope="module")
def core():
    import adapcc_amd._core as c

    return c


def stars(n):
    return [[-1 if r == t else t for r in range(n)] for t in range(n)]


def chains(n, ntrees=2):
    out = []
    for t in range(ntrees):
        order = [(t + i) % n for i in range(n)]
        parents = [0] * n
        parents[order[0]] = -1
        for i in range(1, n):
            parents[order[i]] = order[i - 1]
        out.append(parents)
    return out


def make_world(core, world, parents, chunk_bytes=1 << 20, cap=64 << 20,
               timeout_ms=15000.0):
    engines = [core.Engine(r, world, 0, cap, timeout_ms) for r in range(world)]
    addrs = [e.region_addr() for e in engines]
    for e in engines:
        e.connect_local(addrs)
        e.set_strategy(parents, chunk_bytes)
    return engines


def run_allreduce(core, engines, tensors, active=(), average=False):
    dt = {torch.float32: core.DTYPE_F32, torch.bfloat16: core.DTYPE_BF16,
          torch.float16: core.DTYPE_F16}[tensors[0].dtype]
    op = core.OP_AVG if average else core.OP_SUM
    stream = torch.cuda.current_stream().cuda_stream
    for r, e in enumerate(engines):
        e.allreduce(tensors[r].data_ptr(), tensors[r].numel(), dt, op,
                    list(active), average, stream)
    for e in engines:
        e.synchronize()


@pytest.mark.parametrize("world", [2, 4, 8])
@pytest.mark.parametrize("count", [64, 100_003, 3_000_000])
def test_emulated_stars_sum(core, world, count):
    engines = make_world(core, world, stars(world))
    torch.manual_seed(world * 1000 + count)
    tensors = [torch.randn(count, device="cuda") for _ in range(world)]
    expect = torch.stack(tensors).sum(0)
    run_allreduce(core, engines, tensors)
    for r in range(world):
        torch.testing.assert_close(tensors[r], expect, rtol=1e-5, atol=1e-4)


def test_emulated_chains(core):
    world = 4
    engines = make_world(core, world, chains(world, 2), chunk_bytes=256 << 10)
    tensors = [torch.randn(777_777, device="cuda") for _ in range(world)]
    expect = torch.stack(tensors).sum(0)
    run_allreduce(core, engines, tensors)
    for r in range(world):
        torch.testing.assert_close(tensors[r], expect, rtol=1e-5, atol=1e-4)


def test_emulated_bf16(core):
    world = 4
    engines = make_world(core, world, stars(world))
    tensors = [torch.randn(500_000, device="cuda", dtype=torch.bfloat16)
               for _ in range(world)]
    expect = torch.stack([t.float() for t in tensors]).sum(0).bfloat16()
    run_allreduce(core, engines, tensors)
    for r in range(world):
        torch.testing.assert_close(tensors[r].float(), expect.float(),
                                   rtol=5e-2, atol=5e-2)


def test_emulated_average_and_repeat_calls(core):
    world = 4
    engines = make_world(core, world, stars(world))
    for it in range(5):  # sequence epochs / plan cache / buffer reuse
        count = [4096, 100_000, 4096, 999, 2_000_001][it]
        tensors = [torch.randn(count, device="cuda") for _ in range(world)]
        expect = torch.stack(tensors).mean(0)
        run_allreduce(core, engines, tensors, average=True)
        for r in range(world):
            torch.testing.assert_close(tensors[r], expect, rtol=1e-5, atol=1e-4)


def test_emulated_relay_subset(core):
    world = 8
    engines = make_world(core, world, stars(world))
    active = [0, 2, 3, 5, 6, 7]  # ranks 1 and 4 are straggler relays
    tensors = [torch.randn(300_000, device="cuda") for _ in range(world)]
    expect = torch.stack([tensors[r] for r in active]).sum(0)
    run_allreduce(core, engines, tensors, active=active)
    for r in range(world):
        torch.testing.assert_close(tensors[r], expect, rtol=1e-5, atol=1e-4)


def test_emulated_tiny_tensor(core):
    world = 8
    engines = make_world(core, world, stars(world))
    tensors = [torch.full((16,), float(r + 1), device="cuda")
               for r in range(world)]
    run_allreduce(core, engines, tensors)
    expect = float(sum(range(1, world + 1)))
    for r in range(world):
        torch.testing.assert_close(tensors[r],
                                   torch.full((16,), expect, device="cuda"))
