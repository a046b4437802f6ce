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


def _same_tensor_repeat(rank, world):
    """Back-to-back calls on the SAME tensor: the per-range causality of
    the caller fence must order copy-in(k+1) after bcast-write(k)."""
    os.environ["ADAPCC_TIMEOUT_MS"] = "20000"
    import torch

    torch.cuda.set_device(0)
    from adapcc_amd.runtime.engine import NativeEngine
    from adapcc_amd.strategy.partrees import synthesize_stars

    eng = NativeEngine(rank, world, device=0, cap_bytes=16 << 20)
    eng.bootstrap()
    eng.set_strategy(synthesize_stars(world))
    t = torch.full((500_000,), float(rank + 1), device="cuda")
    expect = sum(range(1, world + 1)) / world
    for _ in range(6):
        eng.all_reduce(t, average=True)
    eng.synchronize()
    # after the first average all ranks hold the mean; averaging the mean
    # is idempotent, so any ordering bug shows up as drift
    assert torch.allclose(t, torch.full_like(t, expect)), t[:3]
    return True


def _weighted_slices(rank, world):
    """Heterogeneity-adapted (non-uniform) slice weights on the native
    engine: same numerics, different per-tree work split."""
    os.environ["ADAPCC_TIMEOUT_MS"] = "20000"
    import torch
    import torch.distributed as dist

    torch.cuda.set_device(0)
    from adapcc_amd.runtime.engine import NativeEngine
    from adapcc_amd.strategy.partrees import synthesize_stars

    eng = NativeEngine(rank, world, device=0, cap_bytes=16 << 20)
    eng.bootstrap()
    strat = synthesize_stars(world)
    strat.slice_weights = [1.0 + 2.0 * t for t in range(world)]
    eng.set_strategy(strat)
    torch.manual_seed(rank)
    t = torch.randn(777_777, device="cuda")
    cpu = t.cpu()
    g = [torch.zeros_like(cpu) for _ in range(world)]
    dist.all_gather(g, cpu)
    eng.all_reduce(t)
    eng.synchronize()
    torch.testing.assert_close(t.cpu(), torch.stack(g).sum(0), rtol=1e-4,
                               atol=1e-4)
    return True


def test_weighted_slices_native():
    assert all(run_mp(_weighted_slices, 4, backend="gloo", timeout=180))


def test_same_tensor_repeat():
    assert all(run_mp(_same_tensor_repeat, 2, backend="gloo", timeout=180))


def test_native_average():
    assert all(run_mp(_world_case, 2, backend="gloo",
                      args=(123_457, [], True, "float32", 2),
                      timeout=300))


def _prims_case(rank, world):
    os.environ["ADAPCC_TIMEOUT_MS"] = "20000"
    import torch
    import torch.distributed as dist

    torch.cuda.set_device(0)
    from adapcc_amd.runtime.engine import NativeEngine
    from adapcc_amd.strategy.partrees import synthesize_stars

    eng = NativeEngine(rank, world, device=0, cap_bytes=64 << 20)
    eng.bootstrap()
    eng.set_strategy(synthesize_stars(world))

    L = 100_000

    def ref_gather(t):
        cpu = t.float().cpu()
        g = [torch.zeros_like(cpu) for _ in range(world)]
        dist.all_gather(g, cpu)
        return g

    # reduce to root 1
    torch.manual_seed(10 + rank)
    t = torch.randn(L, device="cuda")
    g = ref_gather(t)
    eng.reduce(t, root=1 % world)
    eng.synchronize()
    if rank == 1 % world:
        torch.testing.assert_close(t.cpu(), torch.stack(g).sum(0),
                                   rtol=1e-4, atol=1e-4)

    # broadcast from root 0
    torch.manual_seed(20 + rank)
    t = torch.randn(L, device="cuda")
    g = ref_gather(t)
    eng.broadcast(t, root=0)
    eng.synchronize()
    torch.testing.assert_close(t.cpu(), g[0], rtol=1e-6, atol=1e-6)

    # allgather
    torch.manual_seed(30 + rank)
    t = torch.randn(L, device="cuda")
    g = ref_gather(t)
    out = torch.zeros(world * L, device="cuda")
    eng.all_gather(out, t)
    eng.synchronize()
    torch.testing.assert_close(out.cpu(), torch.cat(g), rtol=1e-6, atol=1e-6)

    # reduce_scatter
    torch.manual_seed(40 + rank)
    t = torch.randn(world * L, device="cuda")
    g = ref_gather(t)
    out = torch.zeros(L, device="cuda")
    eng.reduce_scatter(out, t)
    eng.synchronize()
    total = torch.stack(g).sum(0)
    torch.testing.assert_close(out.cpu(), total[rank * L:(rank + 1) * L],
                               rtol=1e-4, atol=1e-4)

    # alltoall
    torch.manual_seed(50 + rank)
    t = torch.randn(world * L, device="cuda")
    g = ref_gather(t)
    out = torch.zeros(world * L, device="cuda")
    eng.all_to_all(out, t)
    eng.synchronize()
    expect = torch.cat([g[s][rank * L:(rank + 1) * L] for s in range(world)])
    torch.testing.assert_close(out.cpu(), expect, rtol=1e-6, atol=1e-6)
    return True


@pytest.mark.parametrize("world", [2, 4])
def test_native_other_primitives(world):
    assert all(run_mp(_prims_case, world, backend="gloo", timeout=300))


def _ddp_hook_gpu(rank, world):
    os.environ["ADAPCC_TIMEOUT_MS"] = "20000"
    os.environ["ADAPCC_TRANSPORT"] = "native"
    import torch
    import torch.nn as nn
    from torch.nn.parallel import DistributedDataParallel as DDP

    torch.cuda.set_device(0)
    from adapcc_amd import AdapCC, CommArgs
    from adapcc_amd.runtime.hook import AdapccDDPState, adapcc_allreduce_hook

    AdapCC.init(CommArgs(entry_point=-1), 0, rank, world)
    AdapCC.setup()

    def make_model():
        torch.manual_seed(7)
        return nn.Sequential(nn.Linear(64, 256), nn.GELU(),
                             nn.Linear(256, 32)).cuda()

    def data():
        torch.manual_seed(900 + rank)
        return (torch.randn(16, 64, device="cuda"),
                torch.randn(16, 32, device="cuda"))

    m1 = DDP(make_model(), bucket_cap_mb=1)
    state = AdapccDDPState(AdapCC.communicator)
    m1.register_comm_hook(state, adapcc_allreduce_hook)
    x, y = data()
    state.on_step(0)
    ((m1(x) - y) ** 2).mean().backward()
    torch.cuda.synchronize()
    g1 = torch.cat([p.grad.flatten() for p in m1.parameters()])

    m2 = DDP(make_model(), bucket_cap_mb=1)
    ((m2(x) - y) ** 2).mean().backward()
    torch.cuda.synchronize()
    g2 = torch.cat([p.grad.flatten() for p in m2.parameters()])

    assert torch.allclose(g1, g2, atol=1e-5), (g1 - g2).abs().max().item()
    AdapCC.clear()
    return True


def test_ddp_hook_native_gpu():
    assert all(run_mp(_ddp_hook_gpu, 2, backend="gloo", timeout=300))


def _timeout_raises(rank, world):
    """A wedged peer must surface as a Python exception within the
    deadline, never a hung GPU: rank 1 simply never joins the collective."""
    os.environ["ADAPCC_TIMEOUT_MS"] = "2500"
    import torch
    import torch.distributed as dist

    torch.cuda.set_device(0)
    from adapcc_amd.runtime.engine import NativeEngine
    from adapcc_amd.strategy.partrees import synthesize_stars

    eng = NativeEngine(rank, world, device=0, cap_bytes=16 << 20)
    eng.bootstrap()
    eng.set_strategy(synthesize_stars(world))
    raised = False
    if rank == 0:
        t = torch.ones(4096, device="cuda")
        eng.all_reduce(t)
        try:
            eng.synchronize()
        except RuntimeError as e:
            raised = "timeout" in str(e) or "error" in str(e)
    # rank 1 stays alive (its memory must remain mapped) until rank 0 is
    # done; then both exit
    dist.barrier()
    return raised if rank == 0 else True


def test_timeout_surfaces_as_exception():
    res = run_mp(_timeout_raises, 2, backend="gloo", timeout=180)
    assert res[0] is True


def _strategy_switch(rank, world):
    """reconstruct_topology path: re-set strategy on a live engine and keep
    reducing correctly (plan cache invalidation + epoch continuity)."""
    os.environ["ADAPCC_TIMEOUT_MS"] = "20000"
    import torch

    torch.cuda.set_device(0)
    from adapcc_amd.runtime.engine import NativeEngine
    from adapcc_amd.strategy.partrees import synthesize_chains, synthesize_stars

    eng = NativeEngine(rank, world, device=0, cap_bytes=16 << 20)
    eng.bootstrap()
    for strat in (synthesize_stars(world),
                  synthesize_chains(world, num_trees=2),
                  synthesize_stars(world)):
        eng.set_strategy(strat)
        t = torch.full((10_000,), float(rank + 1), device="cuda")
        eng.all_reduce(t)
        eng.synchronize()
        expect = float(sum(range(1, world + 1)))
        assert torch.allclose(t, torch.full_like(t, expect)), t[:3]
    return True


def test_strategy_switch_live():
    assert all(run_mp(_strategy_switch, 2, backend="gloo", timeout=180))


def _engine_recreate(rank, world):
    """reconstruct_topology's engine lifecycle: tear the engine down (hipIpc
    close + free) and bring a fresh one up in the same processes."""
    os.environ["ADAPCC_TIMEOUT_MS"] = "20000"
    import torch
    import torch.distributed as dist

    torch.cuda.set_device(0)
    from adapcc_amd.runtime.engine import NativeEngine
    from adapcc_amd.strategy.partrees import synthesize_stars

    for round_ in range(2):
        eng = NativeEngine(rank, world, device=0, cap_bytes=16 << 20)
        eng.bootstrap()
        eng.set_strategy(synthesize_stars(world))
        t = torch.full((8192,), float(rank + 1 + round_), device="cuda")
        eng.all_reduce(t)
        eng.synchronize()
        expect = float(sum(r + 1 + round_ for r in range(world)))
        assert torch.allclose(t, torch.full_like(t, expect))
        dist.barrier()  # nobody frees while a peer might still pull
        del eng
        dist.barrier()
    return True


def test_engine_recreate():
    assert all(run_mp(_engine_recreate, 2, backend="gloo", timeout=180))


def _moe_alltoall_gpu(rank, world):
    os.environ["ADAPCC_TIMEOUT_MS"] = "20000"
    os.environ["ADAPCC_TRANSPORT"] = "native"
    import torch

    torch.cuda.set_device(0)
    from adapcc_amd import AdapCC, CommArgs
    from adapcc_amd.models.moe import MoEMLP

    AdapCC.init(CommArgs(entry_point=-1), 0, rank, world)
    AdapCC.setup()
    comm = AdapCC.communicator

    torch.manual_seed(123 + rank)
    moe = MoEMLP(d_model=64, d_hidden=128, num_local_experts=2, comm=comm,
                 world_size=world, rank=rank, capacity_factor=4.0).cuda()
    x = torch.randn(4, 32, 64, device="cuda", requires_grad=True)
    y = moe(x)
    y.square().mean().backward()
    torch.cuda.synchronize()
    assert x.grad is not None and torch.isfinite(x.grad).all()
    assert torch.isfinite(y).all()
    AdapCC.clear()
    return True


def test_moe_alltoall_native_gpu():
    assert all(run_mp(_moe_alltoall_gpu, 2, backend="gloo", timeout=180))


def _oversized_tensor_split(rank, world):
    """Tensors above engine capacity are transparently split into
    capacity-sized sub-calls by the wrapper."""
    os.environ["ADAPCC_TIMEOUT_MS"] = "20000"
    import torch

    torch.cuda.set_device(0)
    from adapcc_amd.runtime.engine import NativeEngine
    from adapcc_amd.strategy.partrees import synthesize_stars

    eng = NativeEngine(rank, world, device=0, cap_bytes=4 << 20)
    eng.bootstrap()
    eng.set_strategy(synthesize_stars(world))
    n = (10 << 20) // 4  # 10 MB >> 4 MB cap
    torch.manual_seed(rank)
    t = torch.randn(n, device="cuda")
    cpu = t.cpu()
    import torch.distributed as dist

    g = [torch.zeros_like(cpu) for _ in range(world)]
    dist.all_gather(g, cpu)
    eng.all_reduce(t)
    eng.synchronize()
    torch.testing.assert_close(t.cpu(), torch.stack(g).sum(0), rtol=1e-4,
                               atol=1e-4)
    return True


def test_oversized_tensor_split():
    assert all(run_mp(_oversized_tensor_split, 2, backend="gloo", timeout=180))


def _gpt2_ddp_rehearsal(rank, world):
    """Closest 1-GPU rehearsal of the driver's multi-GPU bench: GPT-2 tiny,
    DDP + adapcc hook over the NATIVE engine, bf16-master optimizer, 3
    steps; replicas must remain byte-consistent."""
    os.environ["ADAPCC_TIMEOUT_MS"] = "30000"
    os.environ["ADAPCC_TRANSPORT"] = "native"
    import torch
    import torch.distributed as dist
    from torch.nn.parallel import DistributedDataParallel as DDP

    torch.cuda.set_device(0)
    from adapcc_amd import AdapCC, CommArgs
    from adapcc_amd.models.gpt2 import GPT2, GPT2Config
    from adapcc_amd.runtime.hook import AdapccDDPState, adapcc_allreduce_hook

    AdapCC.init(CommArgs(entry_point=-1), 0, rank, world)
    AdapCC.setup()
    torch.manual_seed(7)
    model = GPT2(GPT2Config.tiny()).cuda().to(torch.bfloat16)
    ddp = DDP(model, bucket_cap_mb=4)
    state = AdapccDDPState(AdapCC.communicator)
    ddp.register_comm_hook(state, adapcc_allreduce_hook)

    params = [p for p in ddp.parameters() if p.requires_grad]
    masters = [p.detach().float().clone() for p in params]
    for m in masters:
        m.grad = torch.zeros_like(m)
    opt = torch.optim.AdamW(masters, lr=1e-3, fused=True)

    torch.manual_seed(100 + rank)
    x = torch.randint(0, 2048, (2, 128), device="cuda")
    losses = []
    for step in range(3):
        state.on_step(step)
        for p in params:
            p.grad = None
        _, loss = ddp(x, x)
        loss.backward()
        torch._foreach_copy_([m.grad for m in masters],
                             [p.grad for p in params])
        opt.step()
        with torch.no_grad():
            torch._foreach_copy_(params, masters)
        losses.append(float(loss))
    torch.cuda.synchronize()
    assert all(map(lambda v: v == v, losses)), losses  # finite

    flat = torch.cat([p.detach().float().flatten() for p in params]).cpu()
    g = [torch.zeros_like(flat) for _ in range(world)]
    dist.all_gather(g, flat)
    for other in g:
        assert torch.allclose(other, g[0], atol=1e-5)
    AdapCC.clear()
    return True


def test_gpt2_ddp_rehearsal():
    assert all(run_mp(_gpt2_ddp_rehearsal, 2, backend="gloo", timeout=300))


def _oversized_collectives(rank, world):
    """AG/A2A/RS payloads beyond engine capacity must split transparently
    (round-1 verdict item 8)."""
    os.environ["ADAPCC_TIMEOUT_MS"] = "20000"
    import torch
    import torch.distributed as dist

    torch.cuda.set_device(0)
    from adapcc_amd.runtime.engine import NativeEngine
    from adapcc_amd.strategy.partrees import synthesize_stars

    cap = 1 << 20  # tiny 1 MB capacity to force splitting
    eng = NativeEngine(rank, world, device=0, cap_bytes=cap)
    eng.bootstrap()
    eng.set_strategy(synthesize_stars(world))

    per = 700_000  # fp32: world*per*4 bytes >> cap
    torch.manual_seed(rank)
    t = torch.randn(per, device="cuda")
    cpu = t.cpu()
    g = [torch.zeros_like(cpu) for _ in range(world)]
    dist.all_gather(g, cpu)

    # all_gather
    out = torch.empty(world * per, device="cuda")
    eng.all_gather(out, t)
    eng.synchronize()
    torch.testing.assert_close(out.cpu(), torch.cat(g), rtol=0, atol=0)

    # reduce_scatter
    big = torch.randn(world * per, device="cuda")
    bc = big.cpu()
    gb = [torch.zeros_like(bc) for _ in range(world)]
    dist.all_gather(gb, bc)
    rs_out = torch.empty(per, device="cuda")
    eng.reduce_scatter(rs_out, big)
    eng.synchronize()
    expect = torch.stack(gb).sum(0).view(world, per)[rank]
    torch.testing.assert_close(rs_out.cpu(), expect, rtol=1e-4, atol=1e-4)

    # all_to_all
    a2a_in = torch.randn(world * per, device="cuda")
    ac = a2a_in.cpu()
    ga = [torch.zeros_like(ac) for _ in range(world)]
    dist.all_gather(ga, ac)
    a2a_out = torch.empty_like(a2a_in)
    eng.all_to_all(a2a_out, a2a_in)
    eng.synchronize()
    expect = torch.cat([ga[r].view(world, per)[rank] for r in range(world)])
    torch.testing.assert_close(a2a_out.cpu(), expect, rtol=0, atol=0)

    # oversized allreduce still fine
    eng.all_reduce(t)
    eng.synchronize()
    torch.testing.assert_close(t.cpu(), torch.stack(g).sum(0), rtol=1e-4,
                               atol=1e-4)
    return True


def test_oversized_collectives_split():
    assert all(run_mp(_oversized_collectives, 2, backend="gloo", timeout=300))


def _scoped_barrier_straggler(rank, world):
    """An excluded straggler must not stall the active set's call: rank 1
    sleeps before its (excluded) call while rank 0 completes alone fast."""
    os.environ["ADAPCC_TIMEOUT_MS"] = "30000"
    import time

    import torch
    import torch.distributed as dist

    torch.cuda.set_device(0)
    from adapcc_amd.runtime.engine import NativeEngine
    from adapcc_amd.strategy.partrees import synthesize_stars

    eng = NativeEngine(rank, world, device=0, cap_bytes=8 << 20)
    eng.bootstrap()
    eng.set_strategy(synthesize_stars(world))
    active = [0, 1]  # exclude ranks 2, 3
    t = torch.full((100_000,), float(rank + 1), device="cuda")
    if rank >= 2:
        time.sleep(4.0)  # wedged-for-a-while straggler
    start = time.monotonic()
    eng.all_reduce(t, active=active)
    eng.synchronize()
    elapsed = time.monotonic() - start
    if rank < 2:
        assert elapsed < 2.5, f"active rank stalled {elapsed:.1f}s by straggler"
        expect = 1.0 + 2.0
        torch.testing.assert_close(t, torch.full_like(t, expect))
    dist.barrier()
    return True


def test_scoped_barrier_straggler():
    assert all(run_mp(_scoped_barrier_straggler, 4, backend="gloo",
                      timeout=300))


def _small_fused(rank, world):
    """Sub-256 KB payloads take the single-launch fused kernel; numerics
    must match across sizes, dtypes and repeats."""
    os.environ["ADAPCC_TIMEOUT_MS"] = "20000"
    import torch
    import torch.distributed as dist

    torch.cuda.set_device(0)
    from adapcc_amd.runtime.engine import NativeEngine
    from adapcc_amd.strategy.partrees import synthesize_stars

    eng = NativeEngine(rank, world, device=0, cap_bytes=8 << 20)
    eng.bootstrap()
    eng.set_strategy(synthesize_stars(world))
    for n in (16, 1024, 16384, 65536, 65537):
        for rep in range(3):
            torch.manual_seed(rank * 97 + n + rep)
            t = torch.randn(n, device="cuda")
            cpu = t.cpu()
            g = [torch.zeros_like(cpu) for _ in range(world)]
            dist.all_gather(g, cpu)
            eng.all_reduce(t)
            eng.synchronize()
            torch.testing.assert_close(t.cpu(), torch.stack(g).sum(0),
                                       rtol=1e-4, atol=1e-4)
    # mixed small/large back-to-back (slot/event interleaving)
    small = torch.full((4096,), float(rank), device="cuda")
    big = torch.full((1 << 20,), float(rank), device="cuda")
    eng.all_reduce(small)
    eng.all_reduce(big)
    eng.all_reduce(small)
    eng.synchronize()
    expect_small = sum(range(world)) * 1.0  # reduced twice? no: allreduce
    # small was allreduced twice: first sum(ranks), then sum over ranks of
    # that (world * sum)
    s1 = sum(range(world))
    torch.testing.assert_close(
        small, torch.full_like(small, float(s1 * world)))
    torch.testing.assert_close(big, torch.full_like(big, float(s1)))
    return True


def test_small_fused_collective():
    assert all(run_mp(_small_fused, 4, backend="gloo", timeout=300))
