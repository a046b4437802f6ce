"""Randomized soak test of the native engine: a long mixed sequence of
primitives, sizes, dtypes and active sets on one engine instance (world 4
processes on one GPU). Catches epoch/flag-protocol races that single-shot
tests miss."""

import os
import random

import pytest
import torch

from util_mp import run_mp

pytestmark = pytest.mark.gpu

N_OPS = 40


def _fuzz(rank, world, seed):
    os.environ["ADAPCC_TIMEOUT_MS"] = "30000"
    import torch
    import torch.distributed as dist

    torch.cuda.set_device(0)
    from adapcc_amd.runtime.engine import NativeEngine
    from adapcc_amd.strategy.partrees import synthesize_chains, synthesize_stars

    eng = NativeEngine(rank, world, device=0, cap_bytes=32 << 20)
    eng.bootstrap()
    eng.set_strategy(synthesize_stars(world))

    rng = random.Random(seed)  # same sequence on every rank

    def ref_gather(t):
        cpu = t.float().cpu()
        g = [torch.zeros_like(cpu) for _ in range(world)]
        dist.all_gather(g, cpu)
        return g

    for step in range(N_OPS):
        op = rng.choice(["allreduce", "allreduce", "allreduce", "reduce",
                         "broadcast", "allgather", "reducescatter",
                         "alltoall", "restrategize"])
        n = rng.choice([64, 1000, 65536, 300_001])
        dtype = rng.choice([torch.float32, torch.float32, torch.bfloat16])
        tol = 1e-4 if dtype == torch.float32 else 0.1
        torch.manual_seed(seed * 1000 + step * 10 + rank)

        if op == "restrategize":
            kind = rng.choice(["stars", "chains", "weighted"])
            if kind == "stars":
                eng.set_strategy(synthesize_stars(world))
            elif kind == "chains":
                eng.set_strategy(synthesize_chains(world, num_trees=2))
            else:
                s = synthesize_stars(world)
                s.slice_weights = [rng.uniform(0.2, 4.0)
                                   for _ in range(world)]
                eng.set_strategy(s)
            continue
        if op == "allreduce":
            active = (sorted(rng.sample(range(world), rng.randint(1, world)))
                      if rng.random() < 0.3 else None)
            avg = rng.random() < 0.5
            t = torch.randn(n, device="cuda", dtype=dtype)
            g = ref_gather(t)
            act = active or list(range(world))
            expect = torch.stack([g[r] for r in act]).sum(0)
            if avg:
                expect /= len(act)
            eng.all_reduce(t, active=active, average=avg)
            eng.synchronize()
            torch.testing.assert_close(t.float().cpu(), expect, rtol=tol,
                                       atol=tol), (op, step)
        elif op == "reduce":
            root = rng.randrange(world)
            t = torch.randn(n, device="cuda", dtype=dtype)
            g = ref_gather(t)
            eng.reduce(t, root=root)
            eng.synchronize()
            if rank == root:
                torch.testing.assert_close(t.float().cpu(),
                                           torch.stack(g).sum(0), rtol=tol,
                                           atol=tol)
        elif op == "broadcast":
            root = rng.randrange(world)
            t = torch.randn(n, device="cuda", dtype=dtype)
            g = ref_gather(t)
            eng.broadcast(t, root=root)
            eng.synchronize()
            torch.testing.assert_close(t.float().cpu(), g[root], rtol=1e-6,
                                       atol=1e-6)
        elif op == "allgather":
            t = torch.randn(n, device="cuda", dtype=dtype)
            g = ref_gather(t)
            out = torch.empty(world * n, device="cuda", dtype=dtype)
            eng.all_gather(out, t)
            eng.synchronize()
            torch.testing.assert_close(out.float().cpu(), torch.cat(g),
                                       rtol=1e-6, atol=1e-6)
        elif op == "reducescatter":
            t = torch.randn(world * n, device="cuda", dtype=dtype)
            g = ref_gather(t)
            out = torch.empty(n, device="cuda", dtype=dtype)
            eng.reduce_scatter(out, t)
            eng.synchronize()
            total = torch.stack(g).sum(0)
            torch.testing.assert_close(out.float().cpu(),
                                       total[rank * n:(rank + 1) * n],
                                       rtol=tol, atol=tol)
        elif op == "alltoall":
            t = torch.randn(world * n, device="cuda", dtype=dtype)
            g = ref_gather(t)
            out = torch.empty(world * n, device="cuda", dtype=dtype)
            eng.all_to_all(out, t)
            eng.synchronize()
            expect = torch.cat([g[s][rank * n:(rank + 1) * n]
                                for s in range(world)])
            torch.testing.assert_close(out.float().cpu(), expect, rtol=1e-6,
                                       atol=1e-6)
    return True


@pytest.mark.parametrize("seed", [7, 1234])
def test_engine_fuzz(seed):
    assert all(run_mp(_fuzz, 4, backend="gloo", args=(seed,), timeout=600))
