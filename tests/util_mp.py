"""Multi-process test helpers: spawn `world` fresh interpreters joined by
a FileStore-initialized process group (gloo on CPU; on GPU boxes the
workers still rendezvous over gloo and drive CUDA device 0 directly, which
is how the single-GPU multi-rank engine tests work). Results are collected
per rank; any worker exception fails the test with its traceback."""

from __future__ import annotations

import os
import tempfile

import torch.multiprocessing as mp


def _worker(rank, world, backend, fn, init_file, args, q):
    import torch.distributed as dist

    try:
        if backend == "nccl":
            import torch

            torch.cuda.set_device(rank)
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        dist.init_process_group(
            backend=backend,
            init_method=f"file://{init_file}",
            rank=rank,
            world_size=world,
        )
        res = fn(rank, world, *args)
        q.put((rank, "ok", res))
    except Exception as e:  # pragma: no cover
        import traceback

        q.put((rank, "err", f"{e}\n{traceback.format_exc()}"))
    finally:
        try:
            dist.destroy_process_group()
        except Exception:
            pass


def run_mp(fn, world: int, backend: str = "gloo", args: tuple = (),
           timeout: float = 120.0):
    """Run fn(rank, world, *args) in `world` processes; returns results by
    rank; raises on any worker error."""
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    with tempfile.NamedTemporaryFile(delete=False) as f:
        init_file = f.name
    os.unlink(init_file)
    procs = [
        ctx.Process(target=_worker,
                    args=(r, world, backend, fn, init_file, args, q))
        for r in range(world)
    ]
    for p in procs:
        p.start()
    results = {}
    try:
        for _ in range(world):
            rank, status, res = q.get()
            if status == "err":
                raise AssertionError(f"rank {rank} failed:\n{res}")
            results[rank] = res
    finally:
        for p in procs:
            p.join(timeout=timeout)
            if p.is_alive():
                p.terminate()
                p.join(5)
    return [results[r] for r in range(world)]
