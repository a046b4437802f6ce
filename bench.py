#!/usr/bin/env python3
"""Flagship benchmark: GPT-2 small DDP training step over the adapcc engine.

Measures whole-job tokens/sec (bf16 compute with fp32-master AdamW by
default, synthetic tokens, random-init weights) with gradient buckets
allreduced by the adapcc_amd engine via the DDP communication hook — the
reference's train_ddp.py workload re-targeted at GPT-2 small per
BASELINE.json.

Launch (driver contract):
    python bench.py --gpus 1 --steps K --warmup W
    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
        --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

os.environ.setdefault("GPU_MAX_HW_QUEUES", "8")  # comm/compute overlap

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from adapcc_amd import AdapCC, CommArgs  # noqa: E402
from adapcc_amd.models.gpt2 import GPT2, GPT2Config  # noqa: E402
from adapcc_amd.runtime.hook import AdapccDDPState, adapcc_allreduce_hook  # noqa: E402


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--batch", type=int, default=128, help="per-GPU micro batch")
    p.add_argument("--seq", type=int, default=1024)
    p.add_argument("--model", type=str, default="small",
                   choices=["tiny", "small", "medium"])
    p.add_argument("--bucket-mb", type=int, default=100)
    p.add_argument("--device", type=str, default=None)
    p.add_argument("--transport", type=str, default=None,
                   help="override ADAPCC_TRANSPORT")
    p.add_argument("--precision", type=str, default="bf16",
                   choices=["autocast", "bf16"],
                   help="autocast: fp32 params + bf16 autocast; bf16: bf16 "
                        "params/grads with fp32 master AdamW (Megatron-style)")
    p.add_argument("--graphs", type=str, default="auto",
                   choices=["auto", "on", "off"],
                   help="capture the whole step in a hipGraph (world==1)")
    return p.parse_args()


def main() -> None:
    args = parse_args()
    if args.transport:
        os.environ["ADAPCC_TRANSPORT"] = args.transport

    rank = int(os.environ.get("RANK", 0))
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))

    use_cuda = torch.cuda.is_available() and args.device != "cpu"
    dev_idx = local_rank % max(1, torch.cuda.device_count() or 1)
    device = torch.device(f"cuda:{dev_idx}" if use_cuda else "cpu")
    if use_cuda:
        torch.cuda.set_device(device)

    if world > 1:
        # RCCL requires one rank per GPU; when ranks share a device
        # (1-GPU rehearsals) bootstrap over gloo instead — the data plane
        # is the native xGMI engine either way.
        backend = "nccl" if (use_cuda and
                             torch.cuda.device_count() >= world) else "gloo"
        dist.init_process_group(backend=backend, rank=rank, world_size=world)

    torch.manual_seed(1234 + rank)
    cfg = getattr(GPT2Config, args.model)()
    model = GPT2(cfg).to(device)
    n_params = model.num_params()
    bf16_master = args.precision == "bf16" and use_cuda
    if bf16_master:
        model = model.to(torch.bfloat16)

    AdapCC.init(CommArgs(entry_point=-1, policy="par-trees"),
                local_rank, rank, world)
    AdapCC.setup()

    if world > 1:
        ddp = torch.nn.parallel.DistributedDataParallel(
            model, device_ids=[device.index] if use_cuda else None,
            bucket_cap_mb=args.bucket_mb,
        )
        state = AdapccDDPState(AdapCC.communicator)
        ddp.register_comm_hook(state, adapcc_allreduce_hook)
        train_mod = ddp
    else:
        state = None
        train_mod = model

    want_graphs = (args.graphs == "on" or
                   (args.graphs == "auto" and world == 1)) and use_cuda
    if bf16_master:
        # fp32 master copies; bf16 params/grads on the model (and on the
        # wire: the DDP hook then allreduces bf16 buckets — half traffic)
        model_params = [p for p in train_mod.parameters() if p.requires_grad]
        masters = [p.detach().float().clone() for p in model_params]
        for m in masters:
            m.grad = torch.zeros_like(m)
        opt = torch.optim.AdamW(masters, lr=3e-4, betas=(0.9, 0.95),
                                weight_decay=0.1, fused=True,
                                capturable=want_graphs)
    else:
        try:
            opt = torch.optim.AdamW(train_mod.parameters(), lr=3e-4,
                                    betas=(0.9, 0.95), weight_decay=0.1,
                                    fused=use_cuda, capturable=want_graphs)
        except (RuntimeError, ValueError):
            opt = torch.optim.AdamW(train_mod.parameters(), lr=3e-4,
                                    betas=(0.9, 0.95), weight_decay=0.1,
                                    foreach=True)
            want_graphs = False

    B, T = args.batch, min(args.seq, cfg.n_positions)
    data = torch.randint(0, cfg.vocab_size, (B, T + 1), device=device)
    x, y = data[:, :-1], data[:, 1:].contiguous()

    amp_dtype = torch.bfloat16
    autocast = torch.autocast(device_type=device.type, dtype=amp_dtype,
                              enabled=True)

    def eager_step(i: int) -> None:
        if state is not None:
            state.on_step(i)
        if bf16_master:
            for p in model_params:
                p.grad = None
            _, loss = train_mod(x, y)
            loss.backward()
            torch._foreach_copy_([m.grad for m in masters],
                                 [p.grad for p in model_params])
            opt.step()
            with torch.no_grad():
                torch._foreach_copy_(model_params, masters)
        else:
            opt.zero_grad(set_to_none=True)
            with autocast:
                _, loss = train_mod(x, y)
            loss.backward()
            opt.step()

    step = eager_step
    use_graphs = want_graphs
    if use_graphs:
        # whole-step hipGraph: fwd + bwd + fused AdamW in one replay
        # (the multi-GPU path keeps eager enqueue: the engine's cross-call
        # event chain is not capturable)
        try:
            for i in range(2):  # allocator warmup on a side stream
                s = torch.cuda.Stream()
                s.wait_stream(torch.cuda.current_stream())
                with torch.cuda.stream(s):
                    eager_step(i)
                torch.cuda.current_stream().wait_stream(s)
            torch.cuda.synchronize()
            g = torch.cuda.CUDAGraph()
            if bf16_master:
                with torch.cuda.graph(g):
                    for p in model_params:
                        if p.grad is not None:
                            p.grad.zero_()
                    _, loss = train_mod(x, y)
                    loss.backward()
                    torch._foreach_copy_([m.grad for m in masters],
                                         [p.grad for p in model_params])
                    opt.step()
                    with torch.no_grad():
                        torch._foreach_copy_(model_params, masters)
            else:
                opt.zero_grad(set_to_none=False)  # grads stay allocated
                with torch.cuda.graph(g):
                    with autocast:
                        _, loss = train_mod(x, y)
                    loss.backward()
                    opt.step()
                    opt.zero_grad(set_to_none=False)

            def graph_step(i: int) -> None:
                g.replay()

            step = graph_step
        except Exception as e:  # pragma: no cover - graph support varies
            print(f"# hipGraph capture unavailable ({e}); eager path",
                  file=sys.stderr)
            step = eager_step
            use_graphs = False

    for i in range(args.warmup):
        step(i)
    if use_cuda:
        torch.cuda.synchronize()
    if world > 1:
        dist.barrier()

    t0 = time.perf_counter()
    for i in range(args.steps):
        step(args.warmup + i)
    if use_cuda:
        torch.cuda.synchronize()
    if world > 1:
        dist.barrier()
    elapsed = time.perf_counter() - t0

    # max over ranks
    if world > 1:
        t = torch.tensor([elapsed], device=device if use_cuda else "cpu",
                         dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    tokens = world * B * T * args.steps
    toks_per_s = tokens / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    if rank == 0:
        out = {
            "metric": "gpt2_ddp_tokens_per_sec",
            "value": toks_per_s,
            "unit": "tokens/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic",
            "config": {
                "model": f"gpt2-{args.model}-{n_params/1e6:.0f}M",
                "global_batch": world * B,
                "seq_len": T,
                "parallelism": f"dp{world}",
                "bucket_cap_mb": args.bucket_mb,
                "hipgraph": bool(use_graphs),
                "precision": args.precision,
                "transport": AdapCC.communicator.effective_transport or os.environ.get("ADAPCC_TRANSPORT", "auto"),
            },
        }
        print(json.dumps(out), flush=True)

    AdapCC.clear()
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
