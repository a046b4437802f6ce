#!/usr/bin/env python3
"""Elastic DDP training with checkpoint/resume (reference:
models/image-classification/main_elastic.py + launch_elastic.sh).

Launch with torchrun's elastic rendezvous; on membership change torchrun
restarts the workers, which resume from the newest checkpoint: the rank
holding the best (newest-epoch) checkpoint broadcasts its state to
everyone (reference main_elastic.py:306-383).

    python -m torch.distributed.run --nnodes=1:3 --max-restarts=3 \
        --rdzv-backend=c10d --rdzv-endpoint=127.0.0.1:29400 \
        examples/train_elastic.py --steps 50
"""

from __future__ import annotations

import argparse
import io
import os
import sys
import tempfile

import torch
import torch.distributed as dist
from torch.nn.parallel import DistributedDataParallel as DDP

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from adapcc_amd import AdapCC, CommArgs
from adapcc_amd.models.resnet import ResNet18
from adapcc_amd.runtime.hook import AdapccDDPState, adapcc_allreduce_hook


class State:
    """Capture/apply training state (reference main_elastic.py:188-237)."""

    def __init__(self, model, opt):
        self.model = model
        self.opt = opt
        self.step = 0

    def capture(self) -> dict:
        return {
            "step": self.step,
            "model": self.model.state_dict(),
            "opt": self.opt.state_dict(),
        }

    def apply(self, snap: dict) -> None:
        self.step = snap["step"]
        self.model.load_state_dict(snap["model"])
        self.opt.load_state_dict(snap["opt"])

    def save(self, path: str) -> None:
        # atomic save via tmp+rename (reference main_elastic.py:395-408)
        d = os.path.dirname(os.path.abspath(path))
        fd, tmp = tempfile.mkstemp(dir=d)
        with os.fdopen(fd, "wb") as f:
            torch.save(self.capture(), f)
        os.replace(tmp, path)

    def load_and_sync(self, path: str, device) -> None:
        """Resume: the rank with the newest checkpoint broadcasts it
        (reference main_elastic.py:306-383)."""
        step = -1
        blob = b""
        if os.path.exists(path):
            with open(path, "rb") as f:
                blob = f.read()
            step = torch.load(io.BytesIO(blob), map_location="cpu",
                              weights_only=False)["step"]
        if not dist.is_initialized():
            if step >= 0:
                self.apply(torch.load(io.BytesIO(blob), map_location=device,
                                      weights_only=False))
            return
        t = torch.tensor([step, dist.get_rank()], dtype=torch.long)
        best = t.clone()
        dist.all_reduce(best, op=dist.ReduceOp.MAX)
        best_step = int(best[0])
        if best_step < 0:
            return
        # find a rank holding best_step (max of (step==best)*rank)
        holder = torch.tensor(
            [dist.get_rank() if step == best_step else -1], dtype=torch.long)
        dist.all_reduce(holder, op=dist.ReduceOp.MAX)
        objs = [blob if step == best_step else None]
        dist.broadcast_object_list(objs, src=int(holder[0]))
        self.apply(torch.load(io.BytesIO(objs[0]), map_location=device,
                              weights_only=False))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--batch", type=int, default=16)
    p.add_argument("--ckpt", default="checkpoint_elastic.pt")
    p.add_argument("--ckpt_freq", type=int, default=5)
    args = p.parse_args()

    rank = int(os.environ.get("RANK", 0))
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    use_cuda = torch.cuda.is_available()
    dev_idx = local_rank % max(1, torch.cuda.device_count() or 1)
    device = torch.device(f"cuda:{dev_idx}" if use_cuda else "cpu")
    if use_cuda:
        torch.cuda.set_device(device)
    if world > 1:
        dist.init_process_group("nccl" if use_cuda else "gloo")

    torch.manual_seed(5)
    model = ResNet18(num_classes=100).to(device)
    opt = torch.optim.SGD(model.parameters(), lr=0.01, momentum=0.9)
    state = State(model, opt)
    state.load_and_sync(args.ckpt, device)

    AdapCC.init(CommArgs(entry_point=-1), local_rank, rank, world)
    AdapCC.setup()
    if world > 1:
        ddp = DDP(model, device_ids=[device.index] if use_cuda else None)
        hstate = AdapccDDPState(AdapCC.communicator)
        ddp.register_comm_hook(hstate, adapcc_allreduce_hook)
    else:
        ddp, hstate = model, None

    crit = torch.nn.CrossEntropyLoss()
    torch.manual_seed(100 + rank)
    x = torch.randn(args.batch, 3, 64, 64, device=device)
    y = torch.randint(0, 100, (args.batch,), device=device)

    start = state.step
    for step in range(start, args.steps):
        if hstate is not None:
            hstate.on_step(step)
        opt.zero_grad(set_to_none=True)
        loss = crit(ddp(x), y)
        loss.backward()
        opt.step()
        state.step = step + 1
        if (step + 1) % args.ckpt_freq == 0 and rank == 0:
            state.save(args.ckpt)
        if rank == 0:
            print(f"step {step}: loss {loss.item():.4f}", flush=True)

    if rank == 0:
        state.save(args.ckpt)
    AdapCC.clear()
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
