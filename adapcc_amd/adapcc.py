"""AdapCC facade (reference: adapcc.py:15-76).

Class-level API kept call-compatible with the reference so its usage recipes
port 1:1:

    AdapCC.init(args, local_rank, world_rank, world_size)
    AdapCC.setup(Primitive.ALLREDUCE)
    AdapCC.communicator.all_reduce(tensor)
    AdapCC.reconstruct_topology()
    AdapCC.clear()

plus the DDP hook in adapcc_amd.runtime.hook.
"""

from __future__ import annotations

from typing import Optional, Sequence

import torch

from .communicator import CommArgs, Communicator
from .primitives import Primitive


class AdapCC:
    communicator: Optional[Communicator] = None

    @classmethod
    def init(cls, args, local_rank: int, world_rank: int, world_size: int,
             group=None) -> None:
        if not isinstance(args, CommArgs):
            args = CommArgs.from_namespace(args)
        cls.communicator = Communicator(args, local_rank, world_rank,
                                        world_size, group=group)
        cls.communicator.run_entry_point()

    @classmethod
    def setup(cls, primitive: Primitive = Primitive.ALLREDUCE) -> None:
        cls._require_init()
        cls.communicator.setup(primitive)

    @classmethod
    def allreduce(cls, tensor: torch.Tensor,
                  active: Optional[Sequence[int]] = None,
                  average: bool = False) -> torch.Tensor:
        cls._require_init()
        return cls.communicator.all_reduce(tensor, active=active, average=average)

    # reference spelling aliases
    all_reduce = allreduce

    @classmethod
    def reduce(cls, tensor: torch.Tensor, root: int = 0,
               active: Optional[Sequence[int]] = None) -> torch.Tensor:
        cls._require_init()
        return cls.communicator.reduce(tensor, root=root, active=active)

    @classmethod
    def boardcast(cls, tensor: torch.Tensor, root: int = 0) -> torch.Tensor:
        cls._require_init()
        return cls.communicator.broadcast(tensor, root=root)

    broadcast = boardcast

    @classmethod
    def allgather(cls, out: torch.Tensor, tensor: torch.Tensor) -> torch.Tensor:
        cls._require_init()
        return cls.communicator.all_gather(out, tensor)

    @classmethod
    def alltoall(cls, out: torch.Tensor, tensor: torch.Tensor) -> torch.Tensor:
        cls._require_init()
        return cls.communicator.all_to_all(out, tensor)

    @classmethod
    def reducescatter(cls, out: torch.Tensor, tensor: torch.Tensor) -> torch.Tensor:
        cls._require_init()
        return cls.communicator.reduce_scatter(out, tensor)

    @classmethod
    def update_relay(cls, step: int) -> None:
        cls._require_init()
        cls.communicator.update_relay(step)

    @classmethod
    def reconstruct_topology(cls) -> None:
        cls._require_init()
        cls.communicator.reconstruct_topology()

    @classmethod
    def clear(cls) -> None:
        if cls.communicator is not None:
            cls.communicator.clear()
            cls.communicator = None

    @classmethod
    def stats(cls) -> dict:
        cls._require_init()
        return cls.communicator.stats()

    @classmethod
    def _require_init(cls) -> None:
        if cls.communicator is None:
            raise RuntimeError("AdapCC.init() has not been called")


def _primitive_demo() -> None:
    """Golden-output primitive run (reference: adapcc.py:81-117 __main__ and
    log/primitive): each rank allreduces a ones*(rank+1) 16-float tensor and
    prints the world-sum tensor.

        python -m torch.distributed.run --nproc-per-node 4 \
            --master-addr 127.0.0.1 -m adapcc_amd.adapcc
    """
    import os

    import torch.distributed as dist

    rank = int(os.environ.get("RANK", 0))
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    use_cuda = torch.cuda.is_available()
    if use_cuda:
        torch.cuda.set_device(local_rank % max(1, torch.cuda.device_count()))
    if world > 1:
        backend = ("nccl" if use_cuda and torch.cuda.device_count() >= world
                   else "gloo")
        dist.init_process_group(backend)

    AdapCC.init(CommArgs(entry_point=-1), local_rank, rank, world)
    AdapCC.setup(Primitive.ALLREDUCE)
    device = "cuda" if use_cuda else "cpu"
    for it in range(2):
        t = torch.full((16,), float(rank + 1), device=device)
        AdapCC.allreduce(t)
        AdapCC.communicator.synchronize()
        print(f"[Rank {rank}] iter {it}: {t.cpu().tolist()[:4]} ... "
              f"(expect {sum(range(1, world + 1)) * (1.0)})", flush=True)
    AdapCC.clear()
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    _primitive_demo()
