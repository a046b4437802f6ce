"""Process-group fallback engine.

Semantically equivalent collectives over torch.distributed (RCCL on GPU,
gloo on CPU). Used (a) as the correctness reference in tests, (b) on CPU-only
machines, (c) as an explicit opt-in (ADAPCC_TRANSPORT=pg) — never as a
silent substitute for the native engine on a GPU box.

Relay semantics: a sum over the active set only. Implemented by zeroing the
inactive ranks' contribution before the process-group allreduce, which is
bit-wise what the native engine computes (inactive sendbufs are never
pulled).
"""

from __future__ import annotations

from typing import Optional, Sequence

import torch
import torch.distributed as dist

from ..topology.formats import Strategy


class ProcessGroupEngine:
    def __init__(self, rank: int, world_size: int, group=None) -> None:
        self.rank = rank
        self.world_size = world_size
        self.group = group

    def bootstrap(self, group=None) -> None:
        if group is not None:
            self.group = group

    def set_strategy(self, strategy: Strategy) -> None:
        self.strategy = strategy

    def all_reduce(
        self,
        tensor: torch.Tensor,
        active: Optional[Sequence[int]] = None,
        average: bool = False,
    ) -> torch.Tensor:
        if self.world_size == 1:
            return tensor
        n = self.world_size
        if active is not None and len(active) > 0 and len(set(active)) < n:
            act = set(active)
            n = len(act)
            if self.rank not in act:
                tensor.zero_()
        dist.all_reduce(tensor, op=dist.ReduceOp.SUM, group=self.group)
        if average:
            tensor.div_(n)
        return tensor

    def synchronize(self) -> None:
        if torch.cuda.is_available():
            torch.cuda.synchronize()

    def query_error(self):
        return (0, 0)

    def self_test(self) -> None:
        pass
