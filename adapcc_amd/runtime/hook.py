"""DDP gradient-bucket communication hook (reference: commu.py:385-435).

Replaces RCCL's bucket allreduce inside PyTorch DDP with the adapcc engine.
The hook averages over the *active* set (relay control): stragglers excluded
from the active set neither contribute nor aggregate, but still receive the
reduced gradients (a stronger consistency model than the reference's BSP
mode, which let inactive replicas silently diverge — commu.py:424-431).

Usage:
    state = AdapccDDPState(AdapCC.communicator)
    model.register_comm_hook(state, adapcc_allreduce_hook)
    ...
    state.on_step(step)  # per-iteration relay update
"""

from dataclasses import dataclass, field
from typing import List, Optional

import torch

from ..communicator import Communicator


@dataclass
class AdapccDDPState:
    comm: Communicator
    step: int = 0
    bucket_elems: List[int] = field(default_factory=list)
    # BSP mode (reference commu.py:424-426): an inactive (straggler) rank
    # keeps its OWN gradients for the step instead of adopting the active
    # set's average. Default off: the engine already delivers the reduced
    # result to inactive ranks, which keeps replicas consistent.
    bsp_mode: bool = False
    _first_bucket_of_step: bool = True

    def on_step(self, step: int) -> None:
        """Call once per iteration before backward (reference:
        train_ddp.py:43 update_relay)."""
        self.step = step
        self._first_bucket_of_step = True
        self.comm.update_relay(step)

    @property
    def active(self) -> Optional[List[int]]:
        return self.comm.active_ranks

    def bucket_info(self) -> List[int]:
        """Element counts of the stable DDP buckets, recorded at step 1
        (reference: log/model_bucket_info.txt + commu.py:409-418)."""
        return list(self.bucket_elems)


def adapcc_allreduce_hook(
    state: AdapccDDPState, bucket
) -> torch.futures.Future[torch.Tensor]:
    tensor = bucket.buffer()
    if state._first_bucket_of_step:
        state._first_bucket_of_step = False
        if hasattr(state.comm, "notify_hook_ready"):
            state.comm.notify_hook_ready(state.step)
    if state.step == 1:
        # bucket layout is stable from DDP's rebuild at iteration 1 on;
        # record it (reference log/model_bucket_info.txt) and feed the
        # coordinator's rent-or-buy cost model
        state.bucket_elems.append(tensor.numel())
        state.comm._bucket_bytes.append(
            tensor.numel() * tensor.element_size())
    active = state.active
    inactive_bsp = (state.bsp_mode and active is not None
                    and state.comm.rank not in active)
    saved = tensor.clone() if inactive_bsp else None
    state.comm.all_reduce(tensor, active=active, average=True)
    if inactive_bsp:
        # keep the straggler's local gradients (the collective still ran so
        # peers were not blocked — reference BSP semantics)
        tensor.copy_(saved)
    fut: torch.futures.Future = torch.futures.Future()
    fut.set_result(tensor)
    return fut


def adapcc_bf16_compress_hook(
    state: AdapccDDPState, bucket
) -> torch.futures.Future[torch.Tensor]:
    """bf16-compressed variant: halves xGMI traffic for fp32 buckets.
    (PyTorch parity: ddp_comm_hooks.default_hooks.bf16_compress_hook.)"""
    buf = bucket.buffer()
    half = buf.to(torch.bfloat16)
    state.comm.all_reduce(half, active=state.active, average=True)
    fut: torch.futures.Future = torch.futures.Future()
    buf.copy_(half.to(buf.dtype))
    fut.set_result(buf)
    return fut
