"""Tree-strategy engine over point-to-point messages (RCCL p2p on GPU,
gloo on CPU) — the second transport of the north star.

Executes the SAME unit plans as the native hipIpc engine
(adapcc_amd._core.compute_plan — copy/reduce/bcast units with relay
analysis), but moves chunks with torch.distributed isend/recv instead of
direct xGMI pulls. This is the transport for topologies the shared-memory
engine cannot reach (multi-node forests from the synthesizer, WAN-ish
heterogeneous links where AdapCC's adaptive trees beat flat collectives)
and a CPU-executable tree engine for tests.

Message discipline: each rank walks its units in the global (chunk, tree)
key order; produced chunks are pushed to their consumer with non-blocking
isend, consumed chunks are received in unit order (tags encode
(tree, chunk) so concurrent trees cannot cross wires). The same
acyclicity argument as the device protocol applies (docs/DESIGN.md).
"""

from __future__ import annotations

from typing import Dict, List, Optional, Sequence, Tuple

import torch
import torch.distributed as dist

from ..topology.formats import Strategy
from .engine import strategy_parent_arrays

SEND, ACC, RESULT = 0, 1, 2
_MAX_TAG = 1 << 20


def _tag(tree: int, chunk: int, phase: int) -> int:
    return phase * (1 << 16) + tree * (1 << 10) + (chunk % (1 << 10))


class P2PTreeEngine:
    """Same interface as NativeEngine (all_reduce/synchronize/...)."""

    def __init__(self, rank: int, world_size: int, group=None) -> None:
        from adapcc_amd import _core

        self.core = _core
        self.rank = rank
        self.world_size = world_size
        self.group = group
        self.parents: Optional[List[List[int]]] = None
        self.chunk_bytes = 2 << 20

    def bootstrap(self, group=None) -> None:
        if group is not None:
            self.group = group

    def set_strategy(self, strategy: Strategy) -> None:
        self.parents = strategy_parent_arrays(strategy, self.world_size)
        self.chunk_bytes = strategy.chunk_bytes
        self.slice_weights = list(strategy.slice_weights or [])

    # ------------------------------------------------------------------

    def all_reduce(
        self,
        tensor: torch.Tensor,
        active: Optional[Sequence[int]] = None,
        average: bool = False,
    ) -> torch.Tensor:
        if self.world_size == 1:
            return tensor
        self._check(tensor)
        act = sorted(active) if active else list(range(self.world_size))
        plan = self.core.compute_plan(
            self.parents, self.rank, tensor.numel(), tensor.element_size(),
            self.chunk_bytes, act, self.slice_weights)
        self._execute(plan, tensor, tensor, act, average)
        return tensor

    def reduce(self, tensor: torch.Tensor, root: int = 0,
               active: Optional[Sequence[int]] = None,
               average: bool = False) -> torch.Tensor:
        if self.world_size == 1:
            return tensor
        self._check(tensor)
        act = sorted(active) if active else list(range(self.world_size))
        plan = self.core.compute_primitive_plan(
            "reduce", self.world_size, self.rank, tensor.numel(),
            tensor.element_size(), self.chunk_bytes, root=root,
            parents=self.parents, active=act,
            slice_weights=self.slice_weights)
        self._execute(plan, tensor, tensor, act, average)
        return tensor

    def broadcast(self, tensor: torch.Tensor, root: int = 0) -> torch.Tensor:
        if self.world_size == 1:
            return tensor
        self._check(tensor)
        plan = self.core.compute_primitive_plan(
            "broadcast", self.world_size, self.rank, tensor.numel(),
            tensor.element_size(), self.chunk_bytes, root=root)
        self._execute(plan, tensor, tensor, list(range(self.world_size)),
                      False)
        return tensor

    def all_gather(self, out: torch.Tensor, tensor: torch.Tensor) -> torch.Tensor:
        self._check(tensor)
        if out.numel() != tensor.numel() * self.world_size:
            raise ValueError("all_gather: out must be world_size * in")
        if self.world_size == 1:
            out.view(-1).copy_(tensor.view(-1))
            return out
        plan = self.core.compute_primitive_plan(
            "allgather", self.world_size, self.rank, tensor.numel(),
            tensor.element_size(), self.chunk_bytes)
        self._execute(plan, tensor, out, list(range(self.world_size)), False)
        return out

    def reduce_scatter(self, out: torch.Tensor, tensor: torch.Tensor,
                       active: Optional[Sequence[int]] = None,
                       average: bool = False) -> torch.Tensor:
        self._check(tensor)
        if tensor.numel() != out.numel() * self.world_size:
            raise ValueError("reduce_scatter: in must be world_size * out")
        if self.world_size == 1:
            out.view(-1).copy_(tensor.view(-1))
            return out
        act = sorted(active) if active else list(range(self.world_size))
        plan = self.core.compute_primitive_plan(
            "reducescatter", self.world_size, self.rank, out.numel(),
            tensor.element_size(), self.chunk_bytes, active=act)
        self._execute(plan, tensor, out, act, average)
        return out

    def all_to_all(self, out: torch.Tensor, tensor: torch.Tensor) -> torch.Tensor:
        self._check(tensor)
        if out.numel() != tensor.numel():
            raise ValueError("all_to_all: out must match in size")
        if out.data_ptr() == tensor.data_ptr():
            raise ValueError("all_to_all: out must not alias the input")
        if tensor.numel() % self.world_size:
            raise ValueError("all_to_all: size must divide world_size")
        if self.world_size == 1:
            out.view(-1).copy_(tensor.view(-1))
            return out
        plan = self.core.compute_primitive_plan(
            "alltoall", self.world_size, self.rank,
            tensor.numel() // self.world_size, tensor.element_size(),
            self.chunk_bytes)
        self._execute(plan, tensor, out, list(range(self.world_size)), False)
        return out

    def _check(self, tensor: torch.Tensor) -> None:
        if self.parents is None:
            raise RuntimeError("no strategy set")
        if not tensor.is_contiguous():
            raise ValueError("p2p engine requires a contiguous tensor")

    def _execute(self, plan: dict, in_tensor: torch.Tensor,
                 out_tensor: torch.Tensor, act: List[int],
                 average: bool) -> None:
        flat = in_tensor.view(-1)
        out_flat = out_tensor.view(-1)
        # staging buffers per (tree, chunk): received partials / results
        acc: Dict[Tuple[int, int], torch.Tensor] = {}
        result: Dict[Tuple[int, int], torch.Tensor] = {}
        pending = []

        me_active = self.rank in act
        scale = 1.0 / len(act) if average else 1.0

        def push(dst: int, t: int, c: int, buf: torch.Tensor, phase: int):
            if dst == self.rank:
                return  # self-delivery is a local no-op (buf already local)
            pending.append(dist.isend(buf.contiguous(), dst,
                                      group=self.group,
                                      tag=_tag(t, c, phase)))

        def pull(src: int, t: int, c: int, n: int, phase: int) -> torch.Tensor:
            buf = torch.empty(n, dtype=flat.dtype, device=flat.device)
            dist.recv(buf, src, group=self.group, tag=_tag(t, c, phase))
            return buf

        # 1) leaf contributions first — they have no dependencies, and
        # every rank must have its sends in flight before blocking on
        # reduce receives (the device engine's copyin-before-reduce order)
        if me_active:
            for u in plan["copy"]:
                if not u["notify_to"]:
                    continue
                t, c = u["tree"], u["chunk"]
                off, n = u["offset"], u["count"]
                phase = 0 if u["flag_space"] == 0 else 2
                for dst in u["notify_to"]:
                    push(dst, t, c, flat[off:off + n], phase=phase)

        # 2) reduce phase in global (chunk, tree) order
        for u in plan["reduce"]:
            t, c = u["tree"], u["chunk"]
            off, n = u["offset"], u["count"]
            pieces = []
            for (src, kind) in u["srcs"]:
                if src == self.rank:
                    continue
                pieces.append(pull(src, t, c, n, phase=0 if kind == SEND else 1))
            if u["include_self"]:
                pieces.append(flat[off:off + n])
            out = torch.stack(pieces).sum(0) if len(pieces) > 1 else \
                pieces[0].clone()
            acc[(t, c)] = out
            if u["notify"]:
                push(u["consumer"], t, c, out, phase=1)
            if u["is_root"]:
                for k in u["publish_to"]:
                    push(k, t, c, out, phase=2)

        # 3) broadcast/receive phase: pull published chunks into OUT
        for u in plan["bcast"]:
            t, c = u["tree"], u["chunk"]
            soff, doff, n = u["src_offset"], u["dst_offset"], u["count"]
            if u["parent"] < 0:
                buf = acc.get((t, c))
                if buf is None:  # self-staged source (allgather/alltoall/bcast)
                    buf = flat[soff:soff + n]
            else:
                buf = pull(u["parent"], t, c, n, phase=2)
            if u["forward"]:
                result[(t, c)] = buf
                for k in u["publish_to"]:
                    push(k, t, c, buf, phase=2)
            if scale != 1.0:
                out_flat[doff:doff + n] = buf * scale
            else:
                out_flat[doff:doff + n] = buf

        for w in pending:
            w.wait()
        dist.barrier(group=self.group)

    def synchronize(self) -> None:
        if torch.cuda.is_available():
            torch.cuda.synchronize()

    def query_error(self):
        return (0, 0)

    def self_test(self) -> None:
        t = torch.full((256,), float(self.rank + 1))
        if torch.cuda.is_available():
            t = t.cuda()
        self.all_reduce(t)
        expect = float(sum(range(1, self.world_size + 1)))
        if not torch.allclose(t, torch.full_like(t, expect)):
            raise RuntimeError("p2p engine self-test failed")
