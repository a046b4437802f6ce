"""Control plane (reference: commu.py CudaCommu).

Owns the engine (native xGMI pull-engine on GPU, process-group fallback on
CPU or by explicit opt-in), the adaptation flow (detect -> profile ->
synthesize), relay control, and the DDP hook state.

Differences from the reference by design:
 - bootstrap/artifact exchange goes over torch.distributed object
   collectives instead of scp + ctypes + sleep-based readiness
   (ref commu.py:318 time_init_wait)
 - one engine serves all primitives (no per-primitive contexts/ports)
 - transport selection: ADAPCC_TRANSPORT = native | pg | auto (default auto:
   native on GPU, with a guarded self-test at setup; pg on CPU)
"""

from __future__ import annotations

import logging
import os
import time
from dataclasses import dataclass
from typing import List, Optional, Sequence

import torch
import torch.distributed as dist

from .primitives import Primitive
from .strategy.synthesizer import Synthesizer
from .topology.formats import (
    LogicalGraph,
    ProfileMatrices,
    Strategy,
    load_strategy,
    single_node_graph,
)

log = logging.getLogger("adapcc")


@dataclass
class CommArgs:
    """The launcher flag contract (reference: launcher.py:54-62)."""

    port: int = 18000
    entry_point: int = -1
    strategy_file: str = ""
    logical_graph: str = ""
    parallel_degree: int = 0  # 0 = auto (world_size stars on one node)
    profile_freq: int = 0     # reconstruct_topology every N steps (0 = never)
    policy: str = "par-trees"
    chunk_bytes: int = 2 * 1024 * 1024
    relay: bool = False       # straggler-adaptive active sets via coordinator
    coordinator_port: int = 50051
    # size-adaptive transport: tensors below this many bytes take the RCCL
    # collective (latency-optimal); larger ones take the tree engine
    # (bandwidth-optimal star forest). 0 disables the hybrid.
    # With the single-launch fused small-message kernel (kernels.hip
    # small_fused_kernel) the native engine owns every bucket size: the
    # measured 4-256 KB latency is far below the RCCL collective's on the
    # same plans (profiles/small_lat evidence), so the RCCL bypass is off
    # by default and kept only as an operational escape hatch
    # (ADAPCC_SMALL_THRESHOLD=<bytes> re-enables it).
    small_threshold: int = 0

    def __post_init__(self):
        import os as _os

        env = _os.environ.get("ADAPCC_SMALL_THRESHOLD")
        if env is not None:
            self.small_threshold = int(env)

    @classmethod
    def from_namespace(cls, ns) -> "CommArgs":
        kw = {}
        for f in ("port", "entry_point", "strategy_file", "logical_graph",
                  "parallel_degree", "profile_freq", "policy", "chunk_bytes",
                  "relay", "coordinator_port", "small_threshold"):
            if hasattr(ns, f) and getattr(ns, f) is not None:
                kw[f] = getattr(ns, f)
        return cls(**kw)


class Communicator:
    def __init__(
        self,
        args: CommArgs,
        local_rank: int,
        world_rank: int,
        world_size: int,
        group=None,
    ) -> None:
        self.args = args
        self.local_rank = local_rank
        self.rank = world_rank
        self.world_size = world_size
        self.group = group
        self.engine = None
        self.strategy: Optional[Strategy] = None
        self.graph: Optional[LogicalGraph] = None
        self.profile_mats: Optional[ProfileMatrices] = None
        self.transport = os.environ.get("ADAPCC_TRANSPORT", "auto")
        self.use_gpu = torch.cuda.is_available()
        self.coordinator = None   # rank-0 gRPC server (relay/fault)
        self.controller = None    # per-rank controller thread
        self.hooker = None        # per-rank hook-negotiation client
        self.active_ranks: Optional[List[int]] = None  # None = all
        self.fault_worker_list: List[int] = []
        self.effective_transport: Optional[str] = None
        self._setup_done = False
        # recorded DDP bucket sizes (bytes) — feeds the coordinator's
        # rent-or-buy cost model (reference accumulated_size)
        self._bucket_bytes: List[int] = []

        self.synthesizer = Synthesizer(
            policy=args.policy,
            parallel_degree=args.parallel_degree or max(2, world_size),
            chunk_bytes=args.chunk_bytes,
        )

    # ------------------------------------------------------------------
    # Adaptation flow (reference: adapcc.py:16-41 init sequence)
    # ------------------------------------------------------------------

    def run_entry_point(self) -> None:
        ep = self.args.entry_point
        if ep == int(Primitive.DETECT):
            self.detect_topology()
            self.profile_topology()
            self.synthesize()
        elif ep == int(Primitive.PROFILE):
            self.load_or_default_graph()
            self.profile_topology()
            self.synthesize()
        else:
            if self.args.strategy_file and os.path.exists(self.args.strategy_file):
                self.strategy = load_strategy(self.args.strategy_file)
            else:
                self.load_or_default_graph()
                self.synthesize()

    def load_or_default_graph(self) -> None:
        if self.args.logical_graph and os.path.exists(self.args.logical_graph):
            from .topology.formats import load_logical_graph

            self.graph = load_logical_graph(self.args.logical_graph)
        else:
            self.graph = single_node_graph(self.world_size)

    def detect_topology(self) -> None:
        from .topology.detect import detect_node_topology

        self.graph = detect_node_topology(
            self.rank, self.local_rank, self.world_size, group=self.group
        )

    def profile_topology(self) -> None:
        from .topology.profile import profile_links

        t0 = time.time()
        self.profile_mats = profile_links(
            self.rank, self.world_size, self.graph, group=self.group
        )
        log.info("profiling time: %.3f s", time.time() - t0)

    def synthesize(self) -> None:
        if self.graph is None:
            self.load_or_default_graph()
        self.strategy = self.synthesizer.generate_strategy(
            graph=self.graph, profile=self.profile_mats
        )

    # ------------------------------------------------------------------
    # Engine lifecycle (reference: commu.py:301-352 init/exit_threads)
    # ------------------------------------------------------------------

    def setup(self, primitive: Primitive = Primitive.ALLREDUCE) -> None:
        if self.strategy is None:
            self.run_entry_point()
        t0 = time.time()
        transport = self.transport
        multi_node = self.graph is not None and len(self.graph.servers) > 1
        if transport == "auto":
            transport = "native" if (self.use_gpu and not multi_node) else "pg"
        if transport == "native" and multi_node:
            log.warning("native engine is single-node; multi-node runs use "
                        "the process-group transport")

        if transport == "native":
            try:
                self._setup_native()
            except Exception as e:
                if self.transport != "auto":
                    raise
                log.error("NATIVE ENGINE SETUP FAILED (%s); falling back to "
                          "the process-group transport — performance and "
                          "the custom data plane are NOT in effect", e)
                self._setup_pg()
        elif transport == "pg":
            self._setup_pg()
        elif transport == "p2p":
            self._setup_p2p()
        else:
            raise ValueError(f"unknown ADAPCC_TRANSPORT={transport}")
        if self.args.relay and self.world_size > 1:
            self._start_relay_control()
        self._setup_done = True
        log.info("[Rank %d] transmission context setup time: %.1f ms",
                 self.rank, 1000 * (time.time() - t0))

    # ------------------------------------------------------------------
    # Relay control / fault detection (reference: commu.py:81-97, proto/)
    # ------------------------------------------------------------------

    def coordinator_address(self) -> str:
        host = os.environ.get("MASTER_ADDR", "127.0.0.1")
        return f"{host}:{self.args.coordinator_port}"

    def _start_relay_control(self) -> None:
        from .coordinator.client import Controller, Hooker
        from .coordinator.server import CoordinatorServer

        if self.rank == 0 and self.coordinator is None:
            self.coordinator = CoordinatorServer(
                self.world_size, port=self.args.coordinator_port).start()
        if dist.is_initialized():
            dist.barrier(group=self.group)  # server up before clients dial
        addr = self.coordinator_address()
        if self.hooker is None:
            self.hooker = Hooker(addr, self.rank)
        if self.controller is None:
            self.controller = Controller(
                addr, self.rank,
                on_active=self._on_controller_active,
                on_fault=self._on_fault,
            )

    def _on_controller_active(self, active: Optional[List[int]]) -> None:
        pass  # the hook-side negotiation governs the active set

    def _on_fault(self, dead: List[int]) -> None:
        self.fault_worker_list = dead
        log.error("[Rank %d] fault_worker_list=%s — restart or "
                  "reconstruct_topology with the surviving ranks",
                  self.rank, dead)

    def notify_hook_ready(self, step: int) -> None:
        """First DDP bucket of a step: negotiate the straggler-adaptive
        active set (reference: commu.py:387-394 + rpc_server hook_fetch).
        Negotiation latency is tracked per call (reference measured
        ~0.8-1.9 ms per step, proto/latency_0.0.txt)."""
        if self.hooker is None:
            return
        from .utils.metrics import GLOBAL as metrics

        metrics.timer_start("relay_negotiation")
        # feed the rent-or-buy cost model: expected step payload (recorded
        # DDP bucket sizes once stable) and the profiled link bandwidth
        comm_bytes = float(sum(self._bucket_bytes)) if self._bucket_bytes \
            else 0.0
        comm_bw = self._mean_link_bw_Bps()
        active = self.hooker.send_ready_request(
            step, comm_bytes=comm_bytes, comm_bw=comm_bw)
        metrics.timer_stop("relay_negotiation")
        metrics.inc("relay_negotiations")
        self.active_ranks = (
            None if len(active) >= self.world_size else active
        )

    def _mean_link_bw_Bps(self) -> float:
        """Mean profiled link bandwidth in B/s (0 when unprofiled) — the
        rent-or-buy cost model's accumulated_bandwidth analog."""
        if self.profile_mats is None or not self.profile_mats.bandwidth:
            return 0.0
        vals = list(self.profile_mats.bandwidth.values())
        return sum(vals) / len(vals) * 1e9

    def _setup_native(self) -> None:
        from .runtime.engine import NativeEngine

        # Phase 1: local construction. Failures here (e.g. not enough
        # devices) must be agreed on COLLECTIVELY before any engine
        # collective runs, or ranks desynchronize into a cross-transport
        # deadlock (one rank in the native bootstrap, another fallen back).
        ndev = max(1, torch.cuda.device_count())
        dev = self.local_rank % ndev
        err: Optional[Exception] = None
        engine = None
        try:
            torch.cuda.set_device(dev)
            engine = NativeEngine(self.rank, self.world_size, device=dev)
        except Exception as e:  # noqa: BLE001 - agreed on below
            err = e
        if self.world_size > 1 and dist.is_initialized():
            oks: List[Optional[bool]] = [None] * self.world_size
            dist.all_gather_object(oks, err is None, group=self.group)
            if not all(oks):
                del engine
                raise RuntimeError(
                    f"native engine unavailable on some rank "
                    f"(local error: {err})")
        elif err is not None:
            raise err
        # Phase 2: collective bootstrap + strategy + canary.
        engine.bootstrap(group=self.group)
        engine.set_strategy(self.strategy)
        engine.self_test()
        self.engine = engine
        self.effective_transport = "native"

    def _setup_p2p(self) -> None:
        from .runtime.p2p_engine import P2PTreeEngine

        self.engine = P2PTreeEngine(self.rank, self.world_size,
                                    group=self.group)
        self.engine.bootstrap(group=self.group)
        self.engine.set_strategy(self.strategy)
        self.engine.self_test()
        self.effective_transport = "p2p"

    def _setup_pg(self) -> None:
        from .runtime.fallback import ProcessGroupEngine

        self.engine = ProcessGroupEngine(self.rank, self.world_size,
                                         group=self.group)
        self.engine.set_strategy(self.strategy)
        self.effective_transport = "pg"

    def ensure_setup(self) -> None:
        if not self._setup_done:
            self.setup()

    # ------------------------------------------------------------------
    # Collectives (reference: commu.py:360-379)
    # ------------------------------------------------------------------

    def all_reduce(self, tensor: torch.Tensor,
                   active: Optional[Sequence[int]] = None,
                   average: bool = False) -> torch.Tensor:
        self.ensure_setup()
        if active is None:
            active = self.active_ranks
        from .utils.metrics import GLOBAL as metrics

        metrics.inc("allreduce_calls")
        nbytes = tensor.numel() * tensor.element_size()
        metrics.inc("allreduce_bytes", nbytes)
        if active is not None:
            metrics.inc("allreduce_relay_calls")
        # size-adaptive transport: latency-bound small tensors go straight
        # to the RCCL/gloo collective; the tree engine owns the
        # bandwidth-bound regime
        if (
            self.args.small_threshold
            and nbytes < self.args.small_threshold
            and self.effective_transport in ("native", "p2p")
            and self.world_size > 1
            and dist.is_initialized()
        ):
            metrics.inc("allreduce_small_pg_calls")
            n = self.world_size
            if active is not None and len(active) > 0 and                     len(set(active)) < self.world_size:
                n = len(set(active))
                if self.rank not in set(active):
                    tensor.zero_()
            dist.all_reduce(tensor, group=self.group)
            if average:
                tensor.div_(n)
            return tensor
        return self.engine.all_reduce(tensor, active=active, average=average)

    def stats(self) -> dict:
        from .utils.metrics import GLOBAL as metrics

        return metrics.snapshot()

    def synchronize(self) -> None:
        if self.engine is not None:
            self.engine.synchronize()

    # Primitive dispatch: the native engine implements all six primitives;
    # transports without a native method (pg fallback, p2p tier off the
    # allreduce path) route through torch.distributed.

    def reduce(self, tensor: torch.Tensor, root: int = 0,
               active: Optional[Sequence[int]] = None) -> torch.Tensor:
        self.ensure_setup()
        if hasattr(self.engine, "reduce"):
            return self.engine.reduce(tensor, root=root, active=active)
        if active is not None and len(active) > 0 and self.rank not in set(active):
            tensor.zero_()
        dist.reduce(tensor, dst=root, group=self.group)
        return tensor

    def broadcast(self, tensor: torch.Tensor, root: int = 0) -> torch.Tensor:
        self.ensure_setup()
        if hasattr(self.engine, "broadcast"):
            return self.engine.broadcast(tensor, root=root)
        dist.broadcast(tensor, src=root, group=self.group)
        return tensor

    def all_gather(self, out: torch.Tensor, tensor: torch.Tensor) -> torch.Tensor:
        self.ensure_setup()
        if hasattr(self.engine, "all_gather"):
            return self.engine.all_gather(out, tensor)
        dist.all_gather_into_tensor(out, tensor, group=self.group)
        return out

    def reduce_scatter(self, out: torch.Tensor, tensor: torch.Tensor) -> torch.Tensor:
        self.ensure_setup()
        if hasattr(self.engine, "reduce_scatter"):
            return self.engine.reduce_scatter(out, tensor)
        dist.reduce_scatter_tensor(out, tensor, group=self.group)
        return out

    def all_to_all(self, out: torch.Tensor, tensor: torch.Tensor) -> torch.Tensor:
        self.ensure_setup()
        if hasattr(self.engine, "all_to_all"):
            return self.engine.all_to_all(out, tensor)
        dist.all_to_all_single(out, tensor, group=self.group)
        return out

    # ------------------------------------------------------------------
    # On-the-fly re-adaptation (reference: adapcc.py:63-67)
    # ------------------------------------------------------------------

    def reconstruct_topology(self) -> None:
        self.clear(keep_coordinator=True)
        self.profile_topology()
        self.synthesize()
        self.setup()

    def update_relay(self, step: int) -> None:
        """Feed the controller a new step (relay negotiation happens in the
        background controller thread; see coordinator/)."""
        if self.controller is not None:
            self.controller.submit_step(step)

    def clear(self, keep_coordinator: bool = False) -> None:
        if self.engine is not None:
            try:
                self.engine.synchronize()
            except Exception:
                pass
            self.engine = None
            # all ranks drop their hipIpc imports/exports before anyone
            # rebuilds (reconstruct_topology) — dmabuf refcounts make the
            # teardown order safe, the barrier makes it deterministic
            if dist.is_initialized() and self.world_size > 1:
                dist.barrier(group=self.group)
        self._setup_done = False
        if not keep_coordinator:
            if self.controller is not None:
                self.controller.stop()
                self.controller = None
            if self.coordinator is not None:
                self.coordinator.stop()
                self.coordinator = None
