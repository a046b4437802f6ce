"""Central coordinator service (reference: proto/rpc_server.py).

Runs on rank 0. Two RPCs, same semantics as the reference:

- ``hook_fetch(step, rank)``: straggler-adaptive active-set negotiation.
  The first worker to arrive for a step runs a rent-or-buy slot loop
  (slot = 5 ms, threshold = 100 ms, reference rpc_server.py:64-96) with
  the reference's online cost model: per slot, compare the accumulated
  "rent" (time already spent waiting, plus the full-world collective that
  waiting would buy) against the "buy" cost (a collective over only the m
  ready ranks plus the relay-forwarding term), and close the active set
  as soon as waiting is provably uneconomical — not only at the fixed
  threshold. The collective-cost terms come from the workload's bucket
  size and measured bandwidth (EMA-updated from ``size``/``bw`` fields
  the hook clients may attach). Later workers get the snapshot.

- ``controller_fetch(step, rank)``: fault detection. Blocks until all
  world_size heartbeats for the step arrive, or ``fault_tolerant_time``
  (10 s) elapses -> returns the alive subset with status=0
  (reference rpc_server.py:48-62).

Transport: gRPC with msgpack-serialized dicts (the reference used protoc
stubs; msgpack avoids a codegen step with identical wire semantics).
"""

from __future__ import annotations

import threading
import time
from concurrent import futures
from typing import Dict, List, Set

import grpc
import msgpack

SERVICE = "adapcc.Coordinator"

TIME_SLOT_DURATION = 0.005   # reference: time_slot_duration = 5 ms
RELAY_THRESHOLD = 0.1        # reference: relay_threshold = 0.1 s
FAULT_TOLERANT_TIME = 10.0   # reference: fault_tolerant_time = 10 s


class CoordinatorServicer:
    def __init__(
        self,
        world_size: int,
        time_slot: float = TIME_SLOT_DURATION,
        relay_threshold: float = RELAY_THRESHOLD,
        fault_timeout: float = FAULT_TOLERANT_TIME,
        comm_bytes: float = 0.0,
        comm_bw: float = 0.0,
    ) -> None:
        self.world_size = world_size
        self.time_slot = time_slot
        self.relay_threshold = relay_threshold
        self.fault_timeout = fault_timeout
        # Rent-or-buy cost-model inputs (reference rpc_server.py:30-31
        # accumulated_size / accumulated_bandwidth). Zero disables the
        # cost model and leaves only the threshold cut-off. Updated by
        # EMA from hook requests carrying "size" (bytes) / "bw" (B/s).
        self.comm_bytes = float(comm_bytes)
        self.comm_bw = float(comm_bw)
        self._lock = threading.Condition()
        # hook negotiation state per step
        self._hook_arrived: Dict[int, Set[int]] = {}
        self._hook_snapshot: Dict[int, List[int]] = {}
        # highest step whose active set has been closed: late requests at
        # or below this step must NOT re-open negotiation (a >16-step
        # straggler would otherwise self-negotiate a singleton set that
        # diverges from its peers' snapshot).
        self._closed_step = -1
        # controller heartbeats per step
        self._beats: Dict[int, Set[int]] = {}

    # ------------------------------------------------------------------

    def _rent_or_buy_done(self, waited: float, num_ready: int) -> bool:
        """Return True when the first arriver should close the set.

        Online rent-or-buy (reference rpc_server.py:71-88): ``rent`` =
        keep waiting for the full world, paying the wait so far plus the
        full-world allreduce 2(n-1)/n * S/BW; ``buy`` = close now with m
        ready ranks, paying the m-rank allreduce scaled by
        ((m-1)/m)/((n-1)/n) plus the n*S/BW relay-forwarding term.
        """
        n = self.world_size
        if num_ready >= n:
            return True
        if waited >= self.relay_threshold:
            return True
        if num_ready > 1 and self.comm_bytes > 0 and self.comm_bw > 0:
            rent_collective = 2.0 * (n - 1) * self.comm_bytes / self.comm_bw
            co_n = (n - 1) / n
            co_m = (num_ready - 1) / num_ready
            buy_cost = (rent_collective * (co_m / co_n)
                        + n * self.comm_bytes / self.comm_bw)
            if waited + rent_collective >= buy_cost:
                return True
        return False

    # ------------------------------------------------------------------

    def hook_fetch(self, request: dict, context=None) -> dict:
        step = int(request["step"])
        rank = int(request["rank"])
        with self._lock:
            if "size" in request and "bw" in request:
                sz, bw = float(request["size"]), float(request["bw"])
                if sz > 0 and bw > 0:
                    a = 0.25  # EMA so one noisy probe can't swing the model
                    self.comm_bytes = (sz if self.comm_bytes == 0
                                       else (1 - a) * self.comm_bytes + a * sz)
                    self.comm_bw = (bw if self.comm_bw == 0
                                    else (1 - a) * self.comm_bw + a * bw)
            if step <= self._closed_step:
                # Step already closed: a >16-step-late straggler must get
                # the recorded snapshot (or a loud fault status), never
                # re-open negotiation with a singleton set.
                snap = self._hook_snapshot.get(step)
                if snap is not None:
                    return {"step": step, "active": list(snap), "status": 1}
                return {"step": step, "active": [], "status": 0}
            first = step not in self._hook_arrived
            self._hook_arrived.setdefault(step, set()).add(rank)
            self._lock.notify_all()
            if not first:
                # late worker: wait for the snapshot then return it
                deadline = time.monotonic() + self.fault_timeout
                while step not in self._hook_snapshot:
                    remaining = deadline - time.monotonic()
                    if remaining <= 0 or not self._lock.wait(timeout=remaining):
                        break
                active = self._hook_snapshot.get(
                    step, sorted(self._hook_arrived[step]))
                return {"step": step, "active": list(active), "status": 1}

        # first arriver: rent-or-buy slot loop with online cost model
        waited = 0.0
        while True:
            with self._lock:
                num_ready = len(self._hook_arrived[step])
            if self._rent_or_buy_done(waited, num_ready):
                break
            time.sleep(self.time_slot)
            waited += self.time_slot
        with self._lock:
            active = sorted(self._hook_arrived[step])
            self._hook_snapshot[step] = active
            self._closed_step = max(self._closed_step, step)
            self._lock.notify_all()
            self._gc(step)
        return {"step": step, "active": active, "status": 1}

    def controller_fetch(self, request: dict, context=None) -> dict:
        step = int(request["step"])
        rank = int(request["rank"])
        deadline = time.monotonic() + self.fault_timeout
        with self._lock:
            self._beats.setdefault(step, set()).add(rank)
            self._lock.notify_all()
            while len(self._beats[step]) < self.world_size:
                remaining = deadline - time.monotonic()
                if remaining <= 0:
                    alive = sorted(self._beats[step])
                    return {"step": step, "active": alive, "status": 0}
                self._lock.wait(timeout=remaining)
            return {"step": step, "active": sorted(self._beats[step]),
                    "status": 1}

    def _gc(self, step: int, keep: int = 16) -> None:
        for d in (self._hook_arrived, self._hook_snapshot, self._beats):
            for s in [s for s in d if s < step - keep]:
                del d[s]


class CoordinatorServer:
    """gRPC server wrapper; start on rank 0 only."""

    def __init__(self, world_size: int, port: int = 50051, **kw) -> None:
        self.servicer = CoordinatorServicer(world_size, **kw)
        self.port = port
        self._server = grpc.server(futures.ThreadPoolExecutor(max_workers=32))
        handlers = {
            "hook_fetch": grpc.unary_unary_rpc_method_handler(
                self.servicer.hook_fetch,
                request_deserializer=msgpack.unpackb,
                response_serializer=msgpack.packb,
            ),
            "controller_fetch": grpc.unary_unary_rpc_method_handler(
                self.servicer.controller_fetch,
                request_deserializer=msgpack.unpackb,
                response_serializer=msgpack.packb,
            ),
        }
        self._server.add_generic_rpc_handlers(
            (grpc.method_handlers_generic_handler(SERVICE, handlers),)
        )
        self.bound_port = self._server.add_insecure_port(f"0.0.0.0:{port}")

    def start(self) -> "CoordinatorServer":
        self._server.start()
        return self

    def stop(self, grace: float = 0.5) -> None:
        self._server.stop(grace)
