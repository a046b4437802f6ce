"""Central coordinator service (reference: proto/rpc_server.py).

Runs on rank 0. Two RPCs, same semantics as the reference:

- ``hook_fetch(step, rank)``: straggler-adaptive active-set negotiation.
  The first worker to arrive for a step runs a rent-or-buy slot loop
  (slot = 5 ms, threshold = 100 ms, reference rpc_server.py:71-88): keep
  "renting" (waiting a slot for more arrivals) while the accumulated wait
  is below the relay threshold, then "buy" — close the active set with
  whoever has arrived. Later workers get the snapshot.

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
    ) -> None:
        self.world_size = world_size
        self.time_slot = time_slot
        self.relay_threshold = relay_threshold
        self.fault_timeout = fault_timeout
        self._lock = threading.Condition()
        # hook negotiation state per step
        self._hook_arrived: Dict[int, Set[int]] = {}
        self._hook_snapshot: Dict[int, List[int]] = {}
        # controller heartbeats per step
        self._beats: Dict[int, Set[int]] = {}

    # ------------------------------------------------------------------

    def hook_fetch(self, request: dict, context=None) -> dict:
        step = int(request["step"])
        rank = int(request["rank"])
        with self._lock:
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

        # first arriver: rent-or-buy slot loop
        waited = 0.0
        while True:
            with self._lock:
                arrived = self._hook_arrived[step]
                if len(arrived) >= self.world_size:
                    break
            if waited >= self.relay_threshold:
                break
            time.sleep(self.time_slot)
            waited += self.time_slot
        with self._lock:
            active = sorted(self._hook_arrived[step])
            self._hook_snapshot[step] = active
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
