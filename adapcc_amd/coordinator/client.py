"""Coordinator clients (reference: proto/rpc_client.py + commu.py threads).

- ``Hooker``: called by the DDP hook on the first bucket of a step; fetches
  the straggler-adaptive active set (rent-or-buy on the server side).
- ``Controller``: background thread fed steps via ``submit_step``; sends a
  heartbeat per step and learns the fault status. On a fault
  (status=0) it records the dead ranks and stops — communication itself is
  never blocked (reference commu.py:152-157 hang-free semantics).
"""

from __future__ import annotations

import logging
import queue
import threading
from typing import Callable, List, Optional

import grpc
import msgpack

from .server import SERVICE

log = logging.getLogger("adapcc.coordinator")


def _stub(channel, method: str):
    return channel.unary_unary(
        f"/{SERVICE}/{method}",
        request_serializer=msgpack.packb,
        response_deserializer=msgpack.unpackb,
    )


class Hooker:
    def __init__(self, address: str, rank: int) -> None:
        self.rank = rank
        self._channel = grpc.insecure_channel(address)
        self._hook_fetch = _stub(self._channel, "hook_fetch")

    def send_ready_request(self, step: int, timeout: float = 30.0,
                           comm_bytes: float = 0.0,
                           comm_bw: float = 0.0) -> List[int]:
        """Fetch the active set; optionally feed the server's rent-or-buy
        cost model with this step's expected collective size (bytes) and
        the measured link bandwidth (B/s)."""
        req = {"step": step, "rank": self.rank}
        if comm_bytes > 0 and comm_bw > 0:
            req["size"] = comm_bytes
            req["bw"] = comm_bw
        resp = self._hook_fetch(req, timeout=timeout)
        return list(resp["active"])

    def close(self) -> None:
        self._channel.close()


class Controller:
    """Per-rank controller thread (reference: commu.py:143-170)."""

    def __init__(
        self,
        address: str,
        rank: int,
        on_active: Callable[[Optional[List[int]]], None],
        on_fault: Callable[[List[int]], None],
    ) -> None:
        self.rank = rank
        self._channel = grpc.insecure_channel(address)
        self._controller_fetch = _stub(self._channel, "controller_fetch")
        self._on_active = on_active
        self._on_fault = on_fault
        self._steps: "queue.Queue[Optional[int]]" = queue.Queue()
        self.fault_worker_list: List[int] = []
        self._thread = threading.Thread(target=self._run, daemon=True)
        self._thread.start()

    def submit_step(self, step: int) -> None:
        self._steps.put(step)

    def _run(self) -> None:
        while True:
            step = self._steps.get()
            if step is None:
                return
            try:
                resp = self._controller_fetch(
                    {"step": step, "rank": self.rank}, timeout=60.0)
            except grpc.RpcError as e:  # coordinator gone
                log.warning("controller rpc failed: %s", e)
                return
            if resp["status"] == 0:
                alive = list(resp["active"])
                world = max(alive + [self.rank]) + 1
                self.fault_worker_list = [
                    r for r in range(world) if r not in alive
                ]
                log.error("[Rank %d] fault detected; dead ranks: %s",
                          self.rank, self.fault_worker_list)
                self._on_fault(self.fault_worker_list)
                return
            self._on_active(list(resp["active"]))

    def stop(self) -> None:
        self._steps.put(None)
        self._thread.join(timeout=5)
        self._channel.close()
