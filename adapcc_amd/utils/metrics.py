"""Lightweight metrics/observability (the reference had only
rank-tagged printf logging; this adds structured counters kept per process
and dumped on demand)."""

from __future__ import annotations

import json
import time
from collections import defaultdict
from typing import Dict


class Metrics:
    def __init__(self) -> None:
        self.counters: Dict[str, float] = defaultdict(float)
        self.timers: Dict[str, float] = defaultdict(float)
        self._t0: Dict[str, float] = {}

    def inc(self, name: str, value: float = 1.0) -> None:
        self.counters[name] += value

    def timer_start(self, name: str) -> None:
        self._t0[name] = time.perf_counter()

    def timer_stop(self, name: str) -> None:
        if name in self._t0:
            self.timers[name] += time.perf_counter() - self._t0.pop(name)

    def snapshot(self) -> Dict[str, float]:
        out = dict(self.counters)
        out.update({f"{k}_s": v for k, v in self.timers.items()})
        return out

    def dump(self, path: str = "", rank: int = 0) -> str:
        blob = json.dumps({"rank": rank, **self.snapshot()}, sort_keys=True)
        if path:
            with open(path, "a") as f:
                f.write(blob + "\n")
        return blob


GLOBAL = Metrics()
