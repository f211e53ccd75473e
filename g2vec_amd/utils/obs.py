"""Observability: phase timers + structured JSONL metrics.

The reference's only tracing is wall-clock prints around training epochs
(G2Vec.py:260-278); this module adds per-phase timers for all 7 pipeline
steps and a JSONL sink."""
from __future__ import annotations

import json
import time
from contextlib import contextmanager
from typing import Dict, Optional


class JsonlLogger:
    def __init__(self, path: str = ""):
        self.path = path
        self._f = open(path, "a") if path else None

    def emit(self, event: str, **fields) -> None:
        if self._f is None:
            return
        rec = {"ts": time.time(), "event": event}
        rec.update(fields)
        self._f.write(json.dumps(rec) + "\n")
        self._f.flush()

    def close(self) -> None:
        if self._f is not None:
            self._f.close()
            self._f = None


class PhaseTimers:
    def __init__(self, logger: Optional[JsonlLogger] = None):
        self.times: Dict[str, float] = {}
        self.logger = logger

    @contextmanager
    def phase(self, name: str):
        t0 = time.perf_counter()
        try:
            yield
        finally:
            dt = time.perf_counter() - t0
            self.times[name] = self.times.get(name, 0.0) + dt
            if self.logger:
                self.logger.emit("phase", name=name, seconds=dt)

    def summary(self) -> Dict[str, float]:
        return dict(self.times)
