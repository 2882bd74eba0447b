"""Observability — SURVEY.md §5 (tracing / metrics / summaries).

The reference logs scalar summaries to TF event files and steps/sec via
StepCounterHook; tracing comes from RunMetadata Chrome traces. Rebuild:
JSONL metrics (step, loss, lr, examples/sec, per-phase ms), an
in-engine step timer using HIP events on the compute/comm streams, and
a Chrome-trace JSON dump for ``--trace`` runs.
"""

from __future__ import annotations

import json
import os
import time
from typing import Any, Dict, List, Optional

import torch

__all__ = ["MetricsLogger", "StepTimer", "chrome_trace"]


class MetricsLogger:
    """Append-only JSONL scalar log (the event-file equivalent)."""

    def __init__(self, logdir: Optional[str] = None, filename: str = "metrics.jsonl",
                 stdout_every: int = 0):
        self._f = None
        if logdir:
            os.makedirs(logdir, exist_ok=True)
            self._f = open(os.path.join(logdir, filename), "a", buffering=1)
        self._stdout_every = stdout_every
        self._n = 0

    def log(self, **scalars: Any) -> None:
        rec = {"t": round(time.time(), 3), **scalars}
        line = json.dumps(rec)
        if self._f is not None:
            self._f.write(line + "\n")
        self._n += 1
        if self._stdout_every and self._n % self._stdout_every == 0:
            print("[rocps metrics] " + line, flush=True)

    def close(self) -> None:
        if self._f is not None:
            self._f.close()
            self._f = None


class StepTimer:
    """Wall + per-phase timing; HIP-event based on GPU so phases measure
    device time without host syncs until read."""

    def __init__(self, use_cuda: Optional[bool] = None):
        self.use_cuda = torch.cuda.is_available() if use_cuda is None else use_cuda
        self._events: Dict[str, List] = {}
        self._wall: Dict[str, float] = {}
        self._t0: Dict[str, float] = {}

    def start(self, phase: str) -> None:
        if self.use_cuda:
            e = torch.cuda.Event(enable_timing=True)
            e.record()
            self._events.setdefault(phase, []).append([e, None])
        self._t0[phase] = time.perf_counter()

    def stop(self, phase: str) -> None:
        if self.use_cuda and self._events.get(phase):
            e = torch.cuda.Event(enable_timing=True)
            e.record()
            self._events[phase][-1][1] = e
        self._wall[phase] = (self._wall.get(phase, 0.0)
                             + time.perf_counter() - self._t0[phase])

    def device_ms(self, phase: str) -> float:
        if not self.use_cuda:
            return self._wall.get(phase, 0.0) * 1e3
        torch.cuda.synchronize()
        return sum(a.elapsed_time(b) for a, b in self._events.get(phase, [])
                   if b is not None)

    def wall_ms(self, phase: str) -> float:
        return self._wall.get(phase, 0.0) * 1e3

    def reset(self) -> None:
        self._events.clear()
        self._wall.clear()


def chrome_trace(records: List[Dict[str, Any]], path: str) -> None:
    """records: [{"name", "ts" (us), "dur" (us), "tid"}] -> Chrome JSON."""
    events = [{"name": r["name"], "ph": "X", "ts": r["ts"], "dur": r["dur"],
               "pid": 0, "tid": r.get("tid", 0)} for r in records]
    with open(path, "w") as f:
        json.dump({"traceEvents": events}, f)
