"""Thread Coordinator — capability C14 of SURVEY.md §2b.

Mirrors ``tf.train.Coordinator`` / queue-runner semantics: background
threads (input prefetch, async PS shard appliers, heartbeats) register
with a coordinator; any exception requests a clean stop of all of them;
``join`` re-raises the first exception.
"""

from __future__ import annotations

import threading
import traceback
from typing import Callable, List, Optional

__all__ = ["Coordinator", "LooseVersionThread"]


class Coordinator:
    def __init__(self):
        self._stop_event = threading.Event()
        self._lock = threading.Lock()
        self._exc: Optional[BaseException] = None
        self._threads: List[threading.Thread] = []

    def register_thread(self, t: threading.Thread) -> None:
        with self._lock:
            self._threads.append(t)

    def should_stop(self) -> bool:
        return self._stop_event.is_set()

    def wait_for_stop(self, timeout: Optional[float] = None) -> bool:
        return self._stop_event.wait(timeout)

    def request_stop(self, ex: Optional[BaseException] = None) -> None:
        with self._lock:
            if ex is not None and self._exc is None:
                self._exc = ex
        self._stop_event.set()

    def stop_on_exception(self):
        """Context manager for thread bodies (mirrors TF's)."""
        coord = self

        class _Ctx:
            def __enter__(self):
                return self

            def __exit__(self, exc_type, exc, tb):
                if exc is not None:
                    traceback.print_exc()
                    coord.request_stop(exc)
                    return True  # swallow; surfaced in join()
                return False

        return _Ctx()

    def join(self, threads: Optional[List[threading.Thread]] = None,
             stop_grace_period_secs: float = 120.0) -> None:
        threads = list(threads or []) + self._threads
        self._stop_event.wait()
        for t in threads:
            t.join(stop_grace_period_secs)
        with self._lock:
            if self._exc is not None:
                raise self._exc

    def raise_requested_exception(self) -> None:
        with self._lock:
            if self._exc is not None:
                raise self._exc


class LooseVersionThread(threading.Thread):
    """A coordinator-aware looping thread (the queue-runner equivalent):
    runs ``fn()`` repeatedly until the coordinator stops."""

    def __init__(self, coord: Coordinator, fn: Callable[[], None],
                 name: str = "rocps-runner"):
        super().__init__(name=name, daemon=True)
        self._coord = coord
        self._fn = fn
        coord.register_thread(self)

    def run(self) -> None:
        with self._coord.stop_on_exception():
            while not self._coord.should_stop():
                self._fn()
