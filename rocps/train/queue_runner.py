"""QueueRunner — capability C14's input-feeding half.

The reference feeds batches through TF input queues driven by
``tf.train.QueueRunner`` threads started with
``tf.train.start_queue_runners(sess, coord)`` and stopped through the
``Coordinator`` (TF python/training/queue_runner_impl.py semantics:
N enqueue threads per runner, exceptions routed to the coordinator,
``queue.close()`` on stop).  Eager recast: a bounded prefetch queue fed
by background threads calling a user batch function; the training loop
pulls ready batches (host→device copy can overlap compute since the
producer runs ahead).

MI355X note: batches destined for the GPU can be produced into pinned
host memory (``pin_memory=True``) so the subsequent ``.to("cuda",
non_blocking=True)`` is a true DMA on the copy engine.
"""

from __future__ import annotations

import queue
import threading
from typing import Any, Callable, List, Optional

from rocps.train.coordinator import Coordinator

__all__ = ["QueueRunner", "start_queue_runners"]


class _Closed:
    pass


_CLOSED = _Closed()


class QueueRunner:
    """N producer threads filling a bounded queue (TF QueueRunner).

    ``batch_fn(index)`` returns one item; ``index`` is a monotonically
    increasing per-runner counter (use it to seed/shard deterministic
    synthetic data).  Exceptions in producers request coordinator stop
    (the reference routes enqueue errors the same way) and close the
    queue so a blocked consumer wakes up.
    """

    def __init__(self, batch_fn: Callable[[int], Any], capacity: int = 4,
                 num_threads: int = 1, pin_memory: bool = False):
        self.batch_fn = batch_fn
        self.capacity = capacity
        self.num_threads = num_threads
        self.pin_memory = pin_memory
        self._q: "queue.Queue[Any]" = queue.Queue(maxsize=capacity)
        self._threads: List[threading.Thread] = []
        self._idx = 0
        self._idx_mu = threading.Lock()
        self._coord: Optional[Coordinator] = None

    # -- producer side ---------------------------------------------------
    def _next_index(self) -> int:
        with self._idx_mu:
            i = self._idx
            self._idx += 1
            return i

    def _run(self, coord: Coordinator) -> None:
        try:
            while not coord.should_stop():
                item = self.batch_fn(self._next_index())
                if self.pin_memory:
                    item = _pin(item)
                while not coord.should_stop():
                    try:
                        self._q.put(item, timeout=0.1)
                        break
                    except queue.Full:
                        continue
        except BaseException as e:  # noqa: BLE001 — routed to coordinator
            coord.request_stop(e)
        finally:
            # close so a blocked dequeue wakes (queue.close() semantics)
            try:
                self._q.put_nowait(_CLOSED)
            except queue.Full:
                pass

    def create_threads(self, coord: Coordinator,
                       start: bool = False) -> List[threading.Thread]:
        self._coord = coord
        ts = [threading.Thread(target=self._run, args=(coord,), daemon=True,
                               name=f"rocps-queue-runner-{i}")
              for i in range(self.num_threads)]
        self._threads += ts
        for t in ts:
            coord.register_thread(t)
            if start:
                t.start()
        return ts

    # -- consumer side ---------------------------------------------------
    def dequeue(self, timeout: float = 60.0) -> Any:
        """Next ready batch; raises RuntimeError once stopped/closed."""
        deadline = None
        while True:
            try:
                item = self._q.get(timeout=min(timeout, 0.5))
            except queue.Empty:
                if self._coord is not None and self._coord.should_stop():
                    raise RuntimeError("queue runner stopped")
                if deadline is None:
                    import time
                    deadline = time.time() + timeout
                else:
                    import time
                    if time.time() > deadline:
                        raise TimeoutError("dequeue timed out")
                continue
            if isinstance(item, _Closed):
                if self._coord is not None and self._coord.should_stop():
                    raise RuntimeError("queue runner stopped")
                continue
            return item


def _pin(item):
    import torch

    if isinstance(item, torch.Tensor):
        return item.pin_memory() if not item.is_cuda else item
    if isinstance(item, (tuple, list)):
        return type(item)(_pin(x) for x in item)
    return item


def start_queue_runners(runners: List[QueueRunner],
                        coord: Optional[Coordinator] = None) -> Coordinator:
    """tf.train.start_queue_runners: spawn every runner's threads."""
    coord = coord or Coordinator()
    for r in runners:
        r.create_threads(coord, start=True)
    return coord
