"""Session-run hooks — capability C13 of SURVEY.md §2b.

Mirrors the reference's ``tf.train.SessionRunHook`` family
(``basic_session_run_hooks.py``): the same five-phase lifecycle
(begin / after_create_session / before_run / after_run / end) and the
standard hooks the reference wires through MonitoredTrainingSession.
"""

from __future__ import annotations

import math
import time

import torch
from typing import Any, Dict, Optional

__all__ = ["SessionRunHook", "StopAtStepHook", "StepCounterHook",
           "NanTensorHook", "CheckpointSaverHook", "SummarySaverHook",
           "LoggingTensorHook", "ProfilerHook"]


class RunContext:
    """Passed to before_run/after_run; mirrors tf's SessionRunContext."""

    def __init__(self, session: "Any"):
        self.session = session
        self._stop_requested = False

    @property
    def global_step(self) -> int:
        return self.session.global_step

    def request_stop(self) -> None:
        self._stop_requested = True


class RunValues:
    """after_run payload: the step's loss (+ anything run() returned)."""

    def __init__(self, results: Any):
        self.results = results


class SessionRunHook:
    def begin(self) -> None: ...
    def after_create_session(self, session: Any) -> None: ...
    def before_run(self, run_context: RunContext) -> None: ...
    def after_run(self, run_context: RunContext, run_values: RunValues) -> None: ...
    def end(self, session: Any) -> None: ...


class StopAtStepHook(SessionRunHook):
    """Stop after global_step reaches last_step (or num_steps more)."""

    def __init__(self, num_steps: Optional[int] = None,
                 last_step: Optional[int] = None):
        if (num_steps is None) == (last_step is None):
            raise ValueError("exactly one of num_steps / last_step")
        self._num_steps = num_steps
        self._last_step = last_step

    def after_create_session(self, session) -> None:
        if self._last_step is None:
            self._last_step = session.global_step + self._num_steps

    def after_run(self, run_context, run_values) -> None:
        if run_context.global_step >= self._last_step:
            run_context.request_stop()


class StepCounterHook(SessionRunHook):
    """Logs steps/sec (and examples/sec if batch size known) every N steps."""

    def __init__(self, every_n_steps: int = 100, batch_size: Optional[int] = None,
                 metrics=None):
        self._every = every_n_steps
        self._batch = batch_size
        self._metrics = metrics
        self._t0: Optional[float] = None
        self._s0 = 0
        self.last_steps_per_sec: Optional[float] = None

    def after_run(self, run_context, run_values) -> None:
        gs = run_context.global_step
        if self._t0 is None:
            self._t0, self._s0 = time.perf_counter(), gs
            return
        if gs - self._s0 >= self._every:
            dt = time.perf_counter() - self._t0
            sps = (gs - self._s0) / dt
            self.last_steps_per_sec = sps
            rec: Dict[str, Any] = {"step": gs, "steps_per_sec": round(sps, 3)}
            if self._batch:
                rec["examples_per_sec"] = round(sps * self._batch, 1)
            if self._metrics is not None:
                self._metrics.log(**rec)
            else:
                print(f"[rocps] step {gs}: {sps:.2f} steps/s"
                      + (f", {sps * self._batch:.0f} ex/s" if self._batch else ""),
                      flush=True)
            self._t0, self._s0 = time.perf_counter(), gs


class NanTensorHook(SessionRunHook):
    """Stops (or raises) when the loss is NaN — mirrors tf.train.NanTensorHook."""

    def __init__(self, fail_on_nan_loss: bool = True):
        self._fail = fail_on_nan_loss

    def after_run(self, run_context, run_values) -> None:
        loss = run_values.results
        if loss is None:
            return
        v = float(loss if not hasattr(loss, "item") else loss.item())
        if math.isnan(v) or math.isinf(v):
            if self._fail:
                raise RuntimeError(f"NaN/Inf loss at step {run_context.global_step}")
            run_context.request_stop()


class CheckpointSaverHook(SessionRunHook):
    """Periodic checkpointing (reference: CheckpointSaverHook).

    Installed on EVERY rank in sync mode: collecting the state is a
    collective in the reduce-scatter engine (each rank all-gathers its
    optimizer-state shard), so all ranks must reach it together; only
    the chief writes the file. Time-based saves are decided by the
    chief and broadcast so ranks can't straddle the deadline.
    """

    def __init__(self, checkpoint_dir: str, save_secs: Optional[float] = None,
                 save_steps: Optional[int] = None, saver=None,
                 checkpoint_basename: str = "model.ckpt"):
        if (save_secs is None) == (save_steps is None):
            raise ValueError("exactly one of save_secs / save_steps")
        self._dir = checkpoint_dir
        self._save_secs = save_secs
        self._save_steps = save_steps
        self._saver = saver
        self._basename = checkpoint_basename
        self._last_save_t = time.time()
        self._last_save_s = 0
        self._is_chief = True

    def _prefix(self) -> str:
        import os
        os.makedirs(self._dir, exist_ok=True)
        return os.path.join(self._dir, self._basename)

    def after_create_session(self, session) -> None:
        self._saver = self._saver or session.saver
        self._last_save_s = session.global_step
        self._is_chief = getattr(session, "is_chief", True)
        self._comm = getattr(session.trainer, "comm", None)

    def _due(self, gs: int) -> bool:
        if self._save_steps is not None:
            # deterministic across ranks: global_step is synchronized
            return gs - self._last_save_s >= self._save_steps
        due = time.time() - self._last_save_t >= self._save_secs
        comm = self._comm
        if comm is not None and comm.is_distributed:
            # chief decides; ranks at the same step must agree or the
            # collective state collection deadlocks
            import torch as _torch
            flag = _torch.tensor(
                [1 if (due and self._is_chief) else 0], dtype=_torch.int64,
                device=comm.device if comm.backend == "nccl" else "cpu")
            comm.broadcast(flag, src=0)
            due = bool(flag.item())
        return due

    def _save(self, session, gs: int) -> None:
        # async mode has no collectives: non-chief ranks skip collection
        if self._is_chief or session.trainer.mode == "sync":
            tensors = session.trainer.state_tensors()  # collective in rs_ag
            if self._is_chief:
                self._saver.save(tensors, self._prefix(), global_step=gs)
        self._last_save_t, self._last_save_s = time.time(), gs

    def after_run(self, run_context, run_values) -> None:
        gs = run_context.global_step
        if self._due(gs):
            self._save(run_context.session, gs)

    def end(self, session) -> None:
        if self._saver is not None and session.global_step != self._last_save_s:
            self._save(session, session.global_step)


class ProfilerHook(SessionRunHook):
    """Chrome-trace capture every N steps (reference: tf.train.
    ProfilerHook / RunMetadata timelines). Uses torch.profiler, which on
    ROCm records HIP kernel + memcpy activity; traces open in
    chrome://tracing or perfetto."""

    def __init__(self, save_steps: int, output_dir: str,
                 profile_steps: int = 1):
        self._save_steps = int(save_steps)
        self._dir = output_dir
        self._profile_steps = int(profile_steps)
        self._prof = None
        self._steps_left = 0

    def before_run(self, run_context) -> None:
        gs = run_context.global_step
        if self._prof is None and gs and gs % self._save_steps == 0:
            import torch.profiler as tp
            acts = [tp.ProfilerActivity.CPU]
            if torch.cuda.is_available():
                acts.append(tp.ProfilerActivity.CUDA)
            self._prof = tp.profile(activities=acts)
            self._prof.__enter__()
            self._steps_left = self._profile_steps
            self._start_step = gs

    def after_run(self, run_context, run_values) -> None:
        if self._prof is None:
            return
        self._steps_left -= 1
        if self._steps_left <= 0:
            import os
            self._prof.__exit__(None, None, None)
            os.makedirs(self._dir, exist_ok=True)
            path = os.path.join(self._dir,
                                f"trace_step{self._start_step}.json")
            self._prof.export_chrome_trace(path)
            self._prof = None

    def end(self, session) -> None:
        if self._prof is not None:
            self._prof.__exit__(None, None, None)
            self._prof = None


class SummarySaverHook(SessionRunHook):
    """Writes scalar summaries (loss, lr) to the metrics JSONL writer."""

    def __init__(self, metrics, every_n_steps: int = 10):
        self._metrics = metrics
        self._every = every_n_steps

    def after_run(self, run_context, run_values) -> None:
        gs = run_context.global_step
        if gs % self._every == 0 and run_values.results is not None:
            loss = run_values.results
            v = float(loss if not hasattr(loss, "item") else loss.item())
            self._metrics.log(step=gs, loss=v)


class LoggingTensorHook(SessionRunHook):
    def __init__(self, every_n_iter: int = 100, formatter=None):
        self._every = every_n_iter
        self._fmt = formatter

    def after_run(self, run_context, run_values) -> None:
        gs = run_context.global_step
        if gs % self._every == 0:
            loss = run_values.results
            v = None if loss is None else float(
                loss if not hasattr(loss, "item") else loss.item())
            msg = (self._fmt(gs, v) if self._fmt
                   else f"[rocps] step {gs} loss {v}")
            print(msg, flush=True)
