"""``tf.train.Supervisor`` parity shim — the older session manager some
reference-family scripts use instead of MonitoredTrainingSession
(SURVEY §0). ``managed_session()`` yields the same monitored session."""

from __future__ import annotations

from contextlib import contextmanager
from typing import Optional

from rocps.train.session import MonitoredTrainingSession, Trainer

__all__ = ["Supervisor"]


class Supervisor:
    def __init__(self, is_chief: bool = True, logdir: Optional[str] = None,
                 trainer: Optional[Trainer] = None,
                 save_model_secs: float = 600.0,
                 global_step=None, init_op=None, summary_op=None,
                 saver=None):
        self.is_chief = is_chief
        self.logdir = logdir
        self.trainer = trainer
        self.save_model_secs = save_model_secs
        self._sess: Optional[MonitoredTrainingSession] = None

    @contextmanager
    def managed_session(self, master: str = "", config=None):
        sess = MonitoredTrainingSession(
            self.trainer, master=master, is_chief=self.is_chief,
            checkpoint_dir=self.logdir,
            save_checkpoint_secs=self.save_model_secs or None)
        self._sess = sess
        with sess as s:
            yield s

    def should_stop(self) -> bool:
        return self._sess.should_stop() if self._sess else False

    def request_stop(self) -> None:
        if self._sess:
            self._sess.request_stop()

    def stop(self) -> None:
        self.request_stop()
