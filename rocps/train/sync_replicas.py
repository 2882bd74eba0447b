"""SyncReplicasOptimizer — API parity for the reference's ``--issync`` path.

Reference semantics: per-variable ConditionalAccumulators
on the PS take exactly ``replicas_to_aggregate`` fresh gradients
(stale ones — local_step < global_step — dropped), the chief applies
the average, a token queue releases workers.

MI355X recast: with ``replicas_to_aggregate == total_num_replicas``
(the demo default) exactly-N aggregation + all-release is semantically
a reduce-scatter → apply → all-gather barrier, which is what
``SyncEngine`` runs. With ``replicas_to_aggregate < total``
(backup-worker mode) the Trainer routes to the PS-shard *semisync*
engine (rocps.engine.async_engine._semisync_loop): exactly-K fresh
gradients per round, stale-drop with straggler release, token-queue
semantics — the reference's ConditionalAccumulator behavior.
"""

from __future__ import annotations

from typing import Optional

from rocps.train.optimizer import Optimizer

__all__ = ["SyncReplicasOptimizer"]


class SyncReplicasOptimizer(Optimizer):
    def __init__(self, opt: Optimizer, replicas_to_aggregate: int,
                 total_num_replicas: Optional[int] = None):
        super().__init__(opt.lr)
        self.base = opt
        self.replicas_to_aggregate = int(replicas_to_aggregate)
        self.total_num_replicas = int(total_num_replicas
                                      or replicas_to_aggregate)
        self.is_sync = True

    # delegate the functional interface
    def apply_flat(self, params, grads, slot):
        self.base.lr = self.lr
        return self.base.apply_flat(params, grads, slot)

    def begin_step(self):
        if hasattr(self.base, "begin_step"):
            self.base.begin_step()

    def state_dict(self):
        return self.base.state_dict()

    def load_state_dict(self, sd):
        return self.base.load_state_dict(sd)

    # -- reference API stubs (graph-mode artifacts; no-ops eagerly) -----
    def get_chief_queue_runner(self):
        return None

    def get_init_tokens_op(self, num_tokens: int = -1):
        return None

    def make_session_run_hook(self, is_chief: bool, num_tokens: int = -1):
        return None
