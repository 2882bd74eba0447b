"""Sync-SGD data-parallel engine — capabilities C4 + C7 of SURVEY.md §2b.

The reference's sync path (``tf.train.SyncReplicasOptimizer``) collects
exactly-N gradients in per-variable accumulators on the PS, averages,
applies once on the PS, and releases workers through a token queue — a
full barrier per step.

MI355X-native recast (semantically the same N-replica averaging +
barrier), chosen for 7-link xGMI instead of a PS hop:

- gradients accumulate straight into flat per-bucket buffers
  (``rocps.engine.buckets``), collectives launch from
  post-accumulate-grad hooks as each bucket fills → communication
  overlaps the remaining backward (C7's 3-stream overlap; RCCL work
  runs on its own HIP stream, ``work.wait()`` fences the compute
  stream).
- ``rs_ag`` mode (RCCL): reduce-scatter each bucket; rank r's segment
  is its co-located PS shard; the (fused) optimizer updates only that
  shard against fp32 master weights; all-gather broadcasts updated
  bf16 parameters. Bytes per step per GPU ≈ 2×P×(N-1)/N — the same
  as the reference's pull+push, but over xGMI instead of TCP.
- ``allreduce`` mode (gloo / CPU plumbing config): all-reduce buckets,
  every rank applies the identical update — bit-identical trajectory
  to a single process at equal global batch, which is the M1 gate.
"""

from __future__ import annotations

import os
from typing import Dict, List, Optional, Tuple

import torch

from rocps.engine.buckets import Bucket, build_buckets
from rocps.engine.comm import Comm

__all__ = ["SyncEngine"]


class SyncEngine:
    def __init__(self, model: torch.nn.Module, comm: Comm,
                 mode: Optional[str] = None,
                 bucket_cap_mb: float = 25.0,
                 grad_dtype: Optional[torch.dtype] = None,
                 overlap: bool = True):
        # ROCPS_DETERMINISTIC=1 (SURVEY §5 race-debug mode): disable
        # hook-order collective launches — every bucket launches in
        # bucket-index order at finish_backward, giving a fixed
        # reduction schedule run-to-run. Comm.tune_env pins RCCL to a
        # single channel/fixed algorithm under the same flag.
        if os.environ.get("ROCPS_DETERMINISTIC") == "1":
            overlap = False
        self.comm = comm
        self.model = model
        self.mode = mode or ("rs_ag" if comm.supports_reduce_scatter()
                             else "allreduce")
        if self.mode == "rs_ag" and not comm.supports_reduce_scatter():
            raise ValueError("rs_ag mode needs the RCCL (nccl) backend")
        self.overlap = overlap
        named = list(model.named_parameters())
        self.buckets: List[Bucket] = build_buckets(
            named, comm.world_size, bucket_cap_mb=bucket_cap_mb,
            grad_dtype=grad_dtype, flatten_params=True)
        self._param_slot: Dict[int, Tuple[Bucket, int]] = {}
        self._hooks = []
        for b in self.buckets:
            if self.mode == "rs_ag":
                b.grad_shard = torch.empty(
                    b.shard_numel, dtype=b.dtype, device=b.device)
            for i, (name, p) in enumerate(b.named_params):
                self._param_slot[id(p)] = (b, i)
                self._hooks.append(p.register_post_accumulate_grad_hook(
                    self._on_grad_ready))
        self._launch_order: List[Bucket] = []
        self._in_backward = False

    # ------------------------------------------------------------------
    def broadcast_parameters(self, src: int = 0) -> None:
        """Chief init broadcast (reference: chief runs init_op, others wait).
        Includes non-parameter buffers (BN running stats): the reference
        keeps them as shared PS variables, so replicas start identical."""
        for b in self.buckets:
            self.comm.broadcast(b.params_flat, src=src)
        for _, buf in self.model.named_buffers():
            self.comm.broadcast(buf, src=src)

    def sync_buffers(self) -> None:
        """All-reduce-average float buffers (BN running stats) so every
        replica agrees — the eager recast of the reference's PS-shared
        moving statistics. Collective: call on every rank together."""
        if not self.comm.is_distributed:
            return
        for _, buf in self.model.named_buffers():
            if buf.is_floating_point():
                self.comm.all_reduce(buf)
                buf.div_(self.comm.world_size)

    def zero_grad(self) -> None:
        for b in self.buckets:
            b.zero_grads()
            b.reset()
        self._launch_order = []
        self._in_backward = True

    # ------------------------------------------------------------------
    def _on_grad_ready(self, param: torch.nn.Parameter) -> None:
        if not self._in_backward:
            return
        bucket, _ = self._param_slot[id(param)]
        bucket.ready_count += 1
        if bucket.ready_count == bucket.expected and self.overlap:
            self._launch(bucket)

    def _launch(self, bucket: Bucket) -> None:
        if bucket.work is not None:
            return
        if self.mode == "rs_ag":
            bucket.work = self.comm.reduce_scatter(
                bucket.grad_shard, bucket.grads, async_op=True)
        else:
            bucket.work = self.comm.all_reduce(bucket.grads, async_op=True)
        self._launch_order.append(bucket)

    def finish_backward(self) -> None:
        """Launch any stragglers, wait for all in-flight collectives."""
        for b in self.buckets:
            if b.work is None:
                self._launch(b)
        for b in self._launch_order:
            b.work.wait()
        self._in_backward = False
        inv_world = 1.0 / self.comm.world_size
        if self.comm.world_size > 1:
            if self.mode == "rs_ag":
                torch._foreach_mul_([b.grad_shard for b in self.buckets],
                                    inv_world)
            else:
                torch._foreach_mul_([b.grads for b in self.buckets], inv_world)
        elif self.mode == "rs_ag":
            # world==1: reduce_scatter stub copied full flat; shard==flat
            pass

    # ------------------------------------------------------------------
    def apply_and_sync(self, optimizer) -> None:
        """Run the optimizer on shards (rs_ag) or full buckets (allreduce),
        then all-gather updated parameters (rs_ag only)."""
        if self.mode == "rs_ag":
            works = []
            for b in self.buckets:
                pshard = b.my_shard(b.params_flat, self.comm.rank)
                optimizer.apply_flat(pshard, b.grad_shard, slot=("b", b.index))
                works.append((b, self.comm.all_gather(
                    b.params_flat, pshard, async_op=True)))
            for b, w in works:
                w.wait()
        else:
            for b in self.buckets:
                optimizer.apply_flat(b.params_flat, b.grads[:b.numel_padded],
                                     slot=("b", b.index))

    # ------------------------------------------------------------------
    def step(self, loss: torch.Tensor, optimizer) -> None:
        """Backward + comm + optimizer, fully overlapped."""
        loss.backward()
        self.finish_backward()
        self.apply_and_sync(optimizer)

    def detach(self) -> None:
        for h in self._hooks:
            h.remove()
        self._hooks = []

    def grad_global_norm(self) -> torch.Tensor:
        if self.mode == "rs_ag":
            sq = sum((b.grad_shard.float() ** 2).sum() for b in self.buckets)
            if self.comm.is_distributed:
                self.comm.all_reduce(sq)
            return sq.sqrt()
        sq = sum((b.grads.float() ** 2).sum() for b in self.buckets)
        return sq.sqrt()
