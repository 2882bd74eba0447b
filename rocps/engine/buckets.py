"""Gradient bucketing for collective launch — part of C4/C7.

Parameters are packed, in reverse registration order (≈ backward
completion order), into flat buckets capped at ``bucket_cap_mb``.
Default 25 MB: xGMI is 7 p2p links per GPU, RCCL stripes a ring across
>=8 channels, and 25 MB / 8 channels keeps per-channel chunks >=1 MB
(SURVEY §2d bucket-tuning note) while leaving enough buckets to overlap
with the remaining backward.

Each bucket owns:
- ``grads``: one contiguous tensor; every param's ``.grad`` is a view
  into it, so autograd accumulates in place and the collective launches
  on the whole bucket with zero repacking.
- ``params_flat`` (optional, rs_ag mode): contiguous parameter storage,
  ``param.data`` re-pointed as views, so all-gather of updated shards
  writes parameters in place.

Buckets are padded to ``world_size`` divisibility so reduce-scatter
shards are equal; rank r's segment of every bucket is its co-located
"PS shard" (the reference's ps-task variable placement, recast).
"""

from __future__ import annotations

from typing import Iterable, List, Optional, Tuple

import torch

__all__ = ["Bucket", "build_buckets"]


def _align(n: int, a: int) -> int:
    return (n + a - 1) // a * a


class Bucket:
    def __init__(self, index: int, named_params: List[Tuple[str, torch.nn.Parameter]],
                 world_size: int, grad_dtype: Optional[torch.dtype] = None,
                 flatten_params: bool = False):
        self.index = index
        self.named_params = named_params
        self.world_size = world_size
        p0 = named_params[0][1]
        self.device = p0.device
        self.dtype = grad_dtype or p0.dtype
        numel = sum(p.numel() for _, p in named_params)
        # pad so the bucket splits evenly into world_size shards of
        # elements aligned to 64 (wave width / vector loads)
        self.numel_padded = _align(numel, world_size * 64)
        self.shard_numel = self.numel_padded // world_size
        self.numel = numel

        self.grads = torch.zeros(self.numel_padded, dtype=self.dtype,
                                 device=self.device)
        self.offsets: List[int] = []
        off = 0
        for _, p in named_params:
            self.offsets.append(off)
            off += p.numel()

        self.params_flat: Optional[torch.Tensor] = None
        if flatten_params:
            self.params_flat = torch.zeros(self.numel_padded, dtype=p0.dtype,
                                           device=self.device)
            for (name, p), o in zip(named_params, self.offsets):
                self.params_flat[o:o + p.numel()].copy_(p.data.reshape(-1))
                p.data = self.params_flat[o:o + p.numel()].view_as(p.data)

        # wire .grad as views into the flat grad buffer
        for (name, p), o in zip(named_params, self.offsets):
            g = self.grads[o:o + p.numel()].view_as(p.data)
            p.grad = g

        # per-step readiness
        self.expected = len(named_params)
        self.ready_count = 0
        self.work = None  # in-flight collective handle
        # rs_ag working buffers (allocated lazily by the engine)
        self.grad_shard: Optional[torch.Tensor] = None

    def grad_view(self, i: int) -> torch.Tensor:
        name, p = self.named_params[i]
        o = self.offsets[i]
        return self.grads[o:o + p.numel()].view_as(p.data)

    def my_shard(self, flat: torch.Tensor, rank: int) -> torch.Tensor:
        return flat[rank * self.shard_numel:(rank + 1) * self.shard_numel]

    def reset(self) -> None:
        self.ready_count = 0
        self.work = None

    def zero_grads(self) -> None:
        self.grads.zero_()

    def __repr__(self) -> str:
        mb = self.numel_padded * self.grads.element_size() / 1e6
        return (f"Bucket({self.index}, tensors={len(self.named_params)}, "
                f"{mb:.1f} MB, shard={self.shard_numel})")


def build_buckets(named_params: Iterable[Tuple[str, torch.nn.Parameter]],
                  world_size: int, bucket_cap_mb: float = 25.0,
                  grad_dtype: Optional[torch.dtype] = None,
                  flatten_params: bool = False,
                  first_bucket_mb: float = 1.0) -> List[Bucket]:
    """Pack params (reverse order) into capped buckets.

    The first (earliest-launching) bucket is small (``first_bucket_mb``)
    so the first collective starts as early in backward as possible —
    same rationale as DDP's small first bucket.
    """
    params = [(n, p) for n, p in named_params if p.requires_grad]
    params.reverse()
    buckets: List[Bucket] = []
    cur: List[Tuple[str, torch.nn.Parameter]] = []
    cur_bytes = 0
    cap = min(first_bucket_mb, bucket_cap_mb) * 1e6
    for name, p in params:
        nb = p.numel() * (torch.tensor([], dtype=grad_dtype).element_size()
                          if grad_dtype else p.element_size())
        if cur and cur_bytes + nb > cap:
            buckets.append(Bucket(len(buckets), cur, world_size, grad_dtype,
                                  flatten_params))
            cur, cur_bytes = [], 0
            cap = bucket_cap_mb * 1e6
        cur.append((name, p))
        cur_bytes += nb
    if cur:
        buckets.append(Bucket(len(buckets), cur, world_size, grad_dtype,
                              flatten_params))
    return buckets
