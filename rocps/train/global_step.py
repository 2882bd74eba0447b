"""Global step — capability C17 of SURVEY.md §2b.

The reference keeps ``global_step`` as a variable on the PS, bumped by
the (chief's) apply op and read by hooks / SyncReplicasOptimizer
staleness checks. Here it is a small host tensor owned by the Trainer,
mirrored to the control plane (counter ``global_step``) so PS processes
and monitoring can read it; in async mode the PS shard service owns it.
"""

from __future__ import annotations

from typing import Optional

import torch

_GLOBAL_STEP: Optional[torch.Tensor] = None


def get_or_create_global_step() -> torch.Tensor:
    global _GLOBAL_STEP
    if _GLOBAL_STEP is None:
        _GLOBAL_STEP = torch.zeros((), dtype=torch.int64)
    return _GLOBAL_STEP


def get_global_step() -> Optional[torch.Tensor]:
    return _GLOBAL_STEP


def reset_global_step() -> None:
    global _GLOBAL_STEP
    _GLOBAL_STEP = None
