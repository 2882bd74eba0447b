"""Optimizers — capability C8 of SURVEY.md §2b.

The reference's update kernels (``ApplyGradientDescent``,
``ApplyMomentum``, ``ApplyAdam`` — applied per-variable *on the PS
device*) become flat multi-tensor updates applied to each bucket's
contiguous parameter storage (full bucket in allreduce mode, the
rank-local reduce-scattered shard in rs_ag mode — that shard IS the
co-located PS shard).

Numerics: parameters may be bf16; the optimizer keeps fp32 master
weights + fp32 state per slot and writes back the bf16 view. On GPU
the update runs as one fused HIP kernel per flat tensor
(``rocps.ops.fused_optim``; csrc/kernels/fused_optimizers.hip) —
grid-stride, one launch per bucket instead of one per variable.
"""

from __future__ import annotations

from typing import Any, Dict, Hashable

import os

import torch

__all__ = ["Optimizer", "GradientDescentOptimizer", "MomentumOptimizer",
           "AdamOptimizer"]


def _use_fused(p: torch.Tensor) -> bool:
    if p.device.type != "cuda":
        return False
    from rocps.ops import fused_optim
    return fused_optim.available()


_CPU_APPLY = None
_EMPTY = torch.empty(0)


def _cpu_fused(p: torch.Tensor, g: torch.Tensor):
    """Fused single-pass C++ CPU apply (csrc/cpu/cpu_apply.cc) for the
    async PS shard thread — torch's op-at-a-time path makes ~8 memory
    passes per momentum apply, the fused loop ~5 (ROADMAP §Perf 4).
    SGD/momentum are bit-identical to the torch ops (FMA rounding
    mirrored); Adam is within 1 ulp/step (torch's SLEEF sqrt is not
    correctly rounded). ROCPS_CPU_FUSED=0 disables."""
    global _CPU_APPLY
    if p.device.type != "cpu" or os.environ.get("ROCPS_CPU_FUSED") == "0":
        return None
    if not (p.is_contiguous() and g.is_contiguous()):
        return None
    if p.dtype not in (torch.float32, torch.bfloat16):
        return None
    if g.dtype not in (torch.float32, torch.bfloat16):
        return None
    if _CPU_APPLY is None:
        try:
            from rocps import _cpuapply
            _CPU_APPLY = _cpuapply
        except ImportError:
            _CPU_APPLY = False
    return _CPU_APPLY or None


class Optimizer:
    """Functional flat-apply optimizer with lazily-created per-slot state."""

    def __init__(self, learning_rate: float = 0.01):
        self.lr = float(learning_rate)
        self._state: Dict[Hashable, Dict[str, Any]] = {}

    # -- the engine-facing interface ------------------------------------
    def slot_state(self, slot: Hashable, params: torch.Tensor) -> Dict[str, Any]:
        st = self._state.get(slot)
        if st is None:
            st = self._init_state(params)
            self._state[slot] = st
        return st

    def _init_state(self, params: torch.Tensor) -> Dict[str, Any]:
        st: Dict[str, Any] = {}
        if params.dtype != torch.float32:
            st["master"] = params.detach().float()
        return st

    def _master(self, st: Dict[str, Any], params: torch.Tensor) -> torch.Tensor:
        return st.get("master", params)

    def _writeback(self, st: Dict[str, Any], params: torch.Tensor) -> None:
        if "master" in st:
            params.copy_(st["master"])

    def apply_flat(self, params: torch.Tensor, grads: torch.Tensor,
                   slot: Hashable) -> None:
        raise NotImplementedError

    # TF-1.x surface compatibility: minimize() on a loss closure is the
    # Trainer/engine's job; these helpers match the reference call shape.
    def state_dict(self) -> Dict[str, Any]:
        return {"lr": self.lr,
                "state": {repr(k): {n: v for n, v in st.items()}
                          for k, st in self._state.items()}}

    def load_state_dict(self, sd: Dict[str, Any]) -> None:
        """Restore lr + any already-materialised slot state.

        Checkpoint restore of optimizer state goes through
        ``Trainer._load_optimizer_state`` (which materialises slots
        first); this method only covers same-process round trips.
        """
        self.lr = sd["lr"]
        by_repr = {repr(k): k for k in self._state}
        for ks, st in sd["state"].items():
            if ks in by_repr:
                for n, v in st.items():
                    self._state[by_repr[ks]][n].copy_(v)


class GradientDescentOptimizer(Optimizer):
    """plain SGD: p -= lr * g  (reference: ApplyGradientDescent)."""

    def apply_flat(self, params, grads, slot):
        st = self.slot_state(slot, params)
        m = self._master(st, params)
        if _use_fused(params):
            from rocps.ops import fused_optim
            fused_optim.sgd(params, grads, m if "master" in st else None, self.lr)
            return
        capp = _cpu_fused(params, grads)
        if capp is not None:
            capp.sgd(params, grads, st.get("master", _EMPTY), self.lr,
                     "master" in st)
            return
        m.add_(grads.to(m.dtype), alpha=-self.lr)
        self._writeback(st, params)


class MomentumOptimizer(Optimizer):
    """p -= lr * v;  v = mu*v + g  (reference: ApplyMomentum).

    ``use_nesterov`` matches TF's flag.
    """

    def __init__(self, learning_rate: float, momentum: float,
                 use_nesterov: bool = False):
        super().__init__(learning_rate)
        self.momentum = float(momentum)
        self.use_nesterov = use_nesterov

    def _init_state(self, params):
        st = super()._init_state(params)
        st["mom"] = torch.zeros(params.numel(), dtype=torch.float32,
                                device=params.device)
        return st

    def apply_flat(self, params, grads, slot):
        st = self.slot_state(slot, params)
        m = self._master(st, params)
        v = st["mom"]
        if _use_fused(params):
            from rocps.ops import fused_optim
            fused_optim.momentum(params, grads, m if "master" in st else None,
                                 v, self.lr, self.momentum, self.use_nesterov)
            return
        capp = _cpu_fused(params, grads)
        if capp is not None:
            capp.momentum(params, grads, st.get("master", _EMPTY), v,
                          self.lr, self.momentum, self.use_nesterov,
                          "master" in st)
            return
        g32 = grads.to(torch.float32)
        v.mul_(self.momentum).add_(g32)
        if self.use_nesterov:
            m.add_(g32, alpha=-self.lr).add_(v, alpha=-self.lr * self.momentum)
        else:
            m.add_(v, alpha=-self.lr)
        self._writeback(st, params)


class AdamOptimizer(Optimizer):
    """TF-semantics Adam (reference: ApplyAdam): bias-corrected via
    alpha_t = lr * sqrt(1-b2^t) / (1-b1^t), epsilon inside sqrt-denominator
    as eps_hat (TF adds eps to sqrt(v_hat))."""

    def __init__(self, learning_rate: float = 0.001, beta1: float = 0.9,
                 beta2: float = 0.999, epsilon: float = 1e-8):
        super().__init__(learning_rate)
        self.beta1, self.beta2, self.epsilon = beta1, beta2, epsilon
        self._t = 0

    def _init_state(self, params):
        st = super()._init_state(params)
        st["m"] = torch.zeros(params.numel(), dtype=torch.float32,
                              device=params.device)
        st["v"] = torch.zeros(params.numel(), dtype=torch.float32,
                              device=params.device)
        return st

    def begin_step(self) -> None:
        self._t += 1

    def apply_flat(self, params, grads, slot):
        st = self.slot_state(slot, params)
        t = max(self._t, 1)
        alpha = (self.lr * (1 - self.beta2 ** t) ** 0.5 / (1 - self.beta1 ** t))
        mst = self._master(st, params)
        m, v = st["m"], st["v"]
        if _use_fused(params):
            from rocps.ops import fused_optim
            fused_optim.adam(params, grads, mst if "master" in st else None,
                             m, v, alpha, self.beta1, self.beta2, self.epsilon)
            return
        capp = _cpu_fused(params, grads)
        if capp is not None:
            capp.adam(params, grads, st.get("master", _EMPTY), m, v,
                      alpha, self.beta1, self.beta2, self.epsilon,
                      "master" in st)
            return
        g32 = grads.to(torch.float32)
        m.mul_(self.beta1).add_(g32, alpha=1 - self.beta1)
        v.mul_(self.beta2).addcmul_(g32, g32, value=1 - self.beta2)
        mst.addcdiv_(m, v.sqrt().add_(self.epsilon), value=-alpha)
        self._writeback(st, params)
