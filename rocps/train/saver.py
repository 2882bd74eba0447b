"""Saver — the reference's ``tf.train.Saver`` surface over the native
TF-V2 bundle codec (capability C12).

``save()`` writes ``prefix-STEP.index`` + ``prefix-STEP.data-*`` in the
TF checkpoint-V2 format, updates the ``checkpoint`` state text file
(``model_checkpoint_path`` / ``all_model_checkpoint_paths``), and
garbage-collects beyond ``max_to_keep`` — the same observable behavior
as the reference stack.
"""

from __future__ import annotations

import os
import time
from typing import Dict, List, Optional

import torch

__all__ = ["Saver", "latest_checkpoint", "get_checkpoint_state",
           "NewCheckpointReader"]


def _state_file(checkpoint_dir: str) -> str:
    return os.path.join(checkpoint_dir, "checkpoint")


def get_checkpoint_state(checkpoint_dir: str) -> Optional[Dict[str, object]]:
    path = _state_file(checkpoint_dir)
    if not os.path.exists(path):
        return None
    latest, all_paths = None, []
    with open(path) as f:
        for line in f:
            line = line.strip()
            if line.startswith("model_checkpoint_path:"):
                latest = line.split(":", 1)[1].strip().strip('"')
            elif line.startswith("all_model_checkpoint_paths:"):
                all_paths.append(line.split(":", 1)[1].strip().strip('"'))
    if latest is None:
        return None
    return {"model_checkpoint_path": latest,
            "all_model_checkpoint_paths": all_paths}


def latest_checkpoint(checkpoint_dir: str) -> Optional[str]:
    st = get_checkpoint_state(checkpoint_dir)
    if st is None:
        return None
    p = str(st["model_checkpoint_path"])
    if not os.path.isabs(p):
        p = os.path.join(checkpoint_dir, p)
    if os.path.exists(p + ".index"):
        return p
    return None


class Saver:
    def __init__(self, var_source=None, max_to_keep: int = 5,
                 keep_checkpoint_every_n_hours: float = 10000.0):
        """``var_source``: a Trainer (state_tensors/load_state_tensors),
        a dict of tensors, or None (pass tensors to save()).

        ``keep_checkpoint_every_n_hours``: like the reference Saver —
        a checkpoint that would be GC'd by ``max_to_keep`` is preserved
        on disk (though dropped from the state file) if at least this
        many hours passed since the last preserved one."""
        self._source = var_source
        self.max_to_keep = max_to_keep
        self.keep_checkpoint_every_n_hours = keep_checkpoint_every_n_hours
        self._kept: List[str] = []
        self._save_time: Dict[str, float] = {}
        self._last_preserved: float = time.time()

    # ------------------------------------------------------------------
    def _collect(self, session_or_tensors) -> Dict[str, torch.Tensor]:
        obj = session_or_tensors if session_or_tensors is not None else self._source
        if hasattr(obj, "trainer"):  # MonitoredTrainingSession
            obj = obj.trainer
        if hasattr(obj, "state_tensors"):
            return obj.state_tensors()
        if isinstance(obj, dict):
            return obj
        if hasattr(obj, "state_dict"):  # raw nn.Module
            return {k: v for k, v in obj.state_dict().items()}
        raise TypeError(f"cannot collect checkpoint state from {type(obj)}")

    def save(self, session=None, save_path: str = "model.ckpt",
             global_step: Optional[int] = None) -> str:
        from rocps.checkpoint.bundle import write_bundle
        tensors = self._collect(session)
        prefix = (f"{save_path}-{int(global_step)}"
                  if global_step is not None else save_path)
        os.makedirs(os.path.dirname(prefix) or ".", exist_ok=True)
        write_bundle(prefix, tensors)
        self._update_state(prefix)
        return prefix

    def _update_state(self, prefix: str) -> None:
        d = os.path.dirname(prefix) or "."
        base = os.path.basename(prefix)
        self._save_time[base] = time.time()
        self._kept = [p for p in self._kept if p != base] + [base]
        dropped, self._kept = (self._kept[:-self.max_to_keep],
                               self._kept[-self.max_to_keep:])
        with open(_state_file(d), "w") as f:
            f.write(f'model_checkpoint_path: "{base}"\n')
            for p in self._kept:
                f.write(f'all_model_checkpoint_paths: "{p}"\n')
        horizon = self.keep_checkpoint_every_n_hours * 3600.0
        for p in dropped:
            ts = self._save_time.pop(p, 0.0)
            if ts - self._last_preserved >= horizon:
                self._last_preserved = ts  # preserved on disk, off-state
                continue
            for suffix in (".index", ".data-00000-of-00001"):
                try:
                    os.remove(os.path.join(d, p + suffix))
                except OSError:
                    pass

    # ------------------------------------------------------------------
    def restore(self, session, save_path: str) -> None:
        from rocps.checkpoint.bundle import read_bundle
        tensors = read_bundle(save_path)
        obj = session if session is not None else self._source
        if hasattr(obj, "trainer"):
            obj = obj.trainer
        if hasattr(obj, "load_state_tensors"):
            obj.load_state_tensors(tensors)
        elif hasattr(obj, "load_state_dict"):
            obj.load_state_dict(tensors, strict=False)
        else:
            raise TypeError(f"cannot restore into {type(obj)}")

    @staticmethod
    def last_checkpoints(checkpoint_dir: str) -> List[str]:
        st = get_checkpoint_state(checkpoint_dir)
        return list(st["all_model_checkpoint_paths"]) if st else []


class NewCheckpointReader:
    """``tf.train.NewCheckpointReader`` parity: inspect a TF-V2 bundle."""

    def __init__(self, prefix: str):
        from rocps.checkpoint.bundle import BundleReader, TORCH_DTYPE
        self._r = BundleReader(prefix)
        self._dtypes = TORCH_DTYPE

    def get_variable_to_shape_map(self):
        return {e["name"]: list(e["shape"]) for e in self._r.list_tensors()}

    def get_variable_to_dtype_map(self):
        return {e["name"]: self._dtypes[e["dtype"]]
                for e in self._r.list_tensors()}

    def has_tensor(self, name: str) -> bool:
        return self._r.contains(name)

    def get_tensor(self, name: str):
        from rocps.checkpoint.bundle import read_bundle
        return read_bundle_single(self._r, name, self._dtypes)


def read_bundle_single(reader, name: str, dtypes):
    import numpy as np
    import torch as _torch
    dtype_code, shape, _ = reader.entry_info(name)
    dtype = dtypes[dtype_code]
    raw = reader.read(name)
    storage = {_torch.bfloat16: _torch.uint16,
               _torch.bool: _torch.uint8}.get(dtype, dtype)
    np_dtype = _torch.empty(0, dtype=storage).numpy().dtype
    arr = np.frombuffer(raw, dtype=np_dtype).copy()
    return _torch.from_numpy(arr).view(dtype).reshape(list(shape))
