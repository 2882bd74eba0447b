"""Python face of the native TF-V2 tensor-bundle codec (csrc/bundle).

Maps torch tensors <-> the TF ``DataType`` enum + raw little-endian
bytes. The on-disk format is the reference stack's checkpoint format
(SURVEY.md C12): ``prefix.index`` (leveldb-format table of
BundleHeaderProto/BundleEntryProto) + ``prefix.data-00000-of-00001``.
"""

from __future__ import annotations

from typing import Dict, List

import numpy as np
import torch

try:
    from rocps import _bundle as _C
except ImportError as e:  # pragma: no cover
    raise ImportError(
        "rocps native bundle codec not built — run "
        "`python setup.py build_ext --inplace` (csrc/bundle/bundle.cc)"
    ) from e

BundleWriter = _C.BundleWriter
BundleReader = _C.BundleReader

# TF DataType enum values (tensorflow/core/framework/types.proto)
TF_DTYPE: Dict[torch.dtype, int] = {
    torch.float32: 1, torch.float64: 2, torch.int32: 3, torch.uint8: 4,
    torch.int16: 5, torch.int8: 6, torch.int64: 9, torch.bool: 10,
    torch.bfloat16: 14, torch.float16: 19,
}
TORCH_DTYPE: Dict[int, torch.dtype] = {v: k for k, v in TF_DTYPE.items()}


def _to_bytes_view(t: torch.Tensor) -> np.ndarray:
    t = t.detach().contiguous().cpu()
    # bf16 has no numpy dtype: reinterpret as uint16 (raw LE bytes identical)
    if t.dtype == torch.bfloat16:
        t = t.view(torch.uint16)
    if t.dtype == torch.bool:
        t = t.view(torch.uint8)
    return t.numpy()


def write_bundle(prefix: str, tensors: Dict[str, torch.Tensor],
                 num_shards: int = 1) -> None:
    w = BundleWriter(prefix, num_shards)
    for name in sorted(tensors):
        t = tensors[name]
        dt = TF_DTYPE.get(t.dtype)
        if dt is None:
            raise TypeError(f"unsupported checkpoint dtype {t.dtype} for {name}")
        w.add(name, dt, list(t.shape), _to_bytes_view(t))
    w.finish()


def read_bundle(prefix: str, names: List[str] = None) -> Dict[str, torch.Tensor]:
    r = BundleReader(prefix)
    out: Dict[str, torch.Tensor] = {}
    entries = r.list_tensors()
    for e in entries:
        name = e["name"]
        if names is not None and name not in names:
            continue
        dtype = TORCH_DTYPE[e["dtype"]]
        raw = r.read(name)
        storage_dtype = {torch.bfloat16: torch.uint16,
                         torch.bool: torch.uint8}.get(dtype, dtype)
        np_dtype = torch.empty(0, dtype=storage_dtype).numpy().dtype
        arr = np.frombuffer(raw, dtype=np_dtype).copy()
        t = torch.from_numpy(arr).view(dtype).reshape(e["shape"])
        out[name] = t
    return out


def list_bundle(prefix: str):
    return BundleReader(prefix).list_tensors()
