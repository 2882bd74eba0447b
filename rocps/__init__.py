"""rocps — an MI355X-native parameter-server data-parallel training runtime.

A from-scratch rebuild of the capability surface of
``gctian/distributed-tensorflow`` (a TF 1.x parameter-server training
runtime): ClusterSpec / chief-worker-ps bootstrap, sync- and async-SGD
data parallelism, MonitoredTrainingSession with hooks, and TF-V2
checkpoint format — re-designed for AMD MI355X (gfx950):

- PyTorch-ROCm supplies tensors/autograd; one process per GPU.
- Sync SGD: bucketed RCCL reduce-scatter + all-gather over xGMI
  (the reduce-scattered segment on each rank *is* the co-located PS
  shard), launched from backward hooks and overlapped with compute.
- Async SGD: PS shards in host DRAM, pinned hipMemcpyAsync push/pull.
- Hot ops: hand-written HIP/CDNA4 kernels (MFMA + LDS tiling) in
  ``csrc/kernels``; TCP control plane for bootstrap/coordination.

Public API intentionally mirrors the reference's TF-1.x surface
(``rocps.train.ClusterSpec``, ``rocps.train.Server``,
``rocps.train.MonitoredTrainingSession``, ``rocps.train.Saver``, ...)
so reference run commands work with this framework.
"""

__version__ = "0.1.0"

from rocps import config  # noqa: F401
from rocps import app  # noqa: F401
from rocps import train  # noqa: F401

flags = config  # tf.app.flags-style alias: rocps.flags.DEFINE_string(...)
