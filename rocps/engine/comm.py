"""Communication backend — capability C6 of SURVEY.md §2b.

The reference moves every parameter pull / gradient push as a gRPC
``RecvTensor`` over TCP. The MI355X-native data plane is RCCL over
xGMI: ``torch.distributed`` with backend "nccl" (which IS RCCL on
ROCm), one process per GPU. gloo serves the CPU-only plumbing config
(BASELINE.json configs[0]); the rocps control plane does rendezvous.

xGMI note: each MI355X has 7 point-to-point links × ≈153 GB/s, so ring
collectives are per-link bound. ``Comm.tune_env`` sets
``NCCL_MIN_NCHANNELS>=8`` so rings stripe across links; bucket sizing
(25–50 MB, engine default) keeps per-channel chunks >=1 MB.
"""

from __future__ import annotations

import datetime
import os
import socket
from typing import Optional

import torch
import torch.distributed as dist

__all__ = ["Comm"]


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


class Comm:
    """Thin wrapper over a torch.distributed process group.

    world_size == 1 short-circuits every collective (single-GPU bench
    path, and the no-cluster local mode).
    """

    def __init__(self, rank: int = 0, world_size: int = 1,
                 backend: Optional[str] = None,
                 device: Optional[torch.device] = None):
        self.rank = rank
        self.world_size = world_size
        self.device = device or (
            torch.device("cuda", torch.cuda.current_device())
            if torch.cuda.is_available() else torch.device("cpu"))
        self.backend = backend or (
            "nccl" if self.device.type == "cuda" else "gloo")

    # ------------------------------------------------------------------
    @staticmethod
    def tune_env() -> None:
        """RCCL tuning for 7-link xGMI (call before init_process_group).

        ROCPS_DETERMINISTIC=1 (SURVEY §5 debug mode) instead pins RCCL
        to one channel and a fixed ring/simple schedule so the
        floating-point reduction order is identical run-to-run; the
        engine also serializes bucket launches under the same flag."""
        if os.environ.get("ROCPS_DETERMINISTIC") == "1":
            os.environ.setdefault("NCCL_MIN_NCHANNELS", "1")
            os.environ.setdefault("NCCL_MAX_NCHANNELS", "1")
            os.environ.setdefault("NCCL_ALGO", "Ring")
            os.environ.setdefault("NCCL_PROTO", "Simple")
        else:
            os.environ.setdefault("NCCL_MIN_NCHANNELS", "8")
        os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")

    @classmethod
    def from_env(cls) -> "Comm":
        """torchrun path: RANK/WORLD_SIZE/MASTER_ADDR from the env."""
        world = int(os.environ.get("WORLD_SIZE", "1"))
        rank = int(os.environ.get("RANK", "0"))
        local_rank = int(os.environ.get("LOCAL_RANK", rank))
        if torch.cuda.is_available():
            torch.cuda.set_device(local_rank)
            device = torch.device("cuda", local_rank)
            backend = "nccl"
        else:
            device = torch.device("cpu")
            backend = "gloo"
        if world > 1 and not dist.is_initialized():
            cls.tune_env()
            dist.init_process_group(
                backend=backend, rank=rank, world_size=world,
                timeout=datetime.timedelta(seconds=300))
        return cls(rank, world, backend, device)

    @classmethod
    def from_cluster(cls, cluster, job_name: str, task_index: int,
                     control, backend: Optional[str] = None,
                     device: Optional[torch.device] = None,
                     timeout: float = 300.0) -> "Comm":
        """Worker-group rendezvous via the rocps control plane.

        Worker 0 picks a free TCP port, publishes it under the key
        ``torch_master``; every worker then joins an init_method
        ``tcp://`` process group. PS tasks do not join.
        """
        assert job_name == "worker", "only workers join the collective group"
        world = cluster.num_tasks("worker")
        rank = task_index
        if device is None:
            device = (torch.device("cuda", torch.cuda.current_device())
                      if torch.cuda.is_available() else torch.device("cpu"))
        backend = backend or ("nccl" if device.type == "cuda" else "gloo")
        if world == 1:
            return cls(0, 1, backend, device)
        if rank == 0:
            host = cluster.task_address("worker", 0).rsplit(":", 1)[0]
            if host in ("0.0.0.0", ""):
                host = "127.0.0.1"
            port = _free_port()
            control.put("torch_master", f"{host}:{port}")
            addr = f"{host}:{port}"
        else:
            addr = control.get("torch_master", blocking=True, timeout=timeout)
        if not dist.is_initialized():
            cls.tune_env()
            dist.init_process_group(
                backend=backend, init_method=f"tcp://{addr}",
                rank=rank, world_size=world,
                timeout=datetime.timedelta(seconds=timeout))
        return cls(rank, world, backend, device)

    # -- collectives ----------------------------------------------------
    @property
    def is_distributed(self) -> bool:
        return self.world_size > 1

    def all_reduce(self, t: torch.Tensor, async_op: bool = False):
        if not self.is_distributed:
            return _Done()
        return dist.all_reduce(t, op=dist.ReduceOp.SUM, async_op=async_op)

    def reduce_scatter(self, out: torch.Tensor, inp: torch.Tensor,
                       async_op: bool = False):
        """out <- sum over ranks of inp[rank-th chunk]. inp flat, len = world*len(out)."""
        if not self.is_distributed:
            out.copy_(inp)
            return _Done()
        return dist.reduce_scatter_tensor(out, inp, op=dist.ReduceOp.SUM,
                                          async_op=async_op)

    def all_gather(self, out: torch.Tensor, shard: torch.Tensor,
                   async_op: bool = False):
        if not self.is_distributed:
            if out.data_ptr() != shard.data_ptr():
                out.copy_(shard)
            return _Done()
        return dist.all_gather_into_tensor(out, shard, async_op=async_op)

    def broadcast(self, t: torch.Tensor, src: int = 0, async_op: bool = False):
        if not self.is_distributed:
            return _Done()
        return dist.broadcast(t, src=src, async_op=async_op)

    def barrier(self) -> None:
        if self.is_distributed:
            dist.barrier()

    def supports_reduce_scatter(self) -> bool:
        # gloo has no reduce_scatter_tensor; nccl(=RCCL) does.
        return self.backend == "nccl"

    def destroy(self) -> None:
        if dist.is_initialized():
            dist.destroy_process_group()


class _Done:
    """Completed-work stub for the world_size==1 fast path."""

    def wait(self) -> bool:
        return True

    def is_completed(self) -> bool:
        return True
