"""Gang collectives over RCCL/xGMI (or gloo on CPU test hosts).

Design notes (MI355X):
  * one process per GPU, ``torch.distributed`` with backend "nccl"
    (which is RCCL on ROCm) over xGMI;
  * xGMI is point-to-point — 7 links x ~153 GB/s per GPU in the 8-GPU
    hive — so ring all-reduce is per-link bound; RCCL's multi-ring
    schedules use all links.  Bus-bandwidth accounting follows the
    nccl-tests convention: busbw = algbw * 2*(N-1)/N.
  * ``HSA_ENABLE_IPC_MODE_LEGACY=0`` must be exported for multi-process
    GPU work (dmabuf IPC; see repo README).
"""
from __future__ import annotations

import datetime
import os
from typing import Optional


def bus_bandwidth_gbps(payload_bytes: int, seconds: float, world: int) -> float:
    """nccl-tests bus bandwidth for an all-reduce of ``payload_bytes``."""
    if seconds <= 0:
        return 0.0
    alg = payload_bytes / seconds / 1e9
    if world <= 1:
        return alg
    return alg * 2.0 * (world - 1) / world


class GangComm:
    """A rank's view of its gang (thin wrapper over torch.distributed).

    Reads the torchrun/gang-launcher env contract (RANK, WORLD_SIZE,
    LOCAL_RANK, MASTER_ADDR, MASTER_PORT).  Replaces the reference's
    ``$AZ_BATCH_HOST_LIST``-based MPI bootstrap
    (reference convoy/batch.py:4590-4698).
    """

    def __init__(self, backend: Optional[str] = None,
                 timeout_s: float = 300.0) -> None:
        import torch
        import torch.distributed as dist

        self.rank = int(os.environ.get("RANK", "0"))
        self.world = int(os.environ.get("WORLD_SIZE", "1"))
        self.local_rank = int(os.environ.get("LOCAL_RANK", str(self.rank)))
        self.device = None
        if torch.cuda.is_available():
            torch.cuda.set_device(self.local_rank % torch.cuda.device_count())
            self.device = torch.device("cuda", torch.cuda.current_device())
        else:
            self.device = torch.device("cpu")
        self.backend = backend or ("nccl" if self.device.type == "cuda"
                                   else "gloo")
        self._dist = dist
        if self.world > 1 and not dist.is_initialized():
            os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
            os.environ.setdefault("MASTER_PORT", "29517")
            dist.init_process_group(
                backend=self.backend,
                rank=self.rank,
                world_size=self.world,
                timeout=datetime.timedelta(seconds=timeout_s),
            )

    @property
    def is_distributed(self) -> bool:
        return self.world > 1

    def all_reduce_(self, tensor) -> None:
        if self.is_distributed:
            self._dist.all_reduce(tensor)

    def barrier(self) -> None:
        if self.is_distributed:
            if self.backend == "nccl":
                self._dist.barrier(device_ids=[self.device.index])
            else:
                self._dist.barrier()

    def max_scalar(self, value: float) -> float:
        """MAX over ranks of a host scalar (for timing aggregation)."""
        if not self.is_distributed:
            return value
        import torch

        t = torch.tensor([value], dtype=torch.float64,
                         device=self.device if self.backend == "nccl"
                         else "cpu")
        self._dist.all_reduce(t, op=self._dist.ReduceOp.MAX)
        return float(t.item())

    def shutdown(self) -> None:
        if self.is_distributed and self._dist.is_initialized():
            self._dist.destroy_process_group()
