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

    def all_reduce_bucketed_(self, tensors, bucket_bytes: int = 64 << 20
                             ) -> None:
        """All-reduce a list of tensors in flat buckets.

        xGMI is per-link bound (7 p2p links x ~153 GB/s), so buckets
        should be large enough to amortize ring latency but small
        enough to overlap with producers; 64 MiB default.  Buckets are
        flattened/coalesced then scattered back (the DDP gradient
        pattern, without requiring torch DDP)."""
        if not self.is_distributed:
            return
        import torch

        bucket, size = [], 0
        for t in list(tensors) + [None]:
            # flush on size overflow AND on dtype change (torch.cat
            # cannot mix dtypes; mixed fp32/bf16 grads are common)
            flush = t is None or (bucket and (
                size + t.numel() * t.element_size() > bucket_bytes
                or t.dtype != bucket[0].dtype
                or t.device != bucket[0].device))
            if flush and bucket:
                flat = torch.cat([b.reshape(-1) for b in bucket])
                self._dist.all_reduce(flat)
                off = 0
                for b in bucket:
                    n = b.numel()
                    b.copy_(flat[off:off + n].view_as(b))
                    off += n
                bucket, size = [], 0
            if t is not None:
                bucket.append(t)
                size += t.numel() * t.element_size()

    def all_gather(self, tensor):
        """Gather each rank's tensor -> list of world tensors."""
        import torch

        if not self.is_distributed:
            return [tensor]
        out = [torch.empty_like(tensor) for _ in range(self.world)]
        self._dist.all_gather(out, tensor)
        return out

    def reduce_scatter_(self, out, shards) -> None:
        """Reduce a per-rank list of shards, scattering shard i to
        rank i (ZeRO/FSDP pattern)."""
        if not self.is_distributed:
            out.copy_(shards[self.rank])
            return
        self._dist.reduce_scatter(out, list(shards))

    def all_to_all(self, shards):
        """Exchange shard i with rank i (EP dispatch pattern).

        RCCL supports all_to_all natively; gloo does not, so the CPU
        test path emulates it with an all_gather of the stacked shards
        (correct, just not bandwidth-optimal — fine for gloo tests)."""
        import torch

        if not self.is_distributed:
            return list(shards)
        if self.backend == "nccl":
            out = [torch.empty_like(s) for s in shards]
            self._dist.all_to_all(out, list(shards))
            return out
        stacked = torch.stack(list(shards))  # [world, ...]
        gathered = [torch.empty_like(stacked) for _ in range(self.world)]
        self._dist.all_gather(gathered, stacked)
        return [gathered[src][self.rank].clone()
                for src in range(self.world)]

    def broadcast_(self, tensor, src: int = 0) -> None:
        if self.is_distributed:
            self._dist.broadcast(tensor, src=src)

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
