"""Communicator — RCCL over xGMI (GPU) / gloo (CPU tests) / no-op (1 proc).

Replaces the reference's three socket fabrics (LightGBM's
LGBM_NetworkInit TCP ring, VW's ClusterSpanningTree AllReduce, and the
driver ServerSocket rendezvous of LightGBMBase.scala:392-430) with a single
``torch.distributed`` process group: backend "nccl" IS RCCL on ROCm, and the
rendezvous is torchrun's env:// exchange (the analog of the driver
socket rendezvous).  One process per GPU; collectives run over the 7
point-to-point xGMI links per GPU.
"""
from __future__ import annotations

import datetime
import os
from typing import List, Optional

import torch
import torch.distributed as dist


class Comm:
    """Thin collective wrapper; a no-op when world_size == 1.

    All GBDT/VW sync points go through this object so distributed logic is
    testable with gloo/world_size=2 on CPU and identical on RCCL/GPU.
    """

    def __init__(self, group: Optional[object] = None):
        self.group = group
        self._initialized = dist.is_available() and dist.is_initialized()

    # --- topology ---------------------------------------------------------
    @property
    def rank(self) -> int:
        return dist.get_rank(self.group) if self._initialized else 0

    @property
    def world_size(self) -> int:
        return dist.get_world_size(self.group) if self._initialized else 1

    @property
    def is_distributed(self) -> bool:
        return self._initialized and self.world_size > 1

    def native_group(self):
        """The underlying c10d ProcessGroup object, for native (C++) code
        that calls collectives GIL-free (e.g. the GBDT grower's per-split
        histogram all_reduce).  None when not distributed."""
        if not self.is_distributed:
            return None
        return self.group if self.group is not None else dist.group.WORLD

    # --- collectives ------------------------------------------------------
    def all_reduce(self, t: torch.Tensor, op: str = "sum") -> torch.Tensor:
        if self.is_distributed:
            ops = {"sum": dist.ReduceOp.SUM, "max": dist.ReduceOp.MAX,
                   "min": dist.ReduceOp.MIN}
            dist.all_reduce(t, op=ops[op], group=self.group)
        return t

    def all_gather(self, t: torch.Tensor) -> List[torch.Tensor]:
        if not self.is_distributed:
            return [t]
        out = [torch.empty_like(t) for _ in range(self.world_size)]
        dist.all_gather(out, t.contiguous(), group=self.group)
        return out

    def all_gather_object(self, obj) -> list:
        if not self.is_distributed:
            return [obj]
        out = [None] * self.world_size
        dist.all_gather_object(out, obj, group=self.group)
        return out

    def broadcast(self, t: torch.Tensor, src: int = 0) -> torch.Tensor:
        if self.is_distributed:
            dist.broadcast(t, src=src, group=self.group)
        return t

    def reduce_scatter(self, out: torch.Tensor, shards: List[torch.Tensor]):
        if self.is_distributed:
            dist.reduce_scatter(out, [s.contiguous() for s in shards],
                                group=self.group)
        else:
            out.copy_(shards[0])
        return out

    def barrier(self):
        if self.is_distributed:
            if torch.cuda.is_available():
                dist.barrier(group=self.group, device_ids=[torch.cuda.current_device()])
            else:
                dist.barrier(group=self.group)


_GLOBAL_COMM: Optional[Comm] = None


def init_from_env(timeout_s: int = 600) -> Comm:
    """Initialize the process group from torchrun env vars (idempotent).

    RANK/WORLD_SIZE/MASTER_ADDR/MASTER_PORT in the env; backend nccl(=RCCL)
    if a GPU is visible, gloo otherwise.  Also pins this process to its GPU
    (one process per GPU — the MI355X analog of "one Spark task per
    executor core" in the reference's barrier execution mode).
    """
    global _GLOBAL_COMM
    if _GLOBAL_COMM is not None:
        return _GLOBAL_COMM
    if dist.is_available() and not dist.is_initialized() and "RANK" in os.environ \
            and "WORLD_SIZE" in os.environ and int(os.environ["WORLD_SIZE"]) > 1:
        # MMLSPARK_AMD_BACKEND=gloo lets multi-rank runs share one GPU for
        # rehearsal (RCCL refuses duplicate devices); default is RCCL on GPU
        backend = os.environ.get("MMLSPARK_AMD_BACKEND") or (
            "nccl" if torch.cuda.is_available() else "gloo")
        if torch.cuda.is_available():
            local_rank = int(os.environ.get("LOCAL_RANK", os.environ["RANK"]))
            torch.cuda.set_device(local_rank % torch.cuda.device_count())
        dist.init_process_group(backend=backend,
                                timeout=datetime.timedelta(seconds=timeout_s))
    _GLOBAL_COMM = Comm()
    return _GLOBAL_COMM


def get_comm() -> Comm:
    return _GLOBAL_COMM if _GLOBAL_COMM is not None else init_from_env()
