"""Cluster topology discovery (core/utils/ClusterUtil.scala parity).

The reference interrogates the BlockManager for executors×cores and parses
`local[k]` masters (ClusterUtil.getNumTasksPerExecutor:21, getDriverHost:111,
getExecutors:128, getNumExecutorTasks:143).  The MI355X runtime is
one-process-per-GPU gang-launched by torchrun, so topology is a pure
function of the torch.distributed environment plus the local device count —
no RPC needed.
"""
from __future__ import annotations

import os
import socket
from dataclasses import dataclass

import torch


@dataclass(frozen=True)
class ClusterTopology:
    world_size: int      # total ranks (= total GPUs in the job)
    rank: int            # this process's global rank
    local_rank: int      # rank within this node
    local_world_size: int  # ranks on this node (= GPUs used per node)
    n_nodes: int         # world_size / local_world_size
    gpus_visible: int    # torch.cuda.device_count() here
    driver_host: str     # rendezvous master (MASTER_ADDR)
    hostname: str


def get_topology() -> ClusterTopology:
    """Topology from the torchrun env (env:// rendezvous variables); sane
    single-process defaults when launched without torchrun."""
    ws = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    lws = int(os.environ.get("LOCAL_WORLD_SIZE", str(ws)))
    try:
        visible = torch.cuda.device_count()
    except Exception:  # pragma: no cover - broken runtime
        visible = 0
    return ClusterTopology(
        world_size=ws,
        rank=rank,
        local_rank=local_rank,
        local_world_size=max(lws, 1),
        n_nodes=max(ws // max(lws, 1), 1),
        gpus_visible=visible,
        driver_host=os.environ.get("MASTER_ADDR", "127.0.0.1"),
        hostname=socket.gethostname(),
    )


def get_num_executors() -> int:
    """ClusterUtil.getExecutors analog: number of worker processes."""
    return get_topology().world_size


def get_num_tasks_per_executor() -> int:
    """One GPU == one task slot per process on MI355X."""
    return 1


def get_driver_host() -> str:
    """ClusterUtil.getDriverHost analog: the rendezvous master address."""
    return get_topology().driver_host
