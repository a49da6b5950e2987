"""Replica bookkeeping for sharded data loading.

Parity with reference replay/data/nn/parquet/info/: DistributedInfo
(distributed_info.py:6-29 — rank/world from torch.distributed), WorkerInfo
(worker_info.py:15 — dataloader worker id/count), replica count =
workers x world (replicas.py:11).  All three are injectable protocols so
sharding logic is unit-testable single-process (the reference test pattern,
SURVEY §4).
"""

from __future__ import annotations

from dataclasses import dataclass


@dataclass
class DistributedInfo:
    rank: int = 0
    world_size: int = 1

    @classmethod
    def from_env(cls) -> "DistributedInfo":
        import torch.distributed as dist

        if dist.is_available() and dist.is_initialized():
            return cls(rank=dist.get_rank(), world_size=dist.get_world_size())
        return cls()


@dataclass
class WorkerInfo:
    worker_id: int = 0
    num_workers: int = 1

    @classmethod
    def from_env(cls) -> "WorkerInfo":
        import torch.utils.data

        info = torch.utils.data.get_worker_info()
        if info is None:
            return cls()
        return cls(worker_id=info.id, num_workers=info.num_workers)


@dataclass
class ReplicasInfo:
    """Flattened replica = worker x rank (reference replicas.py:11)."""

    curr_replica: int = 0
    num_replicas: int = 1

    @classmethod
    def from_env(cls) -> "ReplicasInfo":
        d = DistributedInfo.from_env()
        w = WorkerInfo.from_env()
        return cls(
            curr_replica=d.rank * w.num_workers + w.worker_id,
            num_replicas=d.world_size * w.num_workers,
        )
