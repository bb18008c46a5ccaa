"""Fault-tolerant data sharding helpers (reference parity: torchft/data.py).

Shards data across ``num_replicas × num_replica_groups``; the global shard
rank is ``group_rank + num_replicas * replica_rank``. Inherently lossy under
fault tolerance (dropped batches on rejoin) — use a stateful dataloader for
exactly-once needs.
"""

from __future__ import annotations

from typing import Optional

import torch.distributed as dist
from torch.utils import data


class DistributedSampler(data.distributed.DistributedSampler):
    def __init__(
        self,
        dataset: data.Dataset,
        replica_rank: int,
        num_replica_groups: int,
        group_rank: Optional[int] = None,
        num_replicas: Optional[int] = None,
        **kwargs: object,
    ) -> None:
        if group_rank is None:
            group_rank = dist.get_rank()
        if num_replicas is None:
            num_replicas = dist.get_world_size()

        self.global_rank: int = group_rank + num_replicas * replica_rank
        self.global_world_size: int = num_replicas * num_replica_groups

        super().__init__(
            dataset,
            rank=self.global_rank,
            num_replicas=self.global_world_size,
            **kwargs,  # pyre-ignore[6]
        )
