"""Replica sharding across GPUs.

Monte-Carlo replicas are embarrassingly parallel (SURVEY §2 "Replica
sharding"): each rank owns a contiguous slab; per-replica RNG streams are
keyed on the GLOBAL replica id so results are independent of the GPU count.
"""
from dataclasses import dataclass


@dataclass(frozen=True)
class ReplicaShard:
    global_replicas: int
    start: int       # first global replica id owned by this rank
    count: int       # replicas on this rank

    @property
    def end(self) -> int:
        return self.start + self.count


def replica_shard(total_replicas: int, rank: int, world: int) -> ReplicaShard:
    base = total_replicas // world
    extra = total_replicas % world
    start = rank * base + min(rank, extra)
    count = base + (1 if rank < extra else 0)
    return ReplicaShard(global_replicas=total_replicas, start=start, count=count)
