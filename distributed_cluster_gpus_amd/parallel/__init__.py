from .dist import (init_distributed, is_distributed, world_size, rank,
                   allreduce_gradients, allreduce_scalar, allreduce_tensor_sum,
                   broadcast_module, barrier)
from .sharding import replica_shard

__all__ = ["init_distributed", "is_distributed", "world_size", "rank",
           "allreduce_gradients", "allreduce_scalar", "allreduce_tensor_sum",
           "broadcast_module", "barrier", "replica_shard"]
