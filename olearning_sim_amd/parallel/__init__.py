from .dist import (init_distributed, is_distributed, get_rank,
                   get_world_size, all_reduce_flat, barrier, DistContext)
from .sharding import shard_clients

__all__ = ["init_distributed", "is_distributed", "get_rank",
           "get_world_size", "all_reduce_flat", "barrier", "DistContext",
           "shard_clients"]
