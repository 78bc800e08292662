from .flat import FlatParams
from .dist import (all_reduce_flat, barrier, get_rank, get_world_size,
                   init_distributed, is_distributed)

__all__ = ["FlatParams", "init_distributed", "all_reduce_flat", "barrier",
           "get_rank", "get_world_size", "is_distributed"]
