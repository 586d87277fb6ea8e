from .dist import (all_reduce_gradients, get_rank, get_world_size,
                   init_distributed_from_env, is_distributed)

__all__ = ["init_distributed_from_env", "is_distributed", "get_rank",
           "get_world_size", "all_reduce_gradients"]
