from .pg import (env_local_rank, env_rank, env_world_size, init_distributed,
                 rank0_first, rank_ordered)

__all__ = ["env_local_rank", "env_rank", "env_world_size",
           "init_distributed", "rank0_first", "rank_ordered"]
