from fei_amd.parallel.pg import (
    ParallelContext, all_gather_cat, all_reduce_sum, get_world, init_from_env,
)

__all__ = ["ParallelContext", "all_gather_cat", "all_reduce_sum", "get_world",
           "init_from_env"]
