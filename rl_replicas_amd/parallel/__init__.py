from .ddp import (
    all_reduce_gradients,
    all_reduce_mean_scalar,
    distributed_is_active,
    enable_data_parallel,
    get_rank,
    get_world_size,
    global_normalize,
    init_from_env,
)

__all__ = [
    "all_reduce_gradients",
    "all_reduce_mean_scalar",
    "distributed_is_active",
    "enable_data_parallel",
    "get_rank",
    "get_world_size",
    "global_normalize",
    "init_from_env",
]
