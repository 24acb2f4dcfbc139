from .runtime import (  # noqa: F401
    ComputeConfig,
    all_gather_object,
    barrier,
    broadcast_model,
    get_rank,
    get_world_size,
    init_distributed,
    shard_list,
)
