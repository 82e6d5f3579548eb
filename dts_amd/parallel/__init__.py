from dts_amd.parallel.dist import (
    get_local_rank,
    get_rank,
    get_world_size,
    init_distributed,
    is_distributed,
)

__all__ = [
    "init_distributed",
    "get_rank",
    "get_world_size",
    "get_local_rank",
    "is_distributed",
]
