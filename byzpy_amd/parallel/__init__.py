from byzpy_amd.parallel.dist import (
    all_gather_rows,
    get_rank,
    get_world_size,
    init_from_env,
    is_initialized,
)

__all__ = [
    "init_from_env",
    "is_initialized",
    "get_rank",
    "get_world_size",
    "all_gather_rows",
]
