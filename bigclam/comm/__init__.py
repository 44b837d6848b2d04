from .dist import (
    all_reduce_,
    backend,
    all_to_all,
    barrier,
    get_rank,
    get_world_size,
    init_distributed,
    is_distributed,
)

__all__ = [
    "all_reduce_",
    "backend",
    "all_to_all",
    "barrier",
    "get_rank",
    "get_world_size",
    "init_distributed",
    "is_distributed",
]
