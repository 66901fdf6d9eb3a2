from .ddp import (
    init_distributed,
    is_distributed,
    get_rank,
    get_world_size,
    FlatGradBucket,
    all_reduce_scalar,
)

__all__ = [
    "init_distributed",
    "is_distributed",
    "get_rank",
    "get_world_size",
    "FlatGradBucket",
    "all_reduce_scalar",
]
