from .dist import (
    init_distributed_mode, is_dist, get_rank, get_world_size, get_local_rank,
    is_main_process, barrier, all_reduce_mean, setup_for_distributed,
)
from .ddp import GradBucketAllReduce

__all__ = [
    "init_distributed_mode", "is_dist", "get_rank", "get_world_size",
    "get_local_rank", "is_main_process", "barrier", "all_reduce_mean",
    "setup_for_distributed", "GradBucketAllReduce",
]
