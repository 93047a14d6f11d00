from .ddp import (
    DataParallelEngine, init_distributed, is_distributed, get_rank,
    get_world_size, all_reduce_mean,
)
