from midgpt_amd.parallel.dist import (  # noqa: F401
    init_distributed, is_main, get_rank, get_world_size, barrier,
    reduce_scatter_flat, all_gather_flat, all_reduce_,
)
from midgpt_amd.parallel.engine import ShardedAdamW  # noqa: F401
