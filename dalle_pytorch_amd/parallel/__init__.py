from dalle_pytorch_amd.parallel.engine import (
    DataParallelEngine, init_distributed, is_distributed, get_rank,
    get_world_size, get_local_rank, barrier, average_scalar)

__all__ = ['DataParallelEngine', 'init_distributed', 'is_distributed',
           'get_rank', 'get_world_size', 'get_local_rank', 'barrier',
           'average_scalar']
