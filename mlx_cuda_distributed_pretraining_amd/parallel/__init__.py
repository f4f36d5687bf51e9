from .dist import (
    all_reduce_scalar,
    barrier,
    get_rank,
    get_world_size,
    init_distributed,
    is_distributed,
)
from .flat import FlatParamSpace
from .ddp import DataParallelGrads

__all__ = [
    "init_distributed", "is_distributed", "get_rank", "get_world_size",
    "all_reduce_scalar", "barrier", "FlatParamSpace", "DataParallelGrads",
]
