from .dist import (barrier, get_rank, get_world_size, init_distributed,
                   is_distributed, is_main_process, local_rank)
from .ddp import FlatReplica
from .optim import FlatAdamW

__all__ = [
    "init_distributed", "is_distributed", "get_rank", "get_world_size",
    "local_rank", "is_main_process", "barrier", "FlatReplica", "FlatAdamW",
]
