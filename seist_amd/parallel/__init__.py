from . import dist
from .dist import (
    barrier,
    broadcast_object,
    gather_tensors_to_list,
    get_local_rank,
    get_rank,
    get_world_size,
    init_distributed_mode,
    is_dist,
    is_main_process,
    reduce_tensor,
)
from .ddp import FlatReplica, wrap_distributed

__all__ = [
    "dist", "barrier", "broadcast_object", "gather_tensors_to_list",
    "get_local_rank", "get_rank", "get_world_size", "init_distributed_mode",
    "is_dist", "is_main_process", "reduce_tensor", "FlatReplica",
    "wrap_distributed",
]
