from .launch import init_distributed, is_main, get_rank, get_world_size, barrier
from .ddp import GradReducer, average_parameters

__all__ = [
    "init_distributed",
    "is_main",
    "get_rank",
    "get_world_size",
    "barrier",
    "GradReducer",
    "average_parameters",
]
