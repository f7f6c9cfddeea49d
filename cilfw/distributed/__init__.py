from .init import (init_distributed_mode, is_main_process, get_rank,
                   get_world_size, setup_for_distributed, barrier)
from .ddp import DataParallelEngine

__all__ = ["init_distributed_mode", "is_main_process", "get_rank",
           "get_world_size", "setup_for_distributed", "barrier",
           "DataParallelEngine"]
