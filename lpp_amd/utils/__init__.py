from .misc import (set_seed, init_distributed, get_rank, get_world_size, StepTimer,
                   rank_zero_first)
from .timers import DeviceTimers

__all__ = ["set_seed", "init_distributed", "get_rank", "get_world_size", "StepTimer",
           "rank_zero_first", "DeviceTimers"]
