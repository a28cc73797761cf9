from .misc import set_seed, init_distributed, get_rank, get_world_size, StepTimer

__all__ = ["set_seed", "init_distributed", "get_rank", "get_world_size", "StepTimer"]
