from .ddp import GradReducer, init_distributed, is_distributed, world_size, rank

__all__ = ["GradReducer", "init_distributed", "is_distributed", "world_size", "rank"]
