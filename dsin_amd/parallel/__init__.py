from .ddp import (FlatGradReducer, GradReducer, init_distributed,
                  is_distributed, rank, world_size)

__all__ = ["FlatGradReducer", "GradReducer", "init_distributed",
           "is_distributed", "world_size", "rank"]
