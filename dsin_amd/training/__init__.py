from .helpers import lr_at_step, num_itr_per_epoch, create_optimizer, LRSchedule
from .trainer import Trainer
from . import checkpoint

__all__ = ["lr_at_step", "num_itr_per_epoch", "create_optimizer", "LRSchedule",
           "Trainer", "checkpoint"]
