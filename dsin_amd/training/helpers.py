"""Learning-rate schedules and optimizer factory.

Mirror of /root/reference/src/training_helpers_imgcomp.py:
  * staircase exponential decay with decay_steps = iters-per-epoch *
    decay_interval (:22-35), where iters-per-epoch = num_training_imgs //
    (batch_size // num_crops_per_img) and AE_only hardcodes the 1,281,000
    ImageNet epoch (:51-60);
  * optimizer :: ADAM | SGD | MOMENTUM(nesterov) (:38-48).
"""

from __future__ import annotations

from typing import Iterable

import torch

IMAGENET_EPOCH = 1_281_000  # reference :56


def num_itr_per_epoch(num_crops_per_img: int, batch_size: int,
                      num_training_imgs: int, ae_only: bool) -> int:
    num_unique_imgs_per_batch = max(batch_size // num_crops_per_img, 1)
    if ae_only:
        num_training_imgs = IMAGENET_EPOCH
    return num_training_imgs // num_unique_imgs_per_batch


def lr_at_step(config, step: int, itr_per_epoch: int) -> float:
    lr = float(config.lr_initial)
    if config.lr_schedule == "FIXED":
        return lr
    if config.lr_schedule == "DECAY":
        decay_steps = itr_per_epoch * int(config.lr_schedule_decay_interval)
        exponent = step / decay_steps
        if config.lr_schedule_decay_staircase:
            exponent = float(int(exponent))
        return lr * float(config.lr_schedule_decay_rate) ** exponent
    raise ValueError(f"invalid lr_schedule {config.lr_schedule}")


def create_optimizer(config, params: Iterable[torch.nn.Parameter]) -> torch.optim.Optimizer:
    lr = float(config.lr_initial)
    kind = config.optimizer
    if kind == "ADAM":
        # TF AdamOptimizer defaults: beta1=.9, beta2=.999, eps=1e-8
        return torch.optim.Adam(params, lr=lr, betas=(0.9, 0.999), eps=1e-8)
    if kind == "SGD":
        return torch.optim.SGD(params, lr=lr)
    if kind == "MOMENTUM":
        return torch.optim.SGD(params, lr=lr, momentum=float(config.optimizer_momentum),
                               nesterov=True)
    raise ValueError(f"invalid optimizer {kind}")


class LRSchedule:
    """Applies lr_at_step to an optimizer each step (staircase decay is a
    closed-form function of the global step, so resume is trivial)."""

    def __init__(self, config, optimizer: torch.optim.Optimizer, itr_per_epoch: int):
        self.config = config
        self.optimizer = optimizer
        self.itr_per_epoch = itr_per_epoch

    def set_step(self, step: int) -> float:
        lr = lr_at_step(self.config, step, self.itr_per_epoch)
        for group in self.optimizer.param_groups:
            group["lr"] = lr
        return lr
