"""Training loop driver: two-optimizer step, LR schedules, autocast, DDP.

One step == the reference's ``siNet_update`` (src/AE.py:108-118): eval-mode
autoencoder pass over y (y_dec), full forward, combined loss, backward, and
BOTH Adam updates (Adam_AE over encoder/decoder/centers/siNet, Adam_PC over
the probclass group — src/AE.py:177-191) driven by one shared global step.
"""

from __future__ import annotations

from typing import Optional

import torch

from ..parallel import GradReducer, is_distributed
from .helpers import LRSchedule, create_optimizer, num_itr_per_epoch


class Trainer:
    def __init__(self, model, ae_config, pc_config, num_training_imgs: int,
                 device: Optional[torch.device] = None,
                 autocast_bf16: bool = False,
                 ddp_bucket_bytes: int = 8 * 1024 * 1024,
                 ddp_comm_dtype: Optional[torch.dtype] = None):
        self.model = model
        self.ae_config = ae_config
        self.device = device or next(model.parameters()).device
        self.autocast_bf16 = autocast_bf16 and self.device.type == "cuda"

        ae_params, pc_params = model.param_groups()
        self.opt_ae = create_optimizer(ae_config, ae_params)
        self.opt_pc = create_optimizer(pc_config, pc_params)
        batch = ae_config.batch_size if ae_config.AE_only else 1
        itr_ep = num_itr_per_epoch(ae_config.num_crops_per_img, batch,
                                   num_training_imgs, ae_config.AE_only)
        self.sched_ae = LRSchedule(ae_config, self.opt_ae, itr_ep)
        self.sched_pc = LRSchedule(pc_config, self.opt_pc, itr_ep)
        self.global_step = 0

        self.reducer = GradReducer(ae_params + pc_params,
                                   bucket_bytes=ddp_bucket_bytes,
                                   comm_dtype=ddp_comm_dtype)
        if is_distributed():
            self.reducer.broadcast_params()

    @property
    def optimizers(self):
        return [self.opt_ae, self.opt_pc]

    def _autocast(self):
        if self.autocast_bf16:
            return torch.autocast(device_type="cuda", dtype=torch.bfloat16)
        import contextlib
        return contextlib.nullcontext()

    def train_step(self, x: torch.Tensor, y: Optional[torch.Tensor]):
        """Returns (loss, bpp) floats-on-device (no host sync)."""
        self.sched_ae.set_step(self.global_step)
        self.sched_pc.set_step(self.global_step)
        self.opt_ae.zero_grad(set_to_none=True)
        self.opt_pc.zero_grad(set_to_none=True)
        self.reducer.prepare()
        with self._autocast():
            out = self.model.train_losses(x, y)
        out["loss"].backward()
        self.reducer.finalize()
        self.opt_ae.step()
        self.opt_pc.step()
        self.global_step += 1
        return out["loss"].detach(), out["bpp"].detach()

    def validate(self, x: torch.Tensor, y: Optional[torch.Tensor]) -> torch.Tensor:
        with self._autocast():
            return self.model.validate_loss(x, y)
