"""Training loop driver: two-optimizer step, LR schedules, autocast, DDP,
and whole-step hipGraph capture.

One step == the reference's ``siNet_update`` (src/AE.py:108-118): eval-mode
autoencoder pass over y (y_dec), full forward, combined loss, backward, and
BOTH Adam updates (Adam_AE over encoder/decoder/centers/siNet, Adam_PC over
the probclass group — src/AE.py:177-191) driven by one shared global step.

hipGraphs: the eager step is ~1400 kernel launches; on MI355X the launch gap
dominates (see profiles/r01_eager_step_breakdown.md). With
``use_cuda_graph=True`` the whole step (zero-grad, forward, backward, both
Adam updates) is captured once after a few warmup steps and replayed; inputs
are copied into static buffers and the learning rate lives in a device
tensor (capturable Adam) so the staircase schedule still applies across
replays. Capture failure falls back to eager with a warning.
"""

from __future__ import annotations

import contextlib
import warnings
from typing import Optional

import torch

from ..parallel import FlatGradReducer, GradReducer, is_distributed
from .helpers import LRSchedule, create_optimizer, num_itr_per_epoch


def _make_optimizer(config, params, device, capturable: bool,
                    fused: Optional[bool] = None):
    kind = config.optimizer
    lr = float(config.lr_initial)
    if kind == "ADAM":
        use_fused = fused
        if use_fused is None:
            if device.type == "cuda":
                from ..ops import hip_available
                use_fused = hip_available()
            else:
                use_fused = False
        if use_fused:
            from ..ops.adam import FusedAdam
            return FusedAdam(params, lr=lr, betas=(0.9, 0.999), eps=1e-8)
        if capturable:
            lr_t = torch.tensor(lr, device=device)
            return torch.optim.Adam(params, lr=lr_t, betas=(0.9, 0.999),
                                    eps=1e-8, capturable=True, foreach=True)
        return torch.optim.Adam(params, lr=lr, betas=(0.9, 0.999), eps=1e-8,
                                foreach=True)
    return create_optimizer(config, params)


class LRScheduleT(LRSchedule):
    """LRSchedule that also handles tensor learning rates (graph mode)."""

    _last_lr: Optional[float] = None

    def set_step(self, step: int) -> float:
        from .helpers import lr_at_step
        lr = lr_at_step(self.config, step, self.itr_per_epoch)
        if lr == self._last_lr:
            return lr  # staircase: unchanged almost every step; no device op
        self._last_lr = lr
        for group in self.optimizer.param_groups:
            if isinstance(group["lr"], torch.Tensor):
                group["lr"].fill_(lr)
            else:
                group["lr"] = lr
        return lr


class Trainer:
    def __init__(self, model, ae_config, pc_config, num_training_imgs: int,
                 device: Optional[torch.device] = None,
                 autocast_bf16: bool = False,
                 use_cuda_graph: bool = False,
                 graph_warmup: int = 3,
                 ddp_bucket_bytes: int = 8 * 1024 * 1024,
                 ddp_comm_dtype: Optional[torch.dtype] = None,
                 fused_adam: Optional[bool] = None):
        self.model = model
        self.ae_config = ae_config
        self.device = device or next(model.parameters()).device
        self.autocast_bf16 = autocast_bf16 and self.device.type == "cuda"
        self.use_cuda_graph = use_cuda_graph and self.device.type == "cuda"
        self.graph_warmup = max(graph_warmup, 2)

        ae_params, pc_params = model.param_groups()
        self.opt_ae = _make_optimizer(ae_config, ae_params, self.device,
                                      self.use_cuda_graph, fused_adam)
        self.opt_pc = _make_optimizer(pc_config, pc_params, self.device,
                                      self.use_cuda_graph, fused_adam)
        batch = ae_config.batch_size if ae_config.AE_only else 1
        itr_ep = num_itr_per_epoch(ae_config.num_crops_per_img, batch,
                                   num_training_imgs, ae_config.AE_only)
        self.sched_ae = LRScheduleT(ae_config, self.opt_ae, itr_ep)
        self.sched_pc = LRScheduleT(pc_config, self.opt_pc, itr_ep)
        self.global_step = 0

        from ..ops.adam import FusedAdam
        self._fused = isinstance(self.opt_ae, FusedAdam)
        self._flat_reducers = []
        if self._fused:
            self._delegate_reg_to_optimizer(model)
            self.reducer = GradReducer([])  # inactive
            if is_distributed():
                # flat gradient buffers double as the DDP communication
                # buffers; buckets all-reduce as backward fills them
                self._flat_reducers = [
                    FlatGradReducer(self.opt_ae, bucket_bytes=ddp_bucket_bytes),
                    FlatGradReducer(self.opt_pc, bucket_bytes=ddp_bucket_bytes),
                ]
        else:
            self.reducer = GradReducer(ae_params + pc_params,
                                       bucket_bytes=ddp_bucket_bytes,
                                       comm_dtype=ddp_comm_dtype)
        if is_distributed():
            if self._fused:
                import torch.distributed as dist
                for opt in (self.opt_ae, self.opt_pc):
                    dist.broadcast(opt.flat_p, src=0)
            else:
                self.reducer.broadcast_params()

        self._graph = None
        self._graph_failed = False
        self._static_x = None
        self._static_y = None
        self._static_out = None

    @property
    def optimizers(self):
        return [self.opt_ae, self.opt_pc]

    def _autocast(self):
        if self.autocast_bf16:
            # autocast's weight-cast cache is incompatible with graph capture
            return torch.autocast(device_type="cuda", dtype=torch.bfloat16,
                                  cache_enabled=not self.use_cuda_graph)
        return contextlib.nullcontext()


    def _delegate_reg_to_optimizer(self, model) -> None:
        """Fold the model's factor/2*sum(w^2) regularizers into the fused
        Adam step as per-element L2 factors (mathematically identical
        update; see ops/adam.py). The loss-side value is then computed from
        the optimizer's flat buffer instead of a ~100-kernel autograd
        subgraph per step."""
        import torch.nn as nn
        ok = True
        if getattr(model, "reg_factor", 0.0):
            for mod in (model.encoder, model.decoder):
                for m in mod.modules():
                    if isinstance(m, (nn.Conv2d, nn.ConvTranspose2d)):
                        ok &= self.opt_ae.set_weight_decay(
                            m.weight, model.reg_factor)
        q = model.encoder.quantizer
        if q.reg_factor:
            ok &= self.opt_ae.set_weight_decay(q.centers, q.reg_factor)
        pc = model.probclass
        if pc.reg_factor is not None:
            for mod in (pc.conv0, pc.res_conv1, pc.res_conv2, pc.conv2):
                ok &= self.opt_pc.set_weight_decay(mod.weight,
                                                   float(pc.reg_factor))
        if ok:
            model.reg_value_fn = lambda: (self.opt_ae.reg_value()
                                          + self.opt_pc.reg_value())

    def _step_inner(self, x: torch.Tensor, y: Optional[torch.Tensor]):
        """zero-grad + forward + backward + both optimizer steps. This is
        what gets graph-captured; it must stay free of host syncs."""
        # bump the W-panel cache INSIDE the captured region: each step (and
        # each graph replay) must rebuild panels from the post-optimizer
        # weights — a bump outside _step_inner would let capture record a
        # cache hit and replay stale panels forever
        from ..ops import conv as _conv
        _conv.begin_step()
        self.opt_ae.zero_grad(set_to_none=not self.use_cuda_graph
                              or self._fused)
        self.opt_pc.zero_grad(set_to_none=not self.use_cuda_graph
                              or self._fused)
        self.reducer.prepare()
        for r in self._flat_reducers:
            r.prepare()
        with self._autocast():
            out = self.model.train_losses(x, y)
        out["loss"].backward()
        if self._flat_reducers:
            # hooks gathered each bucket into flat_g and launched its
            # all-reduce during backward; flush stragglers, wait, average
            for r in self._flat_reducers:
                r.finalize()
        elif self._fused:
            # stolen per-tensor grads -> flat buffers (one fused copy)
            self.opt_ae.gather_grads()
            self.opt_pc.gather_grads()
        else:
            self.reducer.finalize()
        self.opt_ae.step()
        self.opt_pc.step()
        return out["loss"].detach(), out["bpp"].detach()

    def _try_capture(self, x, y):
        # NOTE: capture setup performs two real warmup optimizer updates on
        # the capture batch and the first replay trains on it once more, so
        # that one batch is trained on three consecutive times (each counted
        # in global_step). A one-off distribution artifact of enabling
        # use_cuda_graph, negligible over a training run.
        try:
            self._static_x = x.clone()
            self._static_y = y.clone() if y is not None else None
            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(side):
                for _ in range(2):
                    # real parameter updates on the capture input
                    self._step_inner(self._static_x, self._static_y)
                    self.global_step += 1
            torch.cuda.current_stream().wait_stream(side)
            graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(graph):
                self._static_out = self._step_inner(self._static_x, self._static_y)
            self._graph = graph
        except Exception as e:  # pragma: no cover - device specific
            import traceback
            warnings.warn("hipGraph capture failed, staying eager: "
                          + "".join(traceback.format_exception(e)))
            self._graph_failed = True
            self._graph = None

    def train_step(self, x: torch.Tensor, y: Optional[torch.Tensor]):
        """Returns (loss, bpp) tensors on device (no host sync)."""
        self.sched_ae.set_step(self.global_step)
        self.sched_pc.set_step(self.global_step)

        if self._graph is not None:
            self._static_x.copy_(x, non_blocking=True)
            if self._static_y is not None:
                self._static_y.copy_(y, non_blocking=True)
            self._graph.replay()
            self.global_step += 1
            return self._static_out

        if (self.use_cuda_graph and not self._graph_failed
                and self.global_step >= self.graph_warmup):
            self._try_capture(x, y)
            if self._graph is not None:
                # the two warmup iterations inside capture setup advanced the
                # model; count this call as one replayed step
                self._static_x.copy_(x, non_blocking=True)
                if self._static_y is not None:
                    self._static_y.copy_(y, non_blocking=True)
                self._graph.replay()
                self.global_step += 1
                return self._static_out

        out = self._step_inner(x, y)
        self.global_step += 1
        return out

    def validate(self, x: torch.Tensor, y: Optional[torch.Tensor]) -> torch.Tensor:
        from ..ops import conv as _conv
        _conv.begin_step()  # weights may have stepped since panels were built
        with self._autocast():
            return self.model.validate_loss(x, y)
