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
import os
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
                 fused_adam: Optional[bool] = None,
                 nan_guard: Optional[bool] = None,
                 blackbox_dir: str = "nan_blackbox"):
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

        # failure detection: opt-in per-step
        # non-finite guard that dumps a "black box" (failing batch, model +
        # optimizer state, per-component losses, per-buffer corruption map)
        # and raises — the diagnostic for the open RD-run stability item
        # (profiles/r02_rd_curve.md). Costs one host sync per step, so it is
        # off by default; enable with nan_guard=True or DSIN_NANCHECK=1.
        if nan_guard is None:
            nan_guard = os.environ.get("DSIN_NANCHECK", "0") == "1"
        self.nan_guard = bool(nan_guard)
        self.blackbox_dir = blackbox_dir

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
        """Returns (loss, bpp) tensors on device (no host sync unless
        nan_guard is enabled)."""
        self.sched_ae.set_step(self.global_step)
        self.sched_pc.set_step(self.global_step)

        if self._graph is not None:
            self._static_x.copy_(x, non_blocking=True)
            if self._static_y is not None:
                self._static_y.copy_(y, non_blocking=True)
            self._graph.replay()
            self.global_step += 1
            if self.nan_guard:
                self._check_finite(self._static_out[0], x, y)
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
                if self.nan_guard:
                    self._check_finite(self._static_out[0], x, y)
                return self._static_out

        out = self._step_inner(x, y)
        self.global_step += 1
        if self.nan_guard:
            self._check_finite(out[0], x, y)
        return out

    # -- failure detection --------------------------------------------------

    def _check_finite(self, loss: torch.Tensor, x, y) -> None:
        if bool(torch.isfinite(loss)):
            return
        path = self._dump_blackbox(x, y)
        raise RuntimeError(
            f"non-finite training loss at global step {self.global_step}; "
            f"black box saved to {path}")

    @staticmethod
    def _buf_stats(t: Optional[torch.Tensor]) -> dict:
        if t is None or not isinstance(t, torch.Tensor) or t.numel() == 0:
            return {}
        tf = t.detach().float()
        bad = (~torch.isfinite(tf)).sum()
        return {"numel": t.numel(), "nonfinite": int(bad),
                "absmax": float(tf.nan_to_num_(posinf=0, neginf=0).abs().max())}

    @torch.no_grad()
    def _dump_blackbox(self, x, y) -> str:
        """Everything needed to replay the failing step offline: the batch,
        full model + optimizer state (post-step, i.e. possibly already
        corrupt), per-component loss values from an eager re-run, and a
        per-buffer corruption map locating the first non-finite storage."""
        os.makedirs(self.blackbox_dir, exist_ok=True)
        step = self.global_step
        # snapshot state BEFORE the diagnostic re-run: train_losses runs in
        # train mode and would advance the BN running stats, polluting the
        # at-failure state the box is supposed to preserve
        model_state = {k: v.cpu().clone() for k, v in
                       self.model.state_dict().items()}
        optim_state = [opt.state_dict() for opt in self.optimizers]
        param_map = {n: self._buf_stats(p)
                     for n, p in self.model.named_parameters()}
        opt_map = {}
        for tag, opt in (("ae", self.opt_ae), ("pc", self.opt_pc)):
            for buf in ("flat_p", "flat_g", "exp_avg", "exp_avg_sq"):
                opt_map[f"{tag}.{buf}"] = self._buf_stats(
                    getattr(opt, buf, None))
        components = {}
        try:
            with self._autocast():
                out = self.model.train_losses(x, y)
            for k in ("loss", "bpp", "H_real", "pc_loss", "d_loss", "reg",
                      "loss_sinet"):
                v = out.get(k)
                if isinstance(v, torch.Tensor) and v.numel() == 1:
                    components[k] = float(v.detach())
        except Exception as e:  # the re-run itself may blow up — still dump
            components["rerun_error"] = repr(e)
        path = os.path.join(self.blackbox_dir, f"step_{step}.pt")
        torch.save({
            "global_step": step,
            "x": x.detach().cpu(), "y": None if y is None else y.detach().cpu(),
            "components": components,
            "param_stats": param_map,
            "optimizer_stats": opt_map,
            "model_state": model_state,
            "optim_state": optim_state,
        }, path)
        corrupt = {k: v for k, v in {**param_map, **opt_map}.items()
                   if v.get("nonfinite")}
        warnings.warn(f"nan_guard tripped at step {step}: components="
                      f"{components}; corrupt buffers: "
                      f"{list(corrupt)[:8] or 'none (transient activations)'}")
        return path

    def validate(self, x: torch.Tensor, y: Optional[torch.Tensor]) -> torch.Tensor:
        from ..ops import conv as _conv
        _conv.begin_step()  # weights may have stepped since panels were built
        with self._autocast():
            return self.model.validate_loss(x, y)
