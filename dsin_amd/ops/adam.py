"""Fused flat-buffer Adam optimizer.

All parameters of a group are repointed to views of one contiguous fp32
buffer, gradients to views of a flat gradient buffer (autograd accumulates
in place into existing .grad tensors), and the whole update is ONE kernel
launch over the flat buffers per group — versus hundreds of per-tensor
kernels in a stock optimizer. lr and the step count are device tensors so
the step replays correctly inside a captured hipGraph while the staircase
schedule updates lr externally.

Semantics follow tf.train.AdamOptimizer (the reference's optimizer):
eps is added OUTSIDE the bias-corrected sqrt(v), matching
lr_t = lr sqrt(1-b2^t)/(1-b1^t); p -= lr_t m/(sqrt(v)+eps).

The flat gradient buffer doubles as the DDP communication buffer: the
Trainer's reducer can all-reduce it directly.
"""

from __future__ import annotations

from typing import Iterable, List

import torch

from . import _require_ext


class FusedAdam:
    def __init__(self, params: Iterable[torch.nn.Parameter], lr: float,
                 betas=(0.9, 0.999), eps: float = 1e-8):
        self.params: List[torch.nn.Parameter] = [p for p in params]
        assert self.params, "empty parameter group"
        device = self.params[0].device
        assert all(p.dtype == torch.float32 for p in self.params)
        total = sum(p.numel() for p in self.params)

        self.flat_p = torch.empty(total, device=device)
        self.flat_g = torch.zeros(total, device=device)
        self.exp_avg = torch.zeros(total, device=device)
        self.exp_avg_sq = torch.zeros(total, device=device)
        ofs = 0
        self._slices = []
        for p in self.params:
            n = p.numel()
            self.flat_p[ofs:ofs + n].copy_(p.data.reshape(-1))
            p.data = self.flat_p[ofs:ofs + n].view(p.shape)
            # grads are NOT repointed into flat_g: with p.grad left None,
            # autograd STEALS each produced gradient (zero accumulate-add
            # kernels, ~330 of them per DSIN step) and gather_grads() moves
            # them into the flat buffer with one _foreach_copy_ sweep.
            self._slices.append((ofs, n))
            ofs += n

        self.lr_t = torch.tensor(float(lr), device=device)
        self.step_t = torch.zeros(1, device=device, dtype=torch.int32)
        self.wd = None  # optional per-element L2 factors (see set_weight_decay)
        self.betas = tuple(betas)
        self.eps = float(eps)
        # torch-optimizer-compatible surface for the LR schedule
        self.param_groups = [{"lr": self.lr_t, "params": self.params}]
        self.defaults = {"lr": float(lr)}

    def zero_grad(self, set_to_none: bool = True):
        # gather_grads() overwrites flat_g, so only the python-side grad
        # handles need dropping (autograd then steals fresh tensors)
        for p in self.params:
            p.grad = None

    @torch.no_grad()
    def gather_grads(self):
        """Copy this step's stolen .grad tensors into the flat comm/step
        buffer (one horizontally-fused kernel); missing grads zero their
        slice. Must run after backward, before any all-reduce / step()."""
        dsts, srcs = [], []
        for p, (ofs, n) in zip(self.params, self._slices):
            if p.grad is None:
                self.flat_g[ofs:ofs + n].zero_()
            else:
                dsts.append(self.flat_g[ofs:ofs + n])
                srcs.append(p.grad.reshape(-1))
        if dsts:
            torch._foreach_copy_(dsts, srcs)

    def set_weight_decay(self, param: torch.nn.Parameter, factor: float) -> bool:
        """Register a classic L2 term factor/2*sum(p^2) for `param`: its
        lambda*p gradient is added inside the fused step (identical Adam
        update to building the regularizer in the autograd graph, without
        the ~100 cat/split/add kernels per step that graph costs)."""
        for p, (ofs, n) in zip(self.params, self._slices):
            if p is param:
                if self.wd is None:
                    self.wd = torch.zeros_like(self.flat_g)
                self.wd[ofs:ofs + n].fill_(float(factor))
                return True
        return False

    @torch.no_grad()
    def reg_value(self) -> torch.Tensor:
        """Current value of all registered L2 terms: 0.5*sum(wd * p^2)."""
        if self.wd is None:
            return torch.zeros((), device=self.flat_p.device)
        return 0.5 * (self.wd * self.flat_p.square()).sum()

    @torch.no_grad()
    def step(self):
        self.step_t += 1
        if self.flat_p.is_cuda:
            step_fn = _require_ext("adam_step")
            step_fn(self.flat_p, self.flat_g, self.exp_avg, self.exp_avg_sq,
                    self.lr_t, self.step_t, self.wd, self.betas[0],
                    self.betas[1], self.eps)
            return
        # CPU fallback with identical flat-buffer math (TF-Adam semantics:
        # eps outside the bias-corrected sqrt). Lets gloo CI run the SAME
        # fused-optimizer + FlatGradReducer path the GPU uses.
        b1, b2 = self.betas
        t = float(self.step_t.item())
        g = self.flat_g
        if self.wd is not None:
            g = g + self.wd * self.flat_p
        self.exp_avg.mul_(b1).add_(g, alpha=1.0 - b1)
        self.exp_avg_sq.mul_(b2).addcmul_(g, g, value=1.0 - b2)
        lr_t = float(self.lr_t.item()) * ((1.0 - b2 ** t) ** 0.5) / (1.0 - b1 ** t)
        denom = self.exp_avg_sq.sqrt().add_(self.eps)
        self.flat_p.addcdiv_(self.exp_avg, denom, value=-lr_t)

    def state_dict(self):
        return {
            "flat_p": self.flat_p,
            "exp_avg": self.exp_avg,
            "exp_avg_sq": self.exp_avg_sq,
            "step": self.step_t,
            "lr": self.lr_t,
            "betas": self.betas,
            "eps": self.eps,
        }

    def load_state_dict(self, sd):
        self.flat_p.copy_(sd["flat_p"].to(self.flat_p.device))
        self.exp_avg.copy_(sd["exp_avg"].to(self.exp_avg.device))
        self.exp_avg_sq.copy_(sd["exp_avg_sq"].to(self.exp_avg_sq.device))
        self.step_t.copy_(sd["step"].to(self.step_t.device))
        self.lr_t.copy_(sd["lr"].to(self.lr_t.device))
