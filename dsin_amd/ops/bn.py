"""Fused BatchNorm(+activation) autograd op.

torch.nn.BatchNorm2d semantics (biased batch variance for normalization,
unbiased for the running estimate, running = (1-m)*running + m*batch) with
the following activation fused into the same elementwise pass. bf16 data,
fp32 statistics and affine parameters.
"""

from __future__ import annotations

from typing import Optional

import torch

from . import _require_ext


class _BNActFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, y, gamma, beta, rmean, rvar, momentum, eps, training,
                act, residual=None):
        # fused residual (out = act(bn(y)+res)) is only used with act=0 in
        # the model (adds follow LINEAR BN, ref autoencoder_imgcomp.py:284)
        fwd = _require_ext("bn_fwd")
        assert residual is None or act == 0
        out, mean, rstd = fwd(y, gamma, beta, rmean, rvar, momentum, eps,
                              training, act, residual)
        ctx.save_for_backward(y, out, mean, rstd, gamma)
        ctx.meta = (training, act, residual is not None)
        return out

    @staticmethod
    def backward(ctx, dy):
        y, out, mean, rstd, gamma = ctx.saved_tensors
        training, act, has_res = ctx.meta
        bwd = _require_ext("bn_bwd")
        dy = dy.contiguous().to(torch.bfloat16)
        dx, dgamma, dbeta = bwd(dy, y, out, mean, rstd, gamma, training, act)
        dres = dy if has_res else None
        return (dx, dgamma, dbeta, None, None, None, None, None, None, dres)


def batch_norm_act(y: torch.Tensor, bn: torch.nn.BatchNorm2d, training: bool,
                   act: int = 0,
                   residual: Optional[torch.Tensor] = None) -> torch.Tensor:
    """Apply `bn` (+ activation 0/1/2 = none/relu/lrelu.2) via the fused
    kernels on GPU; torch fallback elsewhere."""
    if not y.is_cuda:
        out = torch.nn.functional.batch_norm(
            y, bn.running_mean, bn.running_var, bn.weight, bn.bias, training,
            bn.momentum, bn.eps)
        if residual is not None:
            out = out + residual
        if act == 1:
            out = torch.relu(out)
        elif act == 2:
            out = torch.nn.functional.leaky_relu(out, 0.2)
        return out
    if training and bn.num_batches_tracked is not None:
        bn.num_batches_tracked += 1
    res = (residual.contiguous().to(torch.bfloat16)
           if residual is not None else None)
    return _BNActFn.apply(y.contiguous().to(torch.bfloat16), bn.weight,
                          bn.bias, bn.running_mean, bn.running_var,
                          bn.momentum, bn.eps, training, act, res)
