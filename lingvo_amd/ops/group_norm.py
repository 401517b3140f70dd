"""Fused GroupNorm autograd wrapper (padded-aware)."""

from __future__ import annotations

from typing import Optional

import torch

from lingvo_amd.ops import _loader


class _GroupNormFn(torch.autograd.Function):

  @staticmethod
  def forward(ctx, x, gamma, beta, paddings, groups, eps):
    ext = _loader.get_ext(required=True)
    pad_b = None if paddings is None else \
        paddings.to(torch.bfloat16).contiguous()
    gamma_b = gamma.to(torch.bfloat16).contiguous()
    y, mean, rstd = ext.group_norm_fwd(
        x, gamma_b, beta.to(torch.bfloat16).contiguous(), pad_b, groups,
        eps)
    ctx.save_for_backward(x, gamma_b, mean, rstd,
                          pad_b if pad_b is not None else torch.empty(0))
    ctx.groups = groups
    ctx.pdtype = gamma.dtype
    return y

  @staticmethod
  def backward(ctx, dy):
    ext = _loader.get_ext(required=True)
    x, gamma_b, mean, rstd, pad_b = ctx.saved_tensors
    pad = pad_b if pad_b.numel() else None
    dx, dgamma, dbeta = ext.group_norm_bwd(
        dy.contiguous().to(torch.bfloat16), x, gamma_b, pad, mean, rstd,
        ctx.groups)
    return (dx, dgamma.to(ctx.pdtype), dbeta.to(ctx.pdtype), None, None,
            None)


def group_norm(x: torch.Tensor, gamma: torch.Tensor, beta: torch.Tensor,
               paddings: Optional[torch.Tensor], groups: int,
               eps: float = 1e-3) -> torch.Tensor:
  orig = x.dtype
  y = _GroupNormFn.apply(x.to(torch.bfloat16).contiguous(), gamma, beta,
                         paddings, groups, eps)
  return y.to(orig) if orig != torch.bfloat16 else y
