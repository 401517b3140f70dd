"""Fused GroupNorm autograd wrapper (padded-aware)."""

from __future__ import annotations

from typing import Optional

import torch

from lingvo_amd.ops import _loader


ACT_CODES = {'NONE': 0, 'SILU': 1, 'SWISH': 1}


class _GroupNormFn(torch.autograd.Function):

  @staticmethod
  def forward(ctx, x, gamma, beta, paddings, groups, eps, act):
    ext = _loader.get_ext(required=True)
    pad_b = None if paddings is None else \
        paddings.to(torch.bfloat16).contiguous()
    gamma_b = gamma.to(torch.bfloat16).contiguous()
    beta_b = beta.to(torch.bfloat16).contiguous()
    y, mean, rstd = ext.group_norm_fwd(
        x, gamma_b, beta_b, pad_b, groups, eps, act)
    ctx.save_for_backward(x, gamma_b, beta_b, mean, rstd,
                          pad_b if pad_b is not None else torch.empty(0))
    ctx.groups = groups
    ctx.act = act
    ctx.pdtype = gamma.dtype
    return y

  @staticmethod
  def backward(ctx, dy):
    ext = _loader.get_ext(required=True)
    x, gamma_b, beta_b, mean, rstd, pad_b = ctx.saved_tensors
    pad = pad_b if pad_b.numel() else None
    dx, dgamma, dbeta = ext.group_norm_bwd(
        dy.contiguous().to(torch.bfloat16), x, gamma_b, beta_b, pad,
        mean, rstd, ctx.groups, ctx.act)
    return (dx, dgamma.to(ctx.pdtype), dbeta.to(ctx.pdtype), None, None,
            None, None)


def group_norm(x: torch.Tensor, gamma: torch.Tensor, beta: torch.Tensor,
               paddings: Optional[torch.Tensor], groups: int,
               eps: float = 1e-3, act: str = 'NONE') -> torch.Tensor:
  """Fused padded GroupNorm; act='SILU' applies swish INSIDE the same
  kernel (fwd epilogue; bwd folds silu' without storing extra state)."""
  orig = x.dtype
  y = _GroupNormFn.apply(x.to(torch.bfloat16).contiguous(), gamma, beta,
                         paddings, groups, eps, ACT_CODES[act.upper()])
  return y.to(orig) if orig != torch.bfloat16 else y
