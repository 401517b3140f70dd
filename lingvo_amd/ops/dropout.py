"""Fused deterministic dropout wrapper (seed-only state)."""

from __future__ import annotations

from typing import Optional

import torch

from lingvo_amd.ops import _loader


class _DropoutFn(torch.autograd.Function):

  @staticmethod
  def forward(ctx, x, residual, seed, keep):
    ext = _loader.get_ext(required=True)
    y = ext.dropout_fwd(x, residual, seed, keep)
    ctx.seed = seed
    ctx.keep = keep
    ctx.has_res = residual is not None
    return y

  @staticmethod
  def backward(ctx, dy):
    ext = _loader.get_ext(required=True)
    dy = dy.contiguous()
    dx = ext.dropout_bwd(dy, ctx.seed, ctx.keep)
    dres = dy if ctx.has_res else None
    return dx, dres, None, None


def dropout(x: torch.Tensor, keep_prob: float, seed: int,
            residual: Optional[torch.Tensor] = None) -> torch.Tensor:
  """y = dropout(x) (+ residual). GPU bf16 fast path; generic otherwise."""
  if x.is_cuda and x.numel() % 8 == 0:
    orig = x.dtype
    y = _DropoutFn.apply(
        x.to(torch.bfloat16).contiguous(),
        None if residual is None else
        residual.to(torch.bfloat16).contiguous(), seed, keep_prob)
    return y.to(orig) if orig != torch.bfloat16 else y
  g = torch.Generator(device=x.device)
  g.manual_seed(seed & 0x7FFFFFFFFFFFFFFF)
  mask = (torch.rand(x.shape, generator=g, device=x.device,
                     dtype=torch.float32) < keep_prob)
  y = x * mask.to(x.dtype) / keep_prob
  return y if residual is None else y + residual
