"""Depthwise time-convolution autograd wrapper (K7)."""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn.functional as F

from lingvo_amd.ops import _loader


class _DwConv1dFn(torch.autograd.Function):

  @staticmethod
  def forward(ctx, x, w, bias, pad):
    ext = _loader.get_ext(required=True)
    y = ext.dwconv1d_fwd(x, w, bias, pad)
    ctx.save_for_backward(x, w)
    ctx.pad = pad
    ctx.has_bias = bias is not None
    ctx.wdtype = w.dtype
    return y

  @staticmethod
  def backward(ctx, dy):
    ext = _loader.get_ext(required=True)
    x, w = ctx.saved_tensors
    dx, dw, db = ext.dwconv1d_bwd(dy.contiguous(), x, w, ctx.pad)
    return (dx, dw.to(ctx.wdtype),
            db.to(ctx.wdtype) if ctx.has_bias else None, None)


def depthwise_conv1d(x: torch.Tensor, w: torch.Tensor,
                     bias: Optional[torch.Tensor] = None,
                     causal: bool = False) -> torch.Tensor:
  """x [B,T,D], w [K,D] -> [B,T,D]; zero padding outside the sequence."""
  k = w.shape[0]
  pad = k - 1 if causal else (k - 1) // 2
  if x.is_cuda:
    orig = x.dtype
    out = _DwConv1dFn.apply(
        x.to(torch.bfloat16).contiguous(), w.to(torch.bfloat16).contiguous(),
        None if bias is None else bias.to(torch.bfloat16).contiguous(), pad)
    return out.to(orig) if orig != torch.bfloat16 else out
  # CPU reference via grouped conv1d.
  xf = x.float().permute(0, 2, 1)  # [B,D,T]
  wf = w.float().t().unsqueeze(1)  # [D,1,K]
  left = pad
  right = (k - 1) - pad
  xf = F.pad(xf, (left, right))
  out = F.conv1d(xf, wf, bias.float() if bias is not None else None,
                 groups=x.shape[2])
  return out.permute(0, 2, 1).to(x.dtype)
