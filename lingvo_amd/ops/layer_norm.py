"""LayerNorm/RMSNorm autograd wrappers over the gfx950 HIP kernels (K5).

CPU reference path (fp32 torch) is used by numerics tests; CUDA tensors
require the HIP extension.
"""

from __future__ import annotations

import torch

from lingvo_amd.ops import _loader


class _LayerNormFn(torch.autograd.Function):

  @staticmethod
  def forward(ctx, x, scale, bias, eps, rms):
    ext = _loader.get_ext(required=True)
    x = x.contiguous()
    scale_b = scale.to(torch.bfloat16).contiguous()
    bias_b = (bias.to(torch.bfloat16).contiguous()
              if bias is not None else scale_b)
    y, mean, rstd = ext.layer_norm_fwd(x, scale_b, bias_b, eps, rms)
    ctx.save_for_backward(x, scale_b, mean, rstd)
    ctx.rms = rms
    ctx.scale_dtype = scale.dtype
    return y

  @staticmethod
  def backward(ctx, dy):
    ext = _loader.get_ext(required=True)
    x, scale_b, mean, rstd = ctx.saved_tensors
    dx, dscale, dbias = ext.layer_norm_bwd(
        dy.contiguous().to(torch.bfloat16), x, scale_b, mean, rstd, ctx.rms)
    dscale = dscale.to(ctx.scale_dtype)
    dbias = None if ctx.rms else dbias.to(ctx.scale_dtype)
    return dx, dscale, dbias, None, None


def layer_norm(x: torch.Tensor, scale: torch.Tensor, bias: torch.Tensor,
               eps: float = 1e-6) -> torch.Tensor:
  """y = (x - mean)/sqrt(var+eps) * (1+scale) + bias over the last dim."""
  if x.is_cuda:
    orig = x.dtype
    y = _LayerNormFn.apply(x.to(torch.bfloat16), scale, bias, eps, False)
    return y.to(orig) if orig != torch.bfloat16 else y
  xf = x.float()
  mean = xf.mean(-1, keepdim=True)
  var = xf.var(-1, unbiased=False, keepdim=True)
  out = (xf - mean) * torch.rsqrt(var + eps)
  out = out * (1.0 + scale.float()) + bias.float()
  return out.to(x.dtype)


def rms_norm(x: torch.Tensor, scale: torch.Tensor,
             eps: float = 1e-6) -> torch.Tensor:
  """y = x * rsqrt(mean(x^2)+eps) * (1+scale) over the last dim."""
  if x.is_cuda:
    orig = x.dtype
    y = _LayerNormFn.apply(x.to(torch.bfloat16), scale, None, eps, True)
    return y.to(orig) if orig != torch.bfloat16 else y
  xf = x.float()
  ms = xf.pow(2).mean(-1, keepdim=True)
  out = xf * torch.rsqrt(ms + eps) * (1.0 + scale.float())
  return out.to(x.dtype)
