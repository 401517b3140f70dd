"""Activation registry (reference lingvo/core/activations.py)."""

from __future__ import annotations

import math
from typing import Callable, Dict

import torch
import torch.nn.functional as F

_ACTIVATIONS: Dict[str, Callable] = {
    'RELU': F.relu,
    'RELU6': lambda x: torch.clamp(x, 0.0, 6.0),
    'SIGMOID': torch.sigmoid,
    'TANH': torch.tanh,
    'GELU': lambda x: F.gelu(x, approximate='none'),
    'GELU_APPROXIMATE': lambda x: F.gelu(x, approximate='tanh'),
    'SWISH': F.silu,
    'SILU': F.silu,
    'SOFTPLUS': F.softplus,
    'NONE': lambda x: x,
}


def GetFn(name: str) -> Callable:
  key = name.upper()
  if key not in _ACTIVATIONS:
    raise KeyError(f'Unknown activation {name!r}; known: '
                   f'{sorted(_ACTIVATIONS)}')
  return _ACTIVATIONS[key]


def Register(name: str, fn: Callable) -> None:
  _ACTIVATIONS[name.upper()] = fn


def Entmax15(x: torch.Tensor, dim: int = -1) -> torch.Tensor:
  """Exact 1.5-entmax (reference lingvo/core/entmax.py; Peters et al.
  2019): sparse softmax family member alpha=1.5, computed by the exact
  sort-based threshold algorithm. Differentiable: the backward of the
  closed-form solution is implemented via the custom Function below."""
  return _Entmax15Fn.apply(x, dim)


class _Entmax15Fn(torch.autograd.Function):

  @staticmethod
  def forward(ctx, x, dim):
    x = x / 2  # alpha-trick: entmax15(x) = argmax <p,x> - sum p^1.5 ...
    x = x - x.max(dim=dim, keepdim=True).values
    srt = torch.sort(x, dim=dim, descending=True).values
    k = torch.arange(1, x.shape[dim] + 1, device=x.device,
                     dtype=x.dtype)
    shape = [1] * x.dim()
    shape[dim] = -1
    k = k.reshape(shape)
    mean = srt.cumsum(dim) / k
    mean_sq = (srt ** 2).cumsum(dim) / k
    ss = k * (mean_sq - mean ** 2)
    delta = (1 - ss) / k
    delta = delta.clamp_min(0)
    tau = mean - delta.sqrt()
    support = (tau <= srt).sum(dim=dim, keepdim=True)
    tau_star = tau.gather(dim, support - 1)
    out = torch.clamp(x - tau_star, min=0) ** 2
    ctx.save_for_backward(out)
    ctx.dim = dim
    return out

  @staticmethod
  def backward(ctx, grad):
    out, = ctx.saved_tensors
    dim = ctx.dim
    sqrt_out = out.sqrt()
    g = grad * sqrt_out
    q = g.sum(dim, keepdim=True) / sqrt_out.sum(dim, keepdim=True
                                                ).clamp_min(1e-30)
    return g - q * sqrt_out, None


Register('ENTMAX15', Entmax15)
