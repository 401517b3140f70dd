"""Funnel-transformer sequence pooling / upsampling.

MI355X-native re-implementation of the reference's funnel layers
(lingvo/core/batch_major_attention.py:8162 FunnelPoolingLayer, :8423
FunnelUpsampleLayer; Dai et al. 2020): pool the sequence between
transformer blocks to shrink the O(T^2) attention cost, then upsample
back for token-level outputs.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch
import torch.nn.functional as F

from lingvo_amd.core import py_utils
from lingvo_amd.core.base_layer import BaseLayer
from lingvo_amd.core.nested_map import NestedMap


class FunnelPoolingLayer(BaseLayer):
  """Pools [B, T, D] -> [B, ceil(T/stride), D] with padding-aware
  avg/max pooling; paddings pool by min (a window with ANY real frame
  is real)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('stride', 2, 'Pooling stride == window.')
    p.Define('pooling_type', 'AVG', 'AVG or MAX.')
    return p

  def FProp(self, theta: NestedMap, x: torch.Tensor,
            paddings: Optional[torch.Tensor] = None
            ) -> Tuple[torch.Tensor, Optional[torch.Tensor]]:
    p = self.p
    s = p.stride
    if s == 1:
      return x, paddings
    b, t, d = x.shape
    pad_t = (-t) % s
    xp = F.pad(x, (0, 0, 0, pad_t))
    mask = None
    if paddings is not None:
      mask = 1.0 - F.pad(paddings, (0, pad_t), value=1.0)  # 1 = real
      xp = xp * mask[:, :, None].to(xp.dtype)
    xw = xp.reshape(b, -1, s, d)
    if p.pooling_type == 'MAX':
      if mask is not None:
        xw = xw.masked_fill(
            (mask.reshape(b, -1, s) < 0.5)[:, :, :, None], -1e30)
      out = xw.max(dim=2).values
      if mask is not None:
        dead = mask.reshape(b, -1, s).max(dim=2).values < 0.5
        out = out.masked_fill(dead[:, :, None], 0.0)
    else:
      denom = (mask.reshape(b, -1, s).sum(-1, keepdim=True)
               if mask is not None else
               torch.full((b, xw.shape[1], 1), float(s), device=x.device))
      out = xw.sum(dim=2) / denom.clamp_min(1.0).to(xw.dtype)
    new_pad = None
    if paddings is not None:
      new_pad = 1.0 - (mask.reshape(b, -1, s).max(dim=2).values)
      out = py_utils.ApplyPadding(new_pad, out)
    return out, new_pad


class FunnelUpsampleLayer(BaseLayer):
  """Upsamples [B, T', D] back to [B, T, D] by nearest-repeat plus an
  optional learned projection (reference FunnelUpsampleLayer)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('stride', 2, 'Upsample factor.')
    p.Define('input_dim', 0, 'D (for the projection).')
    p.Define('use_projection', True, 'Learned per-position projection.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    if p.use_projection:
      self.CreateVariable('proj_w', py_utils.WeightParams(
          [p.input_dim, p.stride * p.input_dim], p.params_init, p.dtype))
      self.CreateVariable('proj_b', py_utils.WeightParams(
          [p.stride * p.input_dim],
          py_utils.WeightInit.Constant(0.0), p.dtype))

  def FProp(self, theta: NestedMap, x: torch.Tensor,
            target_len: Optional[int] = None) -> torch.Tensor:
    p = self.p
    b, t, d = x.shape
    if p.use_projection:
      y = py_utils.MatmulBias(x, theta.proj_w, theta.proj_b)
      y = y.reshape(b, t * p.stride, d)
    else:
      y = x.repeat_interleave(p.stride, dim=1)
    if target_len is not None:
      y = y[:, :target_len]
    return y
