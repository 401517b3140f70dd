"""Reversible (RevNet) residual blocks with O(1) activation memory.

MI355X-native re-implementation of the reference's
lingvo/core/reversible_layers.py (RevNetLayer/StackedRevNetLayer;
Gomez et al. 2017). Forward splits the features into (x1, x2) and
computes
    y1 = x1 + F(x2),   y2 = x2 + G(y1).
Backward RECONSTRUCTS (x1, x2) from (y1, y2) instead of stashing them,
so a stacked revnet stores only the final activations — the HBM saved
goes straight into bigger per-GPU batches.
"""

from __future__ import annotations

from typing import Tuple

import torch

from lingvo_amd.core.base_layer import BaseLayer
from lingvo_amd.core.nested_map import NestedMap


class _RevBlockFn(torch.autograd.Function):
  """One reversible block; inputs are freed, backward reconstructs.

  The layer parameters are passed as tensor varargs so the autograd
  engine tracks the dependency (outputs require grad even when the
  block inputs do not); their gradients are accumulated directly into
  `.grad` by the recompute backward passes, and None is returned for
  them here (the memcnn/RevTorch convention).
  """

  @staticmethod
  def forward(ctx, x1, x2, f_fn, g_fn, n_f, *params):
    ctx.f_fn, ctx.g_fn = f_fn, g_fn
    with torch.no_grad():
      y1 = x1 + f_fn(x2)
      y2 = x2 + g_fn(y1)
    ctx.save_for_backward(y1.detach(), y2.detach())
    ctx.n_params = len(params)
    return y1, y2

  @staticmethod
  def backward(ctx, dy1, dy2):
    y1, y2 = ctx.saved_tensors
    f_fn, g_fn = ctx.f_fn, ctx.g_fn
    # reconstruct inputs, recompute both subgraphs WITH grad
    with torch.enable_grad():
      y1_r = y1.detach().requires_grad_(True)
      g_out = g_fn(y1_r)
      torch.autograd.backward(g_out, dy2)
      x2 = (y2 - g_out).detach()
      x2_r = x2.requires_grad_(True)
      f_out = f_fn(x2_r)
      dy1_total = dy1 + y1_r.grad
      torch.autograd.backward(f_out, dy1_total)
    dx2 = dy2 + x2_r.grad
    return (dy1_total, dx2, None, None, None) + (None,) * ctx.n_params


class RevNetLayer(BaseLayer):
  """One reversible block around two residual sub-layers F and G.

  f_tpl / g_tpl are any BaseLayer params whose FProp(theta, x) maps
  [..., D/2] -> [..., D/2].
  """

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('f_tpl', None, 'Residual function F params.')
    p.Define('g_tpl', None, 'Residual function G params.')
    return p

  def __init__(self, params):
    super().__init__(params)
    self.CreateChild('f', self.p.f_tpl)
    self.CreateChild('g', self.p.g_tpl)

  def FProp(self, theta: NestedMap, x1: torch.Tensor, x2: torch.Tensor
            ) -> Tuple[torch.Tensor, torch.Tensor]:
    f_fn = lambda t: self.f.FProp(theta.f, t)
    g_fn = lambda t: self.g.FProp(theta.g, t)
    params = [v for v in self.parameters() if v.requires_grad]
    return _RevBlockFn.apply(x1, x2, f_fn, g_fn, 0, *params)


class StackedRevNetLayer(BaseLayer):
  """Stack of reversible blocks; only the final (y1, y2) persist.

  FProp takes [..., D] (split in half) and returns [..., D].
  """

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('block_tpls', [], 'List of RevNetLayer params.')
    return p

  def __init__(self, params):
    super().__init__(params)
    self.CreateChildren('blocks', list(self.p.block_tpls))

  def FProp(self, theta: NestedMap, x: torch.Tensor) -> torch.Tensor:
    x1, x2 = x.chunk(2, dim=-1)
    for i, block in enumerate(self.blocks):
      x1, x2 = block.FProp(theta.blocks[i], x1, x2)
    return torch.cat([x1, x2], dim=-1)
