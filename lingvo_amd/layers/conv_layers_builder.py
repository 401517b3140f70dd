"""Builder patterns for padding-aware conv stacks
(reference lingvo/core/conv_layers_builder.py:126 Builder,
:49 CausalPoolingLayer). Each method returns a Params tree; a single
Instantiate() materializes the stack (the builder DSL idiom)."""

from __future__ import annotations

from typing import Optional, Sequence

import torch

from lingvo_amd.core import py_utils
from lingvo_amd.core.base_layer import BaseLayer
from lingvo_amd.core.nested_map import NestedMap
from lingvo_amd.layers import activations
from lingvo_amd.layers import bn_layers
from lingvo_amd.layers import builder_layers
from lingvo_amd.layers import conv_layers_with_time_padding as ctp


class CausalPoolingLayer(BaseLayer):
  """Causal pooling over time on [B, T, F, C] with paddings
  (reference conv_layers_builder.py:49): output t pools over the
  left_context frames ending at t (-1 = cumulative)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('pooling_type', 'AVG', 'AVG or MAX.')
    p.Define('left_context', -1, 'Window (frames); -1 = everything.')
    return p

  def FProp(self, theta: NestedMap, inputs: torch.Tensor,
            paddings: torch.Tensor):
    p = self.p
    b, t = inputs.shape[:2]
    mask = (1.0 - paddings).reshape(
        b, t, *([1] * (inputs.dim() - 2))).to(inputs.dtype)
    x = inputs * mask
    if p.pooling_type == 'AVG':
      csum = torch.cumsum(x, dim=1)
      cnt = torch.cumsum(mask, dim=1)
      if p.left_context > 0:
        w = p.left_context
        shifted = torch.cat(
            [torch.zeros_like(csum[:, :w]), csum[:, :-w]], dim=1)
        cshift = torch.cat(
            [torch.zeros_like(cnt[:, :w]), cnt[:, :-w]], dim=1)
        csum = csum - shifted
        cnt = cnt - cshift
      out = csum / cnt.clamp_min(1.0)
    else:
      neg = inputs.masked_fill(mask == 0, float('-inf'))
      if p.left_context > 0:
        w = p.left_context
        pads = torch.full_like(neg[:, :w - 1], float('-inf')) \
            if w > 1 else neg[:, :0]
        win = torch.cat([pads, neg], dim=1).unfold(1, w, 1)
        out = win.max(dim=-1).values
      else:
        out = torch.cummax(neg, dim=1).values
      out = torch.where(torch.isinf(out), torch.zeros_like(out), out)
    return py_utils.ApplyPadding(paddings, out), paddings


class _PaddedFnLayer(BaseLayer):
  """Wraps fn(inputs) -> outputs, carrying paddings through."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('fn', None, 'Callable on the activation tensor.')
    return p

  def FProp(self, theta, inputs, paddings):
    return self.p.fn(inputs), paddings


class _PaddedSeqLayer(BaseLayer):
  """Sequential over (inputs, paddings) pairs."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('sub', [], 'Sub-layer params.')
    return p

  def __init__(self, params):
    super().__init__(params)
    self.CreateChildren(
        'seq', [sp.Copy().Set(name=sp.name or f'sub_{i}')
                for i, sp in enumerate(self.p.sub)])

  def FProp(self, theta, inputs, paddings):
    for i, layer in enumerate(self.seq):
      inputs, paddings = layer.FProp(theta.seq[i], inputs, paddings)
    return inputs, paddings


class Builder:
  """Conv-stack builder (reference conv_layers_builder.py:126)."""

  def __init__(self, norm: str = 'batch', activation: str = 'RELU'):
    self._norm = norm
    self._activation = activation

  def _Seq(self, name, *subs):
    return _PaddedSeqLayer.Params().Set(name=name, sub=list(subs))

  def _Activation(self, name, activation=None):
    fn = activations.GetFn(activation or self._activation)
    return _PaddedFnLayer.Params().Set(name=name, fn=fn)

  def _Norm(self, name, dims):
    return _NormWrapper.Params().Set(name=name, dim=dims,
                                     norm=self._norm)

  def Conv2D(self, name, filter_shape, filter_stride=(1, 1),
             is_causal=False, activation=None):
    """conv -> norm -> activation, paddings threaded throughout."""
    conv = ctp.Conv2DLayerWithPadding.Params().Set(
        name=f'{name}_conv', filter_shape=tuple(filter_shape),
        filter_stride=tuple(filter_stride), is_causal=is_causal)
    return self._Seq(name, conv, self._Norm(f'{name}_n', filter_shape[-1]),
                     self._Activation(f'{name}_act', activation))

  def DepthwiseConv2D(self, name, filter_shape, filter_stride=(1, 1),
                      is_causal=False, activation=None):
    conv = ctp.DepthwiseConv2DLayer.Params().Set(
        name=f'{name}_conv', filter_shape=tuple(filter_shape),
        filter_stride=tuple(filter_stride), is_causal=is_causal)
    out_ch = filter_shape[2] * filter_shape[3]
    return self._Seq(name, conv, self._Norm(f'{name}_n', out_ch),
                     self._Activation(f'{name}_act', activation))

  def SeparableConv2D(self, name, filter_shape, filter_stride=(1, 1),
                      depth_multiplier=1, is_causal=False,
                      activation=None):
    """Depthwise (th, fw, cin, mult) then pointwise (1,1,cin*mult,cout)."""
    th, fw, cin, cout = filter_shape
    dw = ctp.DepthwiseConv2DLayer.Params().Set(
        name=f'{name}_dw', filter_shape=(th, fw, cin, depth_multiplier),
        filter_stride=tuple(filter_stride), is_causal=is_causal)
    pw = ctp.Conv2DLayerWithPadding.Params().Set(
        name=f'{name}_pw',
        filter_shape=(1, 1, cin * depth_multiplier, cout))
    return self._Seq(name, dw, pw, self._Norm(f'{name}_n', cout),
                     self._Activation(f'{name}_act', activation))

  def GlobalPooling(self, name, pooling_type='AVG'):
    return _GlobalPoolWrapper.Params().Set(name=name,
                                           pooling_type=pooling_type)


class _NormWrapper(BaseLayer):
  """Norm over the channel dim of [B, T, F, C] with time paddings."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('dim', 0, 'Channels.')
    p.Define('norm', 'batch', "'batch' | 'none'.")
    return p

  def __init__(self, params):
    super().__init__(params)
    if self.p.norm == 'batch':
      self.CreateChild('bn', bn_layers.BatchNormLayer.Params().Set(
          dim=self.p.dim))

  def FProp(self, theta, inputs, paddings):
    if self.p.norm == 'none':
      return inputs, paddings
    b, t, f, c = inputs.shape
    flat = inputs.reshape(b, t * f, c)
    pad = paddings.repeat_interleave(f, dim=1)
    out = self.bn.FProp(theta.bn, flat, pad)
    return out.reshape(b, t, f, c), paddings


class _GlobalPoolWrapper(BaseLayer):

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('pooling_type', 'AVG', 'AVG or MAX.')
    return p

  def __init__(self, params):
    super().__init__(params)
    self.CreateChild('pool', ctp.GlobalPoolingLayer.Params().Set(
        pooling_type=self.p.pooling_type))

  def FProp(self, theta, inputs, paddings):
    b, t, f, c = inputs.shape
    out = self.pool.FProp(theta.pool, inputs.reshape(b, t, f * c),
                          paddings)
    return out.reshape(b, f, c), None
