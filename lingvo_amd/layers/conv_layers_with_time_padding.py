"""Convolution layers that carry an explicit time-paddings tensor
(reference lingvo/core/conv_layers_with_time_padding.py:
Conv2DLayerWithPadding:425, causal variant :506, DepthwiseConv2DLayer
:608, CausalDepthwiseConv2DLayer :717, GlobalPoolingLayer :1012).

Depthwise time convs run on the gfx950 HIP kernel (K7); full 2-D convs
use the MIOpen-free im2col+GEMM path.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch
import torch.nn.functional as F

from lingvo_amd.core import py_utils
from lingvo_amd.core.base_layer import BaseLayer
from lingvo_amd.core.nested_map import NestedMap
from lingvo_amd.ops import conv1d as conv1d_ops


class Conv2DLayerWithPadding(BaseLayer):
  """[B, T, F, C] conv that masks padded frames before and after."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('filter_shape', (3, 3, 1, 32), '(th, fw, cin, cout).')
    p.Define('filter_stride', (1, 1), '(time, freq) stride.')
    p.Define('is_causal', False, 'Causal in time.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    th, fw, cin, cout = p.filter_shape
    self.CreateVariable('w', py_utils.WeightParams(
        [th, fw, cin, cout], p.params_init, p.dtype))

  def FProp(self, theta: NestedMap, inputs: torch.Tensor,
            paddings: torch.Tensor):
    p = self.p
    th, fw, cin, cout = p.filter_shape
    st, sf = p.filter_stride
    x = py_utils.ApplyPadding(paddings, inputs)
    x = x.permute(0, 3, 1, 2)  # [B, C, T, F]
    if p.is_causal:
      pad_t = (th - 1, 0)
    else:
      pad_t = ((th - 1) // 2, th - 1 - (th - 1) // 2)
    pad_f = ((fw - 1) // 2, fw - 1 - (fw - 1) // 2)
    x = F.pad(x, (pad_f[0], pad_f[1], pad_t[0], pad_t[1]))
    w = theta.w.permute(3, 2, 0, 1).contiguous()
    bsz = x.shape[0]
    tout = (x.shape[2] - th) // st + 1
    fout = (x.shape[3] - fw) // sf + 1
    cols = F.unfold(x, kernel_size=(th, fw), stride=(st, sf))
    out = torch.bmm(w.reshape(1, cout, -1).expand(bsz, -1, -1), cols)
    out = out.reshape(bsz, cout, tout, fout).permute(0, 2, 3, 1)
    out_paddings = paddings[:, ::st][:, :tout]
    return py_utils.ApplyPadding(out_paddings, out), out_paddings


class CausalConv2DLayerWithPadding(Conv2DLayerWithPadding):

  @classmethod
  def Params(cls):
    p = super().Params()
    p.is_causal = True
    return p


class DepthwiseConv1DLayer(BaseLayer):
  """Depthwise conv over time on [B, T, D] (HIP kernel K7)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('kernel_size', 3, 'Taps.')
    p.Define('dim', 0, 'Channels.')
    p.Define('is_causal', False, 'Causal.')
    p.Define('has_bias', True, 'Bias.')
    p.Define('dilation', 1, 'Time dilation (dilated taps run on the '
             'torch grouped-conv path; the HIP kernel covers 1).')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    self.CreateVariable('w', py_utils.WeightParams(
        [p.kernel_size, p.dim], p.params_init, p.dtype))
    if p.has_bias:
      self.CreateVariable('b', py_utils.WeightParams(
          [p.dim], py_utils.WeightInit.Constant(0.0), p.dtype))

  def FProp(self, theta: NestedMap, inputs: torch.Tensor,
            paddings: Optional[torch.Tensor] = None):
    p = self.p
    x = inputs
    if paddings is not None:
      x = py_utils.ApplyPadding(paddings, x)
    if p.dilation > 1:
      k, d = p.kernel_size, p.dim
      eff = (k - 1) * p.dilation + 1
      pad = (eff - 1, 0) if p.is_causal else           ((eff - 1) // 2, eff - 1 - (eff - 1) // 2)
      xc = F.pad(x.permute(0, 2, 1), pad)
      w = theta.w.t().reshape(d, 1, k)
      out = F.conv1d(xc, w, theta.b if p.has_bias else None,
                     dilation=p.dilation, groups=d).permute(0, 2, 1)
    else:
      out = conv1d_ops.depthwise_conv1d(
          x, theta.w, theta.b if p.has_bias else None,
          causal=p.is_causal)
    if paddings is not None:
      out = py_utils.ApplyPadding(paddings, out)
    return out, paddings


class CausalDepthwiseConv1DLayer(DepthwiseConv1DLayer):

  @classmethod
  def Params(cls):
    p = super().Params()
    p.is_causal = True
    return p


class NormalizedDepthwiseConv1DLayer(DepthwiseConv1DLayer):
  """Softmax-normalized taps (reference :903; lightweight-conv style)."""

  def FProp(self, theta, inputs, paddings=None):
    theta = theta.DeepCopy()
    theta.w = torch.softmax(theta.w.float(), dim=0).to(inputs.dtype)
    return super().FProp(theta, inputs, paddings)


class GlobalPoolingLayer(BaseLayer):
  """Padding-aware global max/avg pooling over time (reference :1012)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('pooling_type', 'AVG', 'AVG or MAX.')
    return p

  def FProp(self, theta: NestedMap, inputs: torch.Tensor,
            paddings: Optional[torch.Tensor] = None) -> torch.Tensor:
    if paddings is None:
      if self.p.pooling_type == 'AVG':
        return inputs.mean(dim=1)
      return inputs.max(dim=1).values
    mask = (1.0 - paddings).unsqueeze(-1)
    if self.p.pooling_type == 'AVG':
      total = (inputs * mask.to(inputs.dtype)).sum(dim=1)
      return total / mask.sum(dim=1).clamp_min(1.0).to(inputs.dtype)
    neg = inputs.masked_fill(mask == 0, float('-inf'))
    return neg.max(dim=1).values


class DepthwiseConv2DLayer(BaseLayer):
  """Depthwise 2-D conv on [B, T, F, C] with explicit time paddings
  (reference conv_layers_with_time_padding.py:608; channel multiplier
  supported)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('filter_shape', (3, 3, 32, 1),
             '(time, freq, in_channels, channel_multiplier).')
    p.Define('filter_stride', (1, 1), '(time, freq) stride.')
    p.Define('dilation_rate', (1, 1), '(time, freq) dilation.')
    p.Define('is_causal', False, 'Causal in time.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    th, fw, cin, mult = p.filter_shape
    self.CreateVariable('w', py_utils.WeightParams(
        [th, fw, cin, mult], p.params_init, p.dtype))

  def FProp(self, theta: NestedMap, inputs: torch.Tensor,
            paddings: torch.Tensor):
    p = self.p
    th, fw, cin, mult = p.filter_shape
    st, sf = p.filter_stride
    dt, df = p.dilation_rate
    x = py_utils.ApplyPadding(paddings, inputs)
    x = x.permute(0, 3, 1, 2)  # [B, C, T, F]
    eff_th = (th - 1) * dt + 1
    eff_fw = (fw - 1) * df + 1
    if p.is_causal:
      pad_t = (eff_th - 1, 0)
    else:
      pad_t = ((eff_th - 1) // 2, eff_th - 1 - (eff_th - 1) // 2)
    pad_f = ((eff_fw - 1) // 2, eff_fw - 1 - (eff_fw - 1) // 2)
    x = F.pad(x, (pad_f[0], pad_f[1], pad_t[0], pad_t[1]))
    # [th, fw, cin, mult] -> [cin*mult, 1, th, fw] grouped filter.
    w = theta.w.permute(2, 3, 0, 1).reshape(cin * mult, 1, th, fw)
    out = F.conv2d(x, w, stride=(st, sf), dilation=(dt, df), groups=cin)
    tout = out.shape[2]
    out = out.permute(0, 2, 3, 1)  # [B, T', F', C*mult]
    out_paddings = paddings[:, ::st][:, :tout]
    return py_utils.ApplyPadding(out_paddings, out), out_paddings


class CausalDepthwiseConv2DLayer(DepthwiseConv2DLayer):
  """Reference conv_layers_with_time_padding.py:717."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.is_causal = True
    return p
