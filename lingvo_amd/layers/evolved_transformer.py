"""Evolved Transformer branched-convolution blocks.

MI355X-native re-implementation of the reference's Evolved Transformer
layers (lingvo/core/layers_with_attention.py:1807
EvolvedTransformerEncoderBranchedConvsLayer, :1885
EvolvedTransformerDecoderBranchedConvsLayer, and the encoder/decoder
wrapper layers; So et al. 2019). The NAS-found cell replaces the plain
FFN with two parallel branches (GELU linear + 3x1 conv, merged, then a
separable 9x1 conv in the encoder; 11x1/7x1 convs in the decoder).
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn.functional as F

from lingvo_amd.core import py_utils
from lingvo_amd.core.base_layer import BaseLayer
from lingvo_amd.core.nested_map import NestedMap
from lingvo_amd.layers import layers as lingvo_layers
from lingvo_amd.layers import transformer as transformer_lib


def _TimeConv(x, w, pad_left, pad_right):
  """[B,T,D] x [K,D,O] -> [B,T,O] time conv with explicit padding."""
  xt = F.pad(x.transpose(1, 2), (pad_left, pad_right))  # [B,D,T+]
  return F.conv1d(xt, w.permute(2, 1, 0)).transpose(1, 2)


class EvolvedTransformerEncoderBranchedConvsLayer(BaseLayer):
  """LN -> [GELU linear || 3x1 conv ReLU] -> merge -> LN ->
  separable 9x1 conv -> residual (reference :1807)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('input_dim', 0, 'Model dim D.')
    p.Define('dropout_prob', 0.0, 'Dropout.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    d = p.input_dim
    self.CreateChild('ln1', lingvo_layers.LayerNorm.Params().Set(
        input_dim=d))
    self.CreateChild('ln2', lingvo_layers.LayerNorm.Params().Set(
        input_dim=d * 4))
    self.CreateVariable('linear_w', py_utils.WeightParams(
        [d, d * 4], p.params_init, p.dtype))
    self.CreateVariable('linear_b', py_utils.WeightParams(
        [d * 4], py_utils.WeightInit.Constant(0.0), p.dtype))
    self.CreateVariable('conv3_w', py_utils.WeightParams(
        [3, d, d * 4], p.params_init, p.dtype))
    self.CreateVariable('conv3_b', py_utils.WeightParams(
        [d * 4], py_utils.WeightInit.Constant(0.0), p.dtype))
    # depthwise 9x1 + pointwise (separable) back to D
    self.CreateVariable('sep_dw_w', py_utils.WeightParams(
        [9, d * 4, 1], p.params_init, p.dtype))
    self.CreateVariable('sep_pw_w', py_utils.WeightParams(
        [d * 4, d], p.params_init, p.dtype))
    self.CreateVariable('sep_b', py_utils.WeightParams(
        [d], py_utils.WeightInit.Constant(0.0), p.dtype))

  def _Drop(self, x):
    if self.p.dropout_prob and not self.do_eval:
      return py_utils.DeterministicDropout(x, 1.0 - self.p.dropout_prob)
    return x

  def FProp(self, theta: NestedMap, inputs: torch.Tensor,
            paddings: Optional[torch.Tensor] = None) -> torch.Tensor:
    x = self.ln1.FProp(theta.ln1, inputs)
    if paddings is not None:
      x = py_utils.ApplyPadding(paddings, x)
    left = F.gelu(py_utils.MatmulBias(x, theta.linear_w, theta.linear_b))
    right = F.relu(_TimeConv(x, theta.conv3_w, 1, 1) + theta.conv3_b)
    h = self._Drop(left + right)
    h = self.ln2.FProp(theta.ln2, h)
    if paddings is not None:
      h = py_utils.ApplyPadding(paddings, h)
    # separable 9x1: depthwise over time then pointwise projection
    hd = F.conv1d(F.pad(h.transpose(1, 2), (4, 4)),
                  theta.sep_dw_w.permute(1, 2, 0),
                  groups=h.shape[-1]).transpose(1, 2)
    out = torch.matmul(hd, theta.sep_pw_w) + theta.sep_b
    out = self._Drop(out)
    res = inputs + out
    if paddings is not None:
      res = py_utils.ApplyPadding(paddings, res)
    return res


class EvolvedTransformerDecoderBranchedConvsLayer(BaseLayer):
  """LN -> [ReLU 11x1 conv || 7x1 conv] -> merge -> LN -> 7x1 conv ->
  residual, all causal (reference :1885)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('input_dim', 0, 'Model dim D.')
    p.Define('dropout_prob', 0.0, 'Dropout.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    d = p.input_dim
    self.CreateChild('ln1', lingvo_layers.LayerNorm.Params().Set(
        input_dim=d))
    self.CreateChild('ln2', lingvo_layers.LayerNorm.Params().Set(
        input_dim=d * 2))
    self.CreateVariable('conv11_w', py_utils.WeightParams(
        [11, d, d * 2], p.params_init, p.dtype))
    self.CreateVariable('conv11_b', py_utils.WeightParams(
        [d * 2], py_utils.WeightInit.Constant(0.0), p.dtype))
    self.CreateVariable('conv7a_w', py_utils.WeightParams(
        [7, d, d * 2], p.params_init, p.dtype))
    self.CreateVariable('conv7a_b', py_utils.WeightParams(
        [d * 2], py_utils.WeightInit.Constant(0.0), p.dtype))
    self.CreateVariable('conv7b_w', py_utils.WeightParams(
        [7, d * 2, d], p.params_init, p.dtype))
    self.CreateVariable('conv7b_b', py_utils.WeightParams(
        [d], py_utils.WeightInit.Constant(0.0), p.dtype))

  def _Drop(self, x):
    if self.p.dropout_prob and not self.do_eval:
      return py_utils.DeterministicDropout(x, 1.0 - self.p.dropout_prob)
    return x

  def FProp(self, theta: NestedMap, inputs: torch.Tensor,
            paddings: Optional[torch.Tensor] = None) -> torch.Tensor:
    x = self.ln1.FProp(theta.ln1, inputs)
    if paddings is not None:
      x = py_utils.ApplyPadding(paddings, x)
    # causal convs: left-pad only
    left = F.relu(_TimeConv(x, theta.conv11_w, 10, 0) + theta.conv11_b)
    right = _TimeConv(x, theta.conv7a_w, 6, 0) + theta.conv7a_b
    h = self._Drop(left + right)
    h = self.ln2.FProp(theta.ln2, h)
    if paddings is not None:
      h = py_utils.ApplyPadding(paddings, h)
    out = _TimeConv(h, theta.conv7b_w, 6, 0) + theta.conv7b_b
    out = self._Drop(out)
    res = inputs + out
    if paddings is not None:
      res = py_utils.ApplyPadding(paddings, res)
    return res


class EvolvedTransformerEncoderLayer(BaseLayer):
  """Branched-convs cell + self-attention block + FFN
  (reference EvolvedTransformerEncoderLayer)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('input_dim', 0, 'Model dim.')
    p.Define('num_heads', 8, 'Attention heads.')
    p.Define('hidden_dim', 0, 'FFN hidden (0 = 4x).')
    p.Define('dropout_prob', 0.0, 'Dropout.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    d = p.input_dim
    self.CreateChild(
        'branched_convs',
        EvolvedTransformerEncoderBranchedConvsLayer.Params().Set(
            input_dim=d, dropout_prob=p.dropout_prob))
    self.CreateChild(
        'atten', transformer_lib.TransformerAttentionLayer.Params().Set(
            input_dim=d, num_heads=p.num_heads,
            residual_dropout_prob=p.dropout_prob))
    self.CreateChild(
        'ffn', transformer_lib.TransformerFeedForwardLayer.Params().Set(
            input_dim=d, hidden_dim=p.hidden_dim or 4 * d,
            residual_dropout_prob=p.dropout_prob))

  def FProp(self, theta: NestedMap, inputs: torch.Tensor,
            paddings: Optional[torch.Tensor] = None) -> torch.Tensor:
    x = self.branched_convs.FProp(theta.branched_convs, inputs, paddings)
    x = self.atten.FProp(theta.atten, x, paddings)
    return self.ffn.FProp(theta.ffn, x, paddings)
