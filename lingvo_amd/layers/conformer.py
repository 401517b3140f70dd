"""Conformer block (reference lingvo/core/conformer_layer.py:35 LConvLayer,
:471 ConformerLayer, CommonParams :566).

Block order (the paper's and the reference default 'mhsa_before_conv'):
  x += 0.5*FFN(x); x += MHSA(x); x += LConv(x); x += 0.5*FFN(x); x = LN(x)
LConv: LN -> pointwise 2D GLU -> depthwise time conv (HIP kernel K7)
 -> norm -> swish -> pointwise -> dropout.
"""

from __future__ import annotations

import os
from typing import Optional

import torch
import torch.nn.functional as F

from lingvo_amd.core import py_utils
from lingvo_amd.core.base_layer import BaseLayer
from lingvo_amd.core.nested_map import NestedMap
from lingvo_amd.layers import attention as attention_lib
from lingvo_amd.layers import bn_layers
from lingvo_amd.layers import layers as lingvo_layers
from lingvo_amd.layers import transformer as transformer_lib
from lingvo_amd.ops import conv1d as conv1d_ops


class LConvLayer(BaseLayer):
  """Lightweight convolution module (reference conformer_layer.py:35)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('input_dim', 0, 'Model dim.')
    p.Define('kernel_size', 32, 'Depthwise conv kernel size.')
    p.Define('is_causal', False, 'Causal depthwise conv.')
    p.Define('conv_norm', 'group', "One of 'batch'|'group'|'layer'.")
    p.Define('num_groups', 32, 'Groups for group norm.')
    p.Define('dropout_prob', 0.0, 'Output dropout.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    d = p.input_dim
    self.CreateChild('ln', lingvo_layers.LayerNorm.Params().Set(
        input_dim=d))
    self.CreateVariable('pw1_w', py_utils.WeightParams(
        [d, 2 * d], p.params_init, p.dtype))
    self.CreateVariable('pw1_b', py_utils.WeightParams(
        [2 * d], py_utils.WeightInit.Constant(0.0), p.dtype))
    self.CreateVariable('dw_w', py_utils.WeightParams(
        [p.kernel_size, d], p.params_init, p.dtype))
    self.CreateVariable('dw_b', py_utils.WeightParams(
        [d], py_utils.WeightInit.Constant(0.0), p.dtype))
    if p.conv_norm == 'batch':
      self.CreateChild('norm', bn_layers.BatchNormLayer.Params().Set(dim=d))
    elif p.conv_norm == 'group':
      self.CreateChild('norm', bn_layers.GroupNormLayer.Params().Set(
          dim=d, num_groups=p.num_groups))
    else:
      self.CreateChild('norm', lingvo_layers.LayerNorm.Params().Set(
          input_dim=d))
    self.CreateVariable('pw2_w', py_utils.WeightParams(
        [d, d], p.params_init, p.dtype))
    self.CreateVariable('pw2_b', py_utils.WeightParams(
        [d], py_utils.WeightInit.Constant(0.0), p.dtype))

  def FProp(self, theta: NestedMap, inputs: torch.Tensor,
            paddings: Optional[torch.Tensor] = None) -> torch.Tensor:
    p = self.p
    x = self.ln.FProp(theta.ln, inputs)
    x = py_utils.MatmulBias(x, theta.pw1_w, theta.pw1_b)
    a, b = x.chunk(2, dim=-1)
    x = a * torch.sigmoid(b)  # GLU
    if paddings is not None:
      x = py_utils.ApplyPadding(paddings, x)
    x = conv1d_ops.depthwise_conv1d(x, theta.dw_w, theta.dw_b,
                                    causal=p.is_causal)
    if p.conv_norm == 'layer':
      x = self.norm.FProp(theta.norm, x)
      if paddings is not None:
        x = py_utils.ApplyPadding(paddings, x)
    else:
      x = self.norm.FProp(theta.norm, x, paddings)
    x = F.silu(x)
    x = py_utils.MatmulBias(x, theta.pw2_w, theta.pw2_b)
    if paddings is not None:
      x = py_utils.ApplyPadding(paddings, x)
    if p.dropout_prob and not self.do_eval:
      return py_utils.DeterministicDropoutAdd(x, 1.0 - p.dropout_prob,
                                              inputs)
    return inputs + x

  def InitStreamState(self, batch: int, device, dtype) -> NestedMap:
    p = self.p
    return NestedMap(conv=torch.zeros(batch, p.kernel_size - 1,
                                      p.input_dim, device=device,
                                      dtype=dtype))

  def StreamStep(self, theta: NestedMap, inputs: torch.Tensor,
                 paddings: torch.Tensor, state: NestedMap):
    """Causal streaming chunk (reference conformer_layer.py:390). Requires
    is_causal=True and a stream-safe norm (conv_norm='layer')."""
    p = self.p
    assert p.is_causal, 'StreamStep requires a causal LConv'
    x = self.ln.FProp(theta.ln, inputs)
    x = py_utils.MatmulBias(x, theta.pw1_w, theta.pw1_b)
    a, b = x.chunk(2, dim=-1)
    x = a * torch.sigmoid(b)
    x = py_utils.ApplyPadding(paddings, x)
    # Depthwise causal conv over [state | chunk].
    full = torch.cat([state.conv.to(x.dtype), x], dim=1)
    y = conv1d_ops.depthwise_conv1d(full, theta.dw_w, theta.dw_b,
                                    causal=True)
    x = y[:, state.conv.shape[1]:]
    new_state = NestedMap(conv=full[:, -(p.kernel_size - 1):]
                          if p.kernel_size > 1 else state.conv)
    if p.conv_norm == 'layer':
      x = self.norm.FProp(theta.norm, x)
      x = py_utils.ApplyPadding(paddings, x)
    else:
      x = self.norm.FProp(theta.norm, x, paddings)
    x = F.silu(x)
    x = py_utils.MatmulBias(x, theta.pw2_w, theta.pw2_b)
    x = py_utils.ApplyPadding(paddings, x)
    return inputs + x, new_state


class ConformerLayer(BaseLayer):
  """½FFN -> MHSA -> LConv -> ½FFN -> LN (reference conformer_layer.py:471)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('input_dim', 0, 'Model dim.')
    p.Define('atten_num_heads', 8, 'MHSA heads.')
    p.Define('atten_left_context', -1, 'Local attention left window.')
    p.Define('atten_right_context', -1, 'Local attention right window.')
    p.Define('use_relative_atten', True, 'Clipped rel-pos bias in MHSA.')
    p.Define('rel_pos_clip', 127, 'Rel-pos clip distance.')
    p.Define('fflayer_hidden_dim', 0, 'FFN hidden (0 = 4x).')
    p.Define('kernel_size', 32, 'LConv kernel size.')
    p.Define('is_causal', False, 'Causal conv + attention.')
    p.Define('conv_norm', 'group', 'LConv norm type.')
    p.Define('dropout_prob', 0.0, 'Dropout throughout.')
    p.Define('remat', False, 'Gradient-checkpoint this layer '
             '(reference conformer_layer.py:548 p.remat).')
    p.Define('moe_num_experts', 0,
             'If > 0, the ENDING half-FFN becomes a top-2 MoE FFN '
             '(reference conformer_layer.py:1006 MoE-FFN option).')
    p.Define('moe_capacity_factor', 2.0, 'MoE capacity factor.')
    return p

  @classmethod
  def CommonParams(cls, input_dim, atten_num_heads=8, kernel_size=32,
                   is_causal=False, dropout_prob=0.0):
    return cls.Params().Set(
        input_dim=input_dim, atten_num_heads=atten_num_heads,
        kernel_size=kernel_size, is_causal=is_causal,
        dropout_prob=dropout_prob)

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    d = p.input_dim
    hidden = p.fflayer_hidden_dim or 4 * d
    ff = transformer_lib.TransformerFeedForwardLayer.Params().Set(
        input_dim=d, hidden_dim=hidden, activation='SWISH',
        residual_weight=0.5, residual_dropout_prob=p.dropout_prob,
        relu_dropout_prob=p.dropout_prob)
    self.CreateChild('fflayer_start', ff.Copy())
    if p.moe_num_experts:
      self.CreateChild(
          'fflayer_end',
          transformer_lib.MoETransformerFeedForwardLayer.Params().Set(
              input_dim=d, hidden_dim=hidden,
              num_experts=p.moe_num_experts,
              expert_capacity_factor=p.moe_capacity_factor,
              residual_dropout_prob=p.dropout_prob))
    else:
      self.CreateChild('fflayer_end', ff.Copy())
    atten = transformer_lib.TransformerAttentionLayer.Params().Set(
        input_dim=d, num_heads=p.atten_num_heads, is_masked=p.is_causal,
        residual_dropout_prob=p.dropout_prob)
    atten.atten_tpl.rel_pos_bias = p.use_relative_atten
    atten.atten_tpl.rel_pos_clip = p.rel_pos_clip
    atten.atten_tpl.left_context = p.atten_left_context
    atten.atten_tpl.right_context = p.atten_right_context
    atten.atten_tpl.atten_dropout_prob = p.dropout_prob
    self.CreateChild('trans_atten', atten)
    self.CreateChild('lconv', LConvLayer.Params().Set(
        input_dim=d, kernel_size=p.kernel_size, is_causal=p.is_causal,
        conv_norm=p.conv_norm, dropout_prob=p.dropout_prob))
    self.CreateChild('final_ln', lingvo_layers.LayerNorm.Params().Set(
        input_dim=d))

  def _Body(self, theta, x, paddings):
    x = self.fflayer_start.FProp(theta.fflayer_start, x, paddings)
    x = self.trans_atten.FProp(theta.trans_atten, x, paddings)
    x = self.lconv.FProp(theta.lconv, x, paddings)
    x = self.fflayer_end.FProp(theta.fflayer_end, x, paddings)
    return self.final_ln.FProp(theta.final_ln, x)

  def FProp(self, theta: NestedMap, inputs: torch.Tensor,
            paddings: Optional[torch.Tensor] = None) -> torch.Tensor:
    if self.p.remat and self.training:
      return torch.utils.checkpoint.checkpoint(
          lambda x: self._Body(theta, x, paddings), inputs,
          use_reentrant=False)
    return self._Body(theta, inputs, paddings)

  # ---- streaming (reference conformer_layer.py:390 StreamStep) --------
  def InitStreamState(self, theta: NestedMap, batch: int, max_len: int,
                      device, dtype=torch.float32) -> NestedMap:
    """Requires is_causal=True, conv_norm='layer' and a finite
    atten_left_context (or max_len-bounded history)."""
    assert self.p.is_causal, 'streaming requires a causal block'
    return NestedMap(
        atten=self.trans_atten.InitStates(theta.trans_atten, batch,
                                          max_len, device, dtype),
        lconv=self.lconv.InitStreamState(batch, device, dtype))

  def StreamStep(self, theta: NestedMap, x_chunk: torch.Tensor,
                 paddings_chunk: torch.Tensor, state: NestedMap):
    x = self.fflayer_start.FProp(theta.fflayer_start, x_chunk,
                                 paddings_chunk)
    x, state.atten = self.trans_atten.StreamStep(
        theta.trans_atten, x, paddings_chunk, state.atten)
    x, state.lconv = self.lconv.StreamStep(theta.lconv, x,
                                           paddings_chunk, state.lconv)
    x = self.fflayer_end.FProp(theta.fflayer_end, x, paddings_chunk)
    return self.final_ln.FProp(theta.final_ln, x), state


class ConvSubsampling(BaseLayer):
  """2x Conv2D stride-2 frontend: [B, T, F] mel -> [B, T/4, D]
  (reference tasks/asr/encoder conv subsampling)."""

  # im2col buffers near 2 GB trip 32-bit byte-offset overflows in the
  # upstream col2im/baddbmm backward kernels (observed as GPU write
  # faults whose occurrence depends on heap layout — box-to-box
  # nondeterminism at B=128, hard fault at B>=160). Keep chunks far
  # below the edge: backward materializes fp32/extra copies of the cols
  # buffer, so the safe bound is ~2^29 bytes, not 2^31.
  # Env-overridable for fault-localization tests.
  MAX_COLS_BYTES = int(os.environ.get('LINGVO_AMD_MAX_COLS_BYTES',
                                      2 ** 28))

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('input_freq_dim', 80, 'Mel bins.')
    p.Define('channels', 0, 'Conv channels (defaults to output_dim).')
    p.Define('output_dim', 512, 'Output model dim.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    ch = p.channels or p.output_dim
    self._ch = ch
    self.CreateVariable('conv1_w', py_utils.WeightParams(
        [3, 3, 1, ch], p.params_init, p.dtype))
    self.CreateVariable('conv1_b', py_utils.WeightParams(
        [ch], py_utils.WeightInit.Constant(0.0), p.dtype))
    self.CreateVariable('conv2_w', py_utils.WeightParams(
        [3, 3, ch, ch], p.params_init, p.dtype))
    self.CreateVariable('conv2_b', py_utils.WeightParams(
        [ch], py_utils.WeightInit.Constant(0.0), p.dtype))
    freq_out = ((p.input_freq_dim + 1) // 2 + 1) // 2
    self.CreateVariable('proj_w', py_utils.WeightParams(
        [freq_out * ch, p.output_dim], p.params_init, p.dtype))
    self.CreateVariable('proj_b', py_utils.WeightParams(
        [p.output_dim], py_utils.WeightInit.Constant(0.0), p.dtype))

  @staticmethod
  def _ConvGemm(x: torch.Tensor, w_oihw: torch.Tensor, bias: torch.Tensor,
                stride: int = 2, pad: int = 1) -> torch.Tensor:
    """3x3 strided conv as im2col + hipBLASLt GEMM. MIOpen's algorithm
    search can fall back to a naive NCHW kernel for these shapes on
    gfx950 (observed ~1000x regression under rocprof); the unfold+GEMM
    path always lands on Tensile.

    The batch is chunked so the im2col buffer stays under 2^31 BYTES:
    beyond that, 32-bit byte offsets overflow in the col2im backward
    (observed as a GPU write fault at per-buffer >= 2.2 GB)."""
    o = w_oihw.shape[0]
    bsz, cin = x.shape[0], x.shape[1]
    hout = (x.shape[2] + 2 * pad - 3) // stride + 1
    wout = (x.shape[3] + 2 * pad - 3) // stride + 1
    cols_bytes_per_ex = cin * 9 * hout * wout * x.element_size()
    max_chunk = max(1, int(ConvSubsampling.MAX_COLS_BYTES //
                           cols_bytes_per_ex))
    outs = []
    for s in range(0, bsz, max_chunk):
      xc = x[s:s + max_chunk]
      bc = xc.shape[0]
      cols = F.unfold(xc, kernel_size=3, stride=stride, padding=pad)
      out = torch.baddbmm(
          bias.reshape(1, o, 1),
          w_oihw.reshape(1, o, -1).expand(bc, -1, -1), cols)
      outs.append(out.reshape(bc, o, hout, wout))
    return outs[0] if len(outs) == 1 else torch.cat(outs, dim=0)

  def FProp(self, theta: NestedMap, inputs: torch.Tensor,
            paddings: torch.Tensor):
    """inputs [B, T, F] -> (out [B, ceil(T/4), D], out_paddings)."""
    x = inputs.unsqueeze(1)  # [B,1,T,F]
    w1 = theta.conv1_w.permute(3, 2, 0, 1).contiguous()
    x = F.relu(self._ConvGemm(x, w1, theta.conv1_b))
    w2 = theta.conv2_w.permute(3, 2, 0, 1).contiguous()
    x = F.relu(self._ConvGemm(x, w2, theta.conv2_b))
    b, ch, t4, f4 = x.shape
    x = x.permute(0, 2, 3, 1).reshape(b, t4, f4 * ch)
    out = torch.addmm(theta.proj_b, x.reshape(-1, f4 * ch),
                      theta.proj_w).reshape(b, t4, -1)
    out_paddings = paddings[:, ::2][:, ::2]
    out_paddings = out_paddings[:, :t4]
    out = py_utils.ApplyPadding(out_paddings, out)
    return out, out_paddings
