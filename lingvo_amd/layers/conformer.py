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
    x = F.glu(x, dim=-1)  # fused GLU (single kernel fwd + bwd)
    if paddings is not None:
      x = py_utils.ApplyPadding(paddings, x)
    x = conv1d_ops.depthwise_conv1d(x, theta.dw_w, theta.dw_b,
                                    causal=p.is_causal)
    if p.conv_norm == 'layer':
      x = self.norm.FProp(theta.norm, x)
      if paddings is not None:
        x = py_utils.ApplyPadding(paddings, x)
      x = F.silu(x)
    elif p.conv_norm == 'group':
      # swish folds into the GroupNorm kernel (fwd epilogue + bwd chain)
      x = self.norm.FProp(theta.norm, x, paddings, act='SILU')
    else:
      x = self.norm.FProp(theta.norm, x, paddings)
      x = F.silu(x)
    x = py_utils.MatmulBias(x, theta.pw2_w, theta.pw2_b)
    if p.dropout_prob and not self.do_eval:
      # padding mask folds into the dropout kernel's elementwise pass.
      return py_utils.DeterministicDropoutAdd(x, 1.0 - p.dropout_prob,
                                              inputs, paddings=paddings)
    if paddings is not None:
      x = py_utils.ApplyPadding(paddings, x)
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
    x = F.glu(x, dim=-1)
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


class _Conv3x3S2Nhwc(torch.autograd.Function):
  """3x3 stride-2 same-pad conv on NHWC input via SPACE-TO-DEPTH +
  4 flat GEMMs.

  The input is space-to-depth'd once (2x2 cells -> 4C channels, block
  order [(a0,b1),(a1,b1),(a1,b0),(a0,b0)]) into a zero-padded flat
  buffer X [B*(Ho+1)*(Wo+1), 4C]. In s2d coordinates the 9 conv taps
  collapse into FOUR cell offsets {(-1,-1),(0,-1),(-1,0),(0,0)}; each
  is a UNIFORM ROW OFFSET of the flat buffer and touches a CONTIGUOUS
  channel range, so forward is 4 addmm_ calls on lda-strided views
  with in-GEMM accumulation — no im2col, no slice copies, and the
  backward's dX lands via 4 sequential in-place GEMMs (no strided
  scatter-adds). Two earlier formulations lost here: im2col's ~2 GB
  cols buffers overflowed 32-bit offsets in the upstream col2im (the
  round-1 GPU faults), and the 9-offset strided-slice form spent
  ~70 ms/step at B=512 on backward copies + skinny GEMMs.

  Tap map (cell = (dtau, dphi); within a cell, taps ordered by s2d
  block; bijection (cell, block) <-> (dt, df)):
    (-1,-1): [w[0,0]]                     blocks [s1]
    ( 0,-1): [w[1,0], w[2,0]]             blocks [s0, s1]
    (-1, 0): [w[0,2], w[0,1]]             blocks [s1, s2]
    ( 0, 0): [w[1,2], w[2,2], w[2,1], w[1,1]]  blocks [s0..s3]
  """

  CELLS = [
      ((-1, -1), [(0, 0, 1)]),
      ((0, -1), [(1, 0, 0), (2, 0, 1)]),
      ((-1, 0), [(0, 2, 1), (0, 1, 2)]),
      ((0, 0), [(1, 2, 0), (2, 2, 1), (2, 1, 2), (1, 1, 3)]),
  ]

  # s2d block order: s0=(0,1), s1=(1,1), s2=(1,0), s3=(0,0) in (a,b)
  # cell coordinates — the order that makes every cell's channel range
  # contiguous.
  BLOCKS = [(0, 1), (1, 1), (1, 0), (0, 0)]

  @staticmethod
  def _s2d(x):
    """[B,H,W,C] (H,W even) -> padded flat [B*(Ho+1)*(Wo+1), 4C].
    GPU bf16 with C%8==0 runs the one-pass HIP kernel; otherwise 4
    strided copies straight into the padded buffer."""
    B, H, W, C = x.shape
    Ho, Wo = H // 2, W // 2
    if x.is_cuda and x.dtype == torch.bfloat16 and C % 8 == 0:
      from lingvo_amd.ops import _loader
      X = _loader.get_ext(required=True).s2d_fwd(x)
      return X.reshape(B * (Ho + 1) * (Wo + 1), 4 * C), Ho, Wo
    X = x.new_zeros(B, Ho + 1, Wo + 1, 4 * C)
    for i, (a, b) in enumerate(_Conv3x3S2Nhwc.BLOCKS):
      X[:, 1:, 1:, i * C:(i + 1) * C] = x[:, a::2, b::2, :]
    return X.reshape(B * (Ho + 1) * (Wo + 1), 4 * C), Ho, Wo

  @staticmethod
  def _cell_weight(w, taps, transpose=False):
    """Per-cell weight [nblk*C, Co] from taps sorted by block."""
    mats = [w[dt, df] for (dt, df, _) in
            sorted(taps, key=lambda t: t[2])]
    cat = torch.cat(mats, dim=0)
    return cat.t().contiguous() if transpose else cat.contiguous()

  @staticmethod
  def forward(ctx, x, w, bias):
    """x [B,H,W,C] NHWC; w [3,3,C,Co] (kh,kw,cin,cout); bias [Co]."""
    B, H, W, C = x.shape
    Co = w.shape[3]
    pad_h, pad_w = H % 2, W % 2
    if pad_h or pad_w:
      x = F.pad(x, (0, 0, 0, pad_w, 0, pad_h))
    Hp2, Wp2 = x.shape[1], x.shape[2]
    X, Ho, Wo = _Conv3x3S2Nhwc._s2d(x)
    Wp = Wo + 1
    Rp = B * (Ho + 1) * Wp
    base = Wp + 1  # first real out row of batch 0
    O = bias.to(x.dtype).expand(Rp, Co).contiguous()
    for (dtau, dphi), taps in _Conv3x3S2Nhwc.CELLS:
      off = dtau * Wp + dphi
      blks = sorted(t[2] for t in taps)
      k0, k1 = blks[0] * C, (blks[-1] + 1) * C
      wc = _Conv3x3S2Nhwc._cell_weight(w.to(x.dtype), taps)
      O[base:].addmm_(X[base + off:Rp + off, k0:k1], wc)
    # Save the s2d buffer (not x): backward reuses it for the dW and
    # dX GEMMs without re-running the transform.
    ctx.save_for_backward(X, w)
    ctx.dims = (B, H, W, C, Ho, Wo, Co, pad_h, pad_w)
    ctx.bias_dtype = bias.dtype
    out = O.reshape(B, Ho + 1, Wp, Co)[:, 1:, 1:, :]
    return out.contiguous()

  @staticmethod
  def backward(ctx, dout):
    X, w = ctx.saved_tensors
    B, H, W, C, Ho, Wo, Co, pad_h, pad_w = ctx.dims
    Wp = Wo + 1
    Rp = B * (Ho + 1) * Wp
    base = Wp + 1
    dout4 = dout.reshape(B, Ho, Wo, Co)
    if dout.is_cuda and dout.dtype == torch.bfloat16 and Co % 8 == 0:
      from lingvo_amd.ops import _loader
      dO = _loader.get_ext(required=True).pad_scatter(
          dout4.contiguous())
    else:
      dO = dout.new_zeros(B, Ho + 1, Wp, Co)
      dO[:, 1:, 1:, :] = dout4
    dO = dO.reshape(Rp, Co)
    dX = torch.zeros_like(X)
    dw = torch.empty_like(w)
    for (dtau, dphi), taps in _Conv3x3S2Nhwc.CELLS:
      off = dtau * Wp + dphi
      blks = sorted(t[2] for t in taps)
      k0, k1 = blks[0] * C, (blks[-1] + 1) * C
      # dW_cell = X_slice^T @ dO: one GEMM per cell, rows split back to
      # taps (each block within a cell is exactly one tap).
      g = X[base + off:Rp + off, k0:k1].t() @ dO[base:]
      for i, (dt, df, _) in enumerate(sorted(taps,
                                             key=lambda t: t[2])):
        dw[dt, df] = g[i * C:(i + 1) * C].to(w.dtype)
      # dX rows/channels overlap across cells: sequential in-place.
      wc_t = _Conv3x3S2Nhwc._cell_weight(w.to(dO.dtype), taps,
                                         transpose=True)
      dX[base + off:Rp + off, k0:k1].addmm_(dO[base:], wc_t)
    # Inverse s2d: one-pass HIP kernel on GPU bf16, else strided
    # scatter of each block's channel slice.
    if dX.is_cuda and dX.dtype == torch.bfloat16 and C % 8 == 0:
      from lingvo_amd.ops import _loader
      dx = _loader.get_ext(required=True).s2d_inv(
          dX.reshape(B, Ho + 1, Wp, 4 * C), C)
    else:
      dXr = dX.reshape(B, Ho + 1, Wp, 4 * C)[:, 1:, 1:, :]
      dx_full = dout.new_empty(B, 2 * Ho, 2 * Wo, C)
      for i, (a, b) in enumerate(_Conv3x3S2Nhwc.BLOCKS):
        dx_full[:, a::2, b::2, :] = dXr[..., i * C:(i + 1) * C]
      dx = dx_full
    if pad_h or pad_w:
      dx = dx[:, :2 * Ho - pad_h, :2 * Wo - pad_w, :].contiguous()
    dbias = dO[base:].float().sum(0).to(ctx.bias_dtype)
    return dx, dw, dbias


class ConvSubsampling(BaseLayer):
  """2x Conv2D stride-2 frontend: [B, T, F] mel -> [B, T/4, D]
  (reference tasks/asr/encoder conv subsampling). conv1 (Cin=1, K=9) is
  a small unfold+GEMM; conv2 (Cin=Cout=C) runs the offset-GEMM
  _Conv3x3S2Nhwc path."""

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

  def FProp(self, theta: NestedMap, inputs: torch.Tensor,
            paddings: torch.Tensor):
    """inputs [B, T, F] -> (out [B, ceil(T/4), D], out_paddings)."""
    b, t, f = inputs.shape
    ch = self._ch
    # conv1 (Cin=1): unfold + GEMM, output NHWC directly. cols are tiny
    # (9 x T*F/4 per example).
    cols = F.unfold(inputs.unsqueeze(1), kernel_size=3, stride=2,
                    padding=1)  # [B, 9, HW]
    h1 = (f + 1) // 2
    t1 = (t + 1) // 2
    x = torch.matmul(cols.transpose(1, 2),
                     theta.conv1_w.reshape(9, ch)) + theta.conv1_b
    x = F.relu(x).reshape(b, t1, h1, ch)  # NHWC
    # conv2: offset-GEMM path (no cols buffer).
    x = F.relu(_Conv3x3S2Nhwc.apply(x, theta.conv2_w, theta.conv2_b))
    _, t4, f4, _ = x.shape
    x = x.reshape(b, t4, f4 * ch)
    out = torch.addmm(theta.proj_b, x.reshape(-1, f4 * ch),
                      theta.proj_w).reshape(b, t4, -1)
    out_paddings = paddings[:, ::2][:, ::2]
    out_paddings = out_paddings[:, :t4]
    out = py_utils.ApplyPadding(out_paddings, out)
    return out, out_paddings
