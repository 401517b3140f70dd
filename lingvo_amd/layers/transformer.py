"""Transformer assembly layers (reference
lingvo/core/batch_major_attention.py:5226 TransformerAttentionLayer,
:6265 TransformerLayer, :7116 StackedTransformerLayers;
layers_with_attention.py:529 TransformerFeedForwardLayer)."""

from __future__ import annotations

from typing import Optional

import torch

from lingvo_amd.core import py_utils
from lingvo_amd.core.base_layer import BaseLayer
from lingvo_amd.core.nested_map import NestedMap
from lingvo_amd.layers import activations
from lingvo_amd.layers import attention as attention_lib
from lingvo_amd.layers import layers as lingvo_layers


class TransformerAttentionLayer(BaseLayer):
  """Pre-LN self/cross attention block with residual."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('input_dim', 0, 'Model dim.')
    p.Define('num_heads', 8, 'Heads.')
    p.Define('is_masked', False, 'Causal self-attention.')
    p.Define('atten_tpl', attention_lib.MultiHeadedAttention.Params(),
             'Attention template.')
    p.Define('residual_dropout_prob', 0.0, 'Residual dropout.')
    p.Define('ln_tpl', lingvo_layers.LayerNorm.Params(), 'LayerNorm tpl.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    atten_p = p.atten_tpl.Copy()
    atten_p.input_dim = p.input_dim
    atten_p.hidden_dim = atten_p.hidden_dim or p.input_dim
    atten_p.num_heads = p.num_heads
    atten_p.causal = p.is_masked
    self.CreateChild('atten', atten_p)
    ln_p = p.ln_tpl.Copy()
    ln_p.input_dim = p.input_dim
    self.CreateChild('layer_norm', ln_p)

  def FProp(self, theta: NestedMap, query_vec: torch.Tensor,
            paddings: Optional[torch.Tensor] = None,
            source_vecs: Optional[torch.Tensor] = None,
            source_paddings: Optional[torch.Tensor] = None,
            segment_ids: Optional[torch.Tensor] = None) -> torch.Tensor:
    p = self.p
    x = self.layer_norm.FProp(theta.layer_norm, query_vec)
    if source_vecs is None:
      ctx = self.atten.FProp(theta.atten, x, paddings,
                             segment_ids=segment_ids)
    else:
      ctx = self.atten.FPropCross(theta.atten, x, source_vecs, source_vecs,
                                  source_paddings)
    if p.residual_dropout_prob and not self.do_eval:
      return py_utils.DeterministicDropoutAdd(
          ctx, 1.0 - p.residual_dropout_prob, query_vec)
    return query_vec + ctx

  def InitStates(self, theta, batch, max_len, device, dtype=torch.bfloat16):
    return self.atten.InitStates(theta.atten, batch, max_len, device, dtype)

  def ExtendStep(self, theta, query_vec, cached_states):
    x = self.layer_norm.FProp(theta.layer_norm, query_vec)
    ctx, states = self.atten.ExtendStep(theta.atten, x, cached_states)
    return query_vec + ctx, states

  def StreamStep(self, theta, x_chunk, paddings_chunk, state):
    """Chunked streaming (causal; left-context-bounded KV history)."""
    x = self.layer_norm.FProp(theta.layer_norm, x_chunk)
    ctx, state = self.atten.StreamStep(theta.atten, x, paddings_chunk,
                                       state)
    return x_chunk + ctx, state


class TransformerFeedForwardLayer(BaseLayer):
  """Pre-LN FFN: LN -> FC(hidden) -> act -> dropout -> FC(out) -> residual
  (reference layers_with_attention.py:529)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('input_dim', 0, 'Model dim.')
    p.Define('hidden_dim', 0, 'FFN hidden dim.')
    p.Define('activation', 'RELU', 'Activation.')
    p.Define('residual_dropout_prob', 0.0, 'Residual dropout.')
    p.Define('relu_dropout_prob', 0.0, 'Hidden dropout.')
    p.Define('residual_weight', 1.0, 'Residual scale (0.5 for macaron).')
    p.Define('ln_tpl', lingvo_layers.LayerNorm.Params(), 'LayerNorm tpl.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    ln_p = p.ln_tpl.Copy()
    ln_p.input_dim = p.input_dim
    self.CreateChild('layer_norm', ln_p)
    self.CreateVariable('w1', py_utils.WeightParams(
        [p.input_dim, p.hidden_dim], p.params_init, p.dtype))
    self.CreateVariable('b1', py_utils.WeightParams(
        [p.hidden_dim], py_utils.WeightInit.Constant(0.0), p.dtype))
    self.CreateVariable('w2', py_utils.WeightParams(
        [p.hidden_dim, p.input_dim], p.params_init, p.dtype))
    self.CreateVariable('b2', py_utils.WeightParams(
        [p.input_dim], py_utils.WeightInit.Constant(0.0), p.dtype))

  def FProp(self, theta: NestedMap, inputs: torch.Tensor,
            paddings: Optional[torch.Tensor] = None) -> torch.Tensor:
    p = self.p
    x = self.layer_norm.FProp(theta.layer_norm, inputs)
    h = py_utils.MatmulBias(x, theta.w1, theta.b1)
    if (p.relu_dropout_prob and not self.do_eval and
        p.activation in ('SWISH', 'RELU')):
      # Activation fused into the dropout kernel: one pass over the
      # [*, hidden] tensor instead of two, each direction.
      h = py_utils.DeterministicDropout(h, 1.0 - p.relu_dropout_prob,
                                        act=p.activation)
    else:
      h = activations.GetFn(p.activation)(h)
      if p.relu_dropout_prob and not self.do_eval:
        h = py_utils.DeterministicDropout(h, 1.0 - p.relu_dropout_prob)
    out = py_utils.MatmulBias(h, theta.w2, theta.b2)
    if p.residual_dropout_prob and not self.do_eval:
      # residual weight + padding mask fold into the dropout kernel.
      return py_utils.DeterministicDropoutAdd(
          out, 1.0 - p.residual_dropout_prob, inputs,
          scale=p.residual_weight, paddings=paddings)
    if p.residual_weight != 1.0:
      out = out * p.residual_weight
    if paddings is not None:
      out = py_utils.ApplyPadding(paddings, out)
    return inputs + out


class MoETransformerFeedForwardLayer(BaseLayer):
  """Pre-LN MoE FFN block: LN -> top-2 MoE -> residual (reference MoE
  transformer via MoEBuilder, gshard_builder.py:55; layers_with_attention
  TransformerFeedForwardLayer MoE variant :832)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('input_dim', 0, 'Model dim.')
    p.Define('hidden_dim', 0, 'Expert hidden dim.')
    p.Define('num_experts', 8, 'Experts.')
    p.Define('expert_capacity_factor', 2.0, 'Capacity factor.')
    p.Define('residual_dropout_prob', 0.0, 'Residual dropout.')
    p.Define('aux_loss_weight', 0.01, 'Load-balance loss weight.')
    p.Define('ln_tpl', lingvo_layers.LayerNorm.Params(), 'LN template.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    self.CreateChild('layer_norm', p.ln_tpl.Copy().Set(
        input_dim=p.input_dim))
    from lingvo_amd.parallel import moe as moe_lib
    self.CreateChild('moe', moe_lib.MoEFeedForwardLayer.Params().Set(
        input_dim=p.input_dim, hidden_dim=p.hidden_dim,
        num_experts=p.num_experts,
        expert_capacity_factor=p.expert_capacity_factor,
        aux_loss_weight=p.aux_loss_weight))

  def FProp(self, theta: NestedMap, inputs: torch.Tensor,
            paddings: Optional[torch.Tensor] = None) -> torch.Tensor:
    p = self.p
    x = self.layer_norm.FProp(theta.layer_norm, inputs)
    out = self.moe.FProp(theta.moe, x, paddings)
    if p.residual_dropout_prob and not self.do_eval:
      return py_utils.DeterministicDropoutAdd(
          out, 1.0 - p.residual_dropout_prob, inputs)
    return inputs + out

  def AuxLoss(self):
    return self.moe.AuxLoss()


class TransformerLayer(BaseLayer):
  """Self-attention (+ optional cross-attention) + FFN
  (reference batch_major_attention.py:6265)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('input_dim', 0, 'Model dim.')
    p.Define('num_heads', 8, 'Heads.')
    p.Define('mask_self_atten', False, 'Causal self attention.')
    p.Define('has_aux_atten', False, 'Add cross-attention (decoder).')
    p.Define('tr_atten_tpl', TransformerAttentionLayer.Params(),
             'Self-attention template.')
    p.Define('tr_fflayer_tpl', TransformerFeedForwardLayer.Params(),
             'FFN template (may be MoETransformerFeedForwardLayer).')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    atten_p = p.tr_atten_tpl.Copy().Set(
        input_dim=p.input_dim, num_heads=p.num_heads,
        is_masked=p.mask_self_atten)
    self.CreateChild('self_atten', atten_p)
    if p.has_aux_atten:
      aux_p = p.tr_atten_tpl.Copy().Set(
          input_dim=p.input_dim, num_heads=p.num_heads, is_masked=False)
      self.CreateChild('cross_atten', aux_p)
    ff_p = p.tr_fflayer_tpl.Copy()
    ff_p.input_dim = p.input_dim
    ff_p.hidden_dim = ff_p.hidden_dim or 4 * p.input_dim
    self.CreateChild('fflayer', ff_p)

  def FProp(self, theta: NestedMap, inputs: torch.Tensor,
            paddings: Optional[torch.Tensor] = None,
            aux_vecs: Optional[torch.Tensor] = None,
            aux_paddings: Optional[torch.Tensor] = None,
            segment_ids: Optional[torch.Tensor] = None) -> torch.Tensor:
    x = self.self_atten.FProp(theta.self_atten, inputs, paddings,
                              segment_ids=segment_ids)
    if self.p.has_aux_atten:
      x = self.cross_atten.FProp(theta.cross_atten, x, paddings,
                                 source_vecs=aux_vecs,
                                 source_paddings=aux_paddings)
    return self.fflayer.FProp(theta.fflayer, x, paddings)

  def InitStates(self, theta, batch, max_len, device, dtype=torch.bfloat16):
    return NestedMap(
        self_atten=self.self_atten.InitStates(theta.self_atten, batch,
                                              max_len, device, dtype))

  def ExtendStep(self, theta, inputs, cached_states, aux_vecs=None,
                 aux_paddings=None):
    x, st = self.self_atten.ExtendStep(theta.self_atten, inputs,
                                       cached_states.self_atten)
    cached_states.self_atten = st
    if self.p.has_aux_atten:
      x = self.cross_atten.FProp(theta.cross_atten, x, None,
                                 source_vecs=aux_vecs,
                                 source_paddings=aux_paddings)
    out = self.fflayer.FProp(theta.fflayer, x)
    return out, cached_states


class StackedTransformerLayers(BaseLayer):
  """N transformer layers + final LN
  (reference batch_major_attention.py:7116)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('model_dim', 0, 'Model dim.')
    p.Define('num_layers', 6, 'Layers.')
    p.Define('num_heads', 8, 'Heads.')
    p.Define('hidden_dim', 0, 'FFN hidden (0 = 4x model).')
    p.Define('mask_self_atten', False, 'Causal.')
    p.Define('has_aux_atten', False, 'Cross-attention per layer.')
    p.Define('transformer_tpl', TransformerLayer.Params(), 'Layer tpl.')
    p.Define('final_ln', True, 'Final LayerNorm.')
    p.Define('remat', False, 'Gradient-checkpoint each layer '
             '(reference RematerializeFn py_utils.py:5005).')
    p.Define('moe_every_n', 0,
             'If >0, every n-th layer uses an MoE FFN (GShard pattern).')
    p.Define('moe_tpl', MoETransformerFeedForwardLayer.Params(),
             'MoE FFN template used by moe_every_n layers.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    layer_ps = []
    for i in range(p.num_layers):
      lp = p.transformer_tpl.Copy().Set(
          name=f'layer_{i}', input_dim=p.model_dim, num_heads=p.num_heads,
          mask_self_atten=p.mask_self_atten, has_aux_atten=p.has_aux_atten)
      if p.moe_every_n and i % p.moe_every_n == p.moe_every_n - 1:
        lp.tr_fflayer_tpl = p.moe_tpl.Copy()
      if p.hidden_dim:
        lp.tr_fflayer_tpl.hidden_dim = p.hidden_dim
      layer_ps.append(lp)
    self.CreateChildren('x_layers', layer_ps)
    if p.final_ln:
      self.CreateChild('final_layer_norm',
                       lingvo_layers.LayerNorm.Params().Set(
                           input_dim=p.model_dim))

  def FProp(self, theta: NestedMap, inputs: torch.Tensor,
            paddings: Optional[torch.Tensor] = None,
            aux_vecs=None, aux_paddings=None,
            segment_ids: Optional[torch.Tensor] = None) -> torch.Tensor:
    """Optional segment_ids [B, T] run packed-input training (the
    reference's packed segment masks, PackedBatchMajorLanguageModel
    model.py:408): cross-segment attention is blocked in-kernel."""
    p = self.p
    x = inputs
    for i, layer in enumerate(self.x_layers):
      if p.remat and self.training:
        x = torch.utils.checkpoint.checkpoint(
            lambda x_, i_=i: self.x_layers[i_].FProp(
                theta.x_layers[i_], x_, paddings, aux_vecs, aux_paddings,
                segment_ids),
            x, use_reentrant=False)
      else:
        x = layer.FProp(theta.x_layers[i], x, paddings, aux_vecs,
                        aux_paddings, segment_ids)
    if p.final_ln:
      x = self.final_layer_norm.FProp(theta.final_layer_norm, x)
    return x

  def InitStates(self, theta, batch, max_len, device, dtype=torch.bfloat16):
    return NestedMap(layers=[
        l.InitStates(theta.x_layers[i], batch, max_len, device, dtype)
        for i, l in enumerate(self.x_layers)
    ])

  def ExtendStep(self, theta, inputs, cached_states, aux_vecs=None,
                 aux_paddings=None):
    x = inputs
    for i, layer in enumerate(self.x_layers):
      x, st = layer.ExtendStep(theta.x_layers[i], x,
                               cached_states.layers[i], aux_vecs,
                               aux_paddings)
      cached_states.layers[i] = st
    if self.p.final_ln:
      x = self.final_layer_norm.FProp(theta.final_layer_norm, x)
    return x, cached_states
