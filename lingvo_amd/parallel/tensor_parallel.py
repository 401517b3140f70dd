"""Explicit tensor parallelism over RCCL/xGMI.

The reference expresses TP as GShard/GSPMD sharding annotations
(gshard_utils.py:39-137 Split/MeshSplit; per-layer device_mesh +
weight_split_dims_mapping on BaseLayer, base_layer.py:262-280) and lets
the XLA partitioner insert collectives. The MI355X-native lowering is
explicit Megatron-style TP: column-parallel Wq/Wk/Wv/W1 and row-parallel
Wo/W2 with an all-reduce at block boundaries (DenseBuilder sharding
pattern, gshard_builder.py:2269; tasks/lm/README.md:100-115), using
autograd-aware collectives on the TP process group.

ShardTransformerStackForTp() is the planner: it rewrites a
StackedTransformerLayers params tree whose layers carry
weight_split_dims_mapping annotations into TP-parallel attention/FFN.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist

from lingvo_amd.core import py_utils
from lingvo_amd.core.base_layer import BaseLayer
from lingvo_amd.core.nested_map import NestedMap


class _CopyToTp(torch.autograd.Function):
  """Identity fwd; all-reduce grads in bwd (column-parallel input)."""

  @staticmethod
  def forward(ctx, x, group):
    ctx.group = group
    return x

  @staticmethod
  def backward(ctx, g):
    if dist.is_initialized() and dist.get_world_size(ctx.group) > 1:
      g = g.contiguous()
      dist.all_reduce(g, group=ctx.group)
    return g, None


class _ReduceFromTp(torch.autograd.Function):
  """All-reduce fwd; identity bwd (row-parallel output)."""

  @staticmethod
  def forward(ctx, x, group):
    if dist.is_initialized() and dist.get_world_size(group) > 1:
      x = x.contiguous()
      dist.all_reduce(x, group=group)
    return x

  @staticmethod
  def backward(ctx, g):
    return g, None


def _TpInfo(group=None):
  if dist.is_available() and dist.is_initialized():
    return dist.get_world_size(group), dist.get_rank(group)
  return 1, 0


class ColumnParallelLinear(BaseLayer):
  """Y_local = X @ W[:, shard] (+ b[shard]); output stays sharded."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('input_dim', 0, 'Input dim.')
    p.Define('output_dim', 0, 'FULL output dim (sharded over TP).')
    p.Define('has_bias', True, 'Bias.')
    p.Define('tp_group', None, 'TP process group.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    world, rank = _TpInfo(p.tp_group)
    assert p.output_dim % world == 0
    self._shard = p.output_dim // world
    # Shard-deterministic init: full-matrix init then slice, so TP=k
    # matches TP=1 numerics.
    self.CreateVariable('w_full_seeded', py_utils.WeightParams(
        [p.input_dim, p.output_dim], p.params_init, p.dtype))
    with torch.no_grad():
      shard = self.w_full_seeded[:, rank * self._shard:
                                 (rank + 1) * self._shard].clone()
    del self._parameters['w_full_seeded']
    self.register_parameter('w', torch.nn.Parameter(shard))
    if p.has_bias:
      self.CreateVariable('b', py_utils.WeightParams(
          [self._shard], py_utils.WeightInit.Constant(0.0), p.dtype))

  def FProp(self, theta: NestedMap, x: torch.Tensor) -> torch.Tensor:
    p = self.p
    x = _CopyToTp.apply(x, p.tp_group)
    return py_utils.MatmulBias(x, theta.w,
                               theta.b if p.has_bias else None)


class RowParallelLinear(BaseLayer):
  """Y = all_reduce(X_local @ W[shard, :]) (+ b); input arrives sharded."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('input_dim', 0, 'FULL input dim (sharded over TP).')
    p.Define('output_dim', 0, 'Output dim.')
    p.Define('has_bias', True, 'Bias (added once, after the reduce).')
    p.Define('tp_group', None, 'TP process group.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    world, rank = _TpInfo(p.tp_group)
    assert p.input_dim % world == 0
    self._shard = p.input_dim // world
    self.CreateVariable('w_full_seeded', py_utils.WeightParams(
        [p.input_dim, p.output_dim], p.params_init, p.dtype))
    with torch.no_grad():
      shard = self.w_full_seeded[rank * self._shard:
                                 (rank + 1) * self._shard].clone()
    del self._parameters['w_full_seeded']
    self.register_parameter('w', torch.nn.Parameter(shard))
    if p.has_bias:
      self.CreateVariable('b', py_utils.WeightParams(
          [p.output_dim], py_utils.WeightInit.Constant(0.0), p.dtype))

  def FProp(self, theta: NestedMap, x: torch.Tensor) -> torch.Tensor:
    p = self.p
    out = torch.matmul(x, theta.w)
    out = _ReduceFromTp.apply(out, p.tp_group)
    if p.has_bias:
      out = out + theta.b
    return out


class TpFeedForwardLayer(BaseLayer):
  """Pre-LN FFN with column-parallel W1 and row-parallel W2 — the
  explicit lowering of wi:[M,H]->(s0,s1) / wo:[H,M] sharding
  (gshard_builder DenseBuilder). Drop-in for TransformerFeedForwardLayer.
  """

  @classmethod
  def Params(cls):
    from lingvo_amd.layers import layers as lingvo_layers
    p = super().Params()
    p.Define('input_dim', 0, 'Model dim.')
    p.Define('hidden_dim', 0, 'FFN hidden dim (sharded over TP).')
    p.Define('activation', 'RELU', 'Activation.')
    p.Define('residual_dropout_prob', 0.0, 'Residual dropout.')
    p.Define('relu_dropout_prob', 0.0, 'Hidden dropout.')
    p.Define('residual_weight', 1.0, 'Residual scale.')
    p.Define('tp_group', None, 'TP group.')
    p.Define('ln_tpl', lingvo_layers.LayerNorm.Params(), 'LN template.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    self.CreateChild('layer_norm', p.ln_tpl.Copy().Set(
        input_dim=p.input_dim))
    self.CreateChild('wi', ColumnParallelLinear.Params().Set(
        input_dim=p.input_dim, output_dim=p.hidden_dim,
        tp_group=p.tp_group))
    self.CreateChild('wo', RowParallelLinear.Params().Set(
        input_dim=p.hidden_dim, output_dim=p.input_dim,
        tp_group=p.tp_group))

  def FProp(self, theta: NestedMap, inputs: torch.Tensor,
            paddings=None) -> torch.Tensor:
    from lingvo_amd.layers import activations
    p = self.p
    x = self.layer_norm.FProp(theta.layer_norm, inputs)
    h = activations.GetFn(p.activation)(self.wi.FProp(theta.wi, x))
    if p.relu_dropout_prob and not self.do_eval:
      h = py_utils.DeterministicDropout(h, 1.0 - p.relu_dropout_prob)
    out = self.wo.FProp(theta.wo, h)
    if p.residual_weight != 1.0:
      out = out * p.residual_weight
    if paddings is not None:
      out = py_utils.ApplyPadding(paddings, out)
    if p.residual_dropout_prob and not self.do_eval:
      return py_utils.DeterministicDropoutAdd(
          out, 1.0 - p.residual_dropout_prob, inputs)
    return inputs + out


def ShardTransformerStackForTp(stack_params, tp_group=None):
  """Planner: rewrites a StackedTransformerLayers params tree to use the
  TP FFN (the lowering of weight_split_dims_mapping annotations).
  Attention TP (column-parallel QKV / row-parallel post) requires
  head-sharding inside the flash kernel wrapper and lands in a later
  round; FFN is ~2/3 of transformer weights."""
  tpl = stack_params.transformer_tpl
  ff = tpl.tr_fflayer_tpl
  new_ff = TpFeedForwardLayer.Params().Set(
      input_dim=ff.input_dim, hidden_dim=ff.hidden_dim,
      activation=ff.activation,
      residual_dropout_prob=ff.residual_dropout_prob,
      relu_dropout_prob=ff.relu_dropout_prob,
      residual_weight=ff.residual_weight, tp_group=tp_group)
  tpl.tr_fflayer_tpl = new_ff
  return stack_params


class TpMultiHeadedAttention(BaseLayer):
  """Head-sharded self-attention: each TP rank owns N/W query heads
  (and their KV heads), runs the flash kernel on its local heads, and
  the row-parallel output projection all-reduces the result — the
  explicit lowering of the reference's attention sharding
  `wq:[M,N,H] -> (M,N)` over the mesh (gshard_builder.py DenseBuilder;
  tasks/lm/README.md:100-115). Drop-in for MultiHeadedAttention.FProp.
  """

  @classmethod
  def Params(cls):
    from lingvo_amd.layers.attention import MultiHeadedAttention
    p = MultiHeadedAttention.Params()
    p.cls = cls
    p.Define('tp_group', None, 'TP process group.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    world, rank = _TpInfo(p.tp_group)
    n = p.num_heads
    assert n % world == 0, 'num_heads must divide TP world size'
    nkv_full = p.num_kv_heads or n
    assert nkv_full % world == 0
    h = p.dim_per_head or (p.hidden_dim or p.input_dim) // n
    self._n = n // world
    self._nkv = nkv_full // world
    self._h = h
    self._world, self._rank = world, rank
    d = p.input_dim
    # Full-matrix seeded init, sliced by head block, so TP=k matches
    # TP=1 numerics exactly.
    self.CreateVariable('qkv_w_full_seeded', py_utils.WeightParams(
        [d, (n + 2 * nkv_full) * h], p.params_init, p.dtype))
    with torch.no_grad():
      w = self.qkv_w_full_seeded
      q_w = w[:, :n * h].reshape(d, world, self._n * h)[:, rank]
      k_w = w[:, n * h:(n + nkv_full) * h].reshape(
          d, world, self._nkv * h)[:, rank]
      v_w = w[:, (n + nkv_full) * h:].reshape(
          d, world, self._nkv * h)[:, rank]
      shard = torch.cat([q_w, k_w, v_w], dim=1).clone()
    del self._parameters['qkv_w_full_seeded']
    self.register_parameter('qkv_w', torch.nn.Parameter(shard))
    if p.use_bias:
      self.CreateVariable('qkv_b', py_utils.WeightParams(
          [(self._n + 2 * self._nkv) * h],
          py_utils.WeightInit.Constant(0.0), p.dtype))
    self.CreateVariable('post_w_full_seeded', py_utils.WeightParams(
        [n * h, d], p.params_init, p.dtype))
    with torch.no_grad():
      pw = self.post_w_full_seeded.reshape(world, self._n * h, d)[rank]
      pshard = pw.clone()
    del self._parameters['post_w_full_seeded']
    self.register_parameter('post_w', torch.nn.Parameter(pshard))
    if p.use_bias:
      # Bias added once after the all-reduce (owned by every rank but
      # scaled so the sum is applied once).
      self.CreateVariable('post_b', py_utils.WeightParams(
          [d], py_utils.WeightInit.Constant(0.0), p.dtype))
    if p.rel_pos_bias:
      self.CreateVariable('rel_bias_full_seeded', py_utils.WeightParams(
          [n, 2 * p.rel_pos_clip + 1], py_utils.WeightInit.Constant(0.0),
          p.dtype))
      with torch.no_grad():
        rb = self.rel_bias_full_seeded.reshape(
            world, self._n, -1)[rank].clone()
      del self._parameters['rel_bias_full_seeded']
      self.register_parameter('rel_bias', torch.nn.Parameter(rb))

  def FProp(self, theta: NestedMap, query_vec: torch.Tensor,
            paddings=None, segment_ids=None) -> torch.Tensor:
    from lingvo_amd.ops import flash_attn
    p = self.p
    n, nkv, h = self._n, self._nkv, self._h
    b, t = query_vec.shape[0], query_vec.shape[1]
    x = _CopyToTp.apply(query_vec, p.tp_group)
    qkv = py_utils.MatmulBias(x, theta.qkv_w,
                              theta.qkv_b if p.use_bias else None)
    q, k, v = qkv.split([n * h, nkv * h, nkv * h], dim=-1)
    q = q.reshape(b, t, n, h)
    k = k.reshape(b, t, nkv, h)
    v = v.reshape(b, t, nkv, h)
    klen = None
    if paddings is not None and segment_ids is None:
      klen = py_utils.LengthsFromPaddings(paddings).to(torch.int32)
    bias = theta.rel_bias if p.rel_pos_bias else None
    win_r = 0 if p.causal else p.right_context
    out = flash_attn.flash_attention(
        q, k, v, klen, bias, p.left_context, win_r, p.rel_pos_clip,
        q_segment_ids=segment_ids, k_segment_ids=segment_ids)
    ctx = out.reshape(b, t, n * h)
    post = torch.matmul(ctx, theta.post_w)
    post = _ReduceFromTp.apply(post, p.tp_group)
    if p.use_bias:
      post = post + theta.post_b
    if paddings is not None:
      post = py_utils.ApplyPadding(paddings, post)
    return post


def ShardAttentionForTp(stack_params, tp_group=None):
  """Planner: swaps the stack's attention template for the TP variant
  (composes with ShardTransformerStackForTp for the FFN)."""
  atten = stack_params.transformer_tpl.tr_atten_tpl.atten_tpl
  new_p = TpMultiHeadedAttention.Params()
  for name, _ in new_p.IterParams():
    if name in ('cls', 'tp_group'):
      continue
    if name in atten:
      setattr(new_p, name, atten.Get(name))
  new_p.tp_group = tp_group
  stack_params.transformer_tpl.tr_atten_tpl.atten_tpl = new_p
  return stack_params


def LowerShardingAnnotations(params, tp_group=None):
  """Annotation-driven planner (the explicit-RCCL lowering of GShard's
  device_mesh / weight_split_dims_mapping annotations, reference
  gshard_utils.py:39-137 + base_layer.py:262-280).

  Recursively walks a Params tree; any StackedTransformerLayers whose
  transformer_tpl carries weight_split_dims_mapping (or whose sub
  templates do) is rewritten to the explicit TP FFN + TP attention
  classes. Model code keeps the same annotation surface as the
  reference; this pass inserts the collectives."""
  from lingvo_amd.core.hyperparams import Params as _Params
  from lingvo_amd.layers import transformer as transformer_lib

  def annotated(p):
    try:
      return (p.Get('weight_split_dims_mapping') is not None or
              p.Get('device_mesh') is not None)
    except Exception:
      return False

  def visit(p):
    if not isinstance(p, _Params):
      return
    cls = p.Get('cls') if 'cls' in p else None
    if cls is transformer_lib.StackedTransformerLayers:
      tpl = p.transformer_tpl
      if annotated(p) or annotated(tpl) or \
          annotated(tpl.tr_fflayer_tpl) or \
          annotated(tpl.tr_atten_tpl.atten_tpl):
        ShardTransformerStackForTp(p, tp_group)
        ShardAttentionForTp(p, tp_group)
      return
    for name, val in p.IterParams():
      if isinstance(val, _Params):
        visit(val)
      elif isinstance(val, (list, tuple)):
        for v in val:
          visit(v) if isinstance(v, _Params) else None

  visit(params)
  return params
