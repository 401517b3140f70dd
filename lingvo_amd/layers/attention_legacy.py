"""Per-step attention classes for RNN decoders.

MI355X-native re-implementation of the reference's seq-major attention
family (lingvo/core/attention.py: AdditiveAttention:547,
DotProductAttention:1015, LocationSensitiveAttention:2334,
MonotonicAttention:2900, GmmMonotonicAttention:3267, MergerLayer:3608).
Batch-major here ([B, S, D] sources, [B, Q] queries): these run once per
decode step inside RNN attention decoders, so they are GEMV/bmm shaped —
composed torch ops (hipBLASLt) are the right backend, not custom
kernels.

Contract (mirrors the reference's three-phase API):
  packed = atten.InitForSourcePacked(theta, src_vecs, src_ctxs, padding)
  state  = atten.ZeroAttentionState(src_len, batch, device, dtype)
  ctx, probs, state = atten.ComputeContextVector(theta, packed, query,
                                                 state)
"""

from __future__ import annotations

import math
from typing import Optional, Tuple

import torch
import torch.nn.functional as F

from lingvo_amd.core import py_utils
from lingvo_amd.core.base_layer import BaseLayer
from lingvo_amd.core.nested_map import NestedMap


class BaseAttentionLayer(BaseLayer):
  """Shared packing + masking for per-step attention."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('source_dim', 0, 'Source vector dim.')
    p.Define('query_dim', 0, 'Query vector dim.')
    p.Define('hidden_dim', 0, 'Attention hidden dim.')
    p.Define('atten_dropout_prob', 0.0, 'Prob dropout.')
    return p

  def InitForSourcePacked(self, theta: NestedMap, source_vecs: torch.Tensor,
                          source_contexts: Optional[torch.Tensor],
                          source_padding: torch.Tensor) -> NestedMap:
    """source_vecs [B,S,D] keys; source_contexts [B,S,Dc] values
    (defaults to the keys); source_padding [B,S]."""
    return NestedMap(
        source_vecs=source_vecs,
        source_contexts=(source_contexts if source_contexts is not None
                         else source_vecs),
        source_padding=source_padding)

  def ZeroAttentionState(self, source_len: int, batch: int, device=None,
                         dtype=torch.float32) -> NestedMap:
    return NestedMap()

  def _Finalize(self, logits: torch.Tensor, packed: NestedMap
                ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Mask padded positions, softmax, weighted-sum the contexts."""
    logits = logits.float().masked_fill(packed.source_padding > 0.5, -1e30)
    probs = torch.softmax(logits, dim=-1)
    if self.p.atten_dropout_prob and not self.do_eval:
      probs = py_utils.DeterministicDropout(
          probs, 1.0 - self.p.atten_dropout_prob)
    ctx = torch.bmm(probs.unsqueeze(1).to(packed.source_contexts.dtype),
                    packed.source_contexts).squeeze(1)
    return ctx, probs

  def ComputeContextVector(self, theta, packed, query_vec, state):
    raise NotImplementedError


class AdditiveAttention(BaseAttentionLayer):
  """v . tanh(W_s s + W_q q) (reference attention.py:547)."""

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    self.CreateVariable('source_var', py_utils.WeightParams(
        [p.source_dim, p.hidden_dim], p.params_init, p.dtype))
    self.CreateVariable('query_var', py_utils.WeightParams(
        [p.query_dim, p.hidden_dim], p.params_init, p.dtype))
    self.CreateVariable('hidden_var', py_utils.WeightParams(
        [p.hidden_dim], p.params_init, p.dtype))

  def InitForSourcePacked(self, theta, source_vecs, source_contexts,
                          source_padding):
    packed = super().InitForSourcePacked(theta, source_vecs,
                                         source_contexts, source_padding)
    # precompute W_s s once per utterance
    packed.projected = torch.matmul(source_vecs, theta.source_var)
    return packed

  def _Logits(self, theta, packed, query_vec):
    q = torch.matmul(query_vec, theta.query_var).unsqueeze(1)  # [B,1,H]
    return torch.einsum('bsh,h->bs', torch.tanh(packed.projected + q),
                        theta.hidden_var.float().to(q.dtype))

  def ComputeContextVector(self, theta, packed, query_vec, state):
    ctx, probs = self._Finalize(self._Logits(theta, packed, query_vec),
                                packed)
    return ctx, probs, state


class DotProductAttention(BaseAttentionLayer):
  """Scaled dot-product (reference attention.py:1015)."""

  def ComputeContextVector(self, theta, packed, query_vec, state):
    scale = 1.0 / math.sqrt(max(1, query_vec.shape[-1]))
    logits = torch.einsum('bsd,bd->bs', packed.source_vecs.float(),
                          query_vec.float()) * scale
    ctx, probs = self._Finalize(logits, packed)
    return ctx, probs, state


class LocationSensitiveAttention(AdditiveAttention):
  """Additive attention + convolved previous alignments (reference
  attention.py:2334; Chorowski et al. 2015). The previous step's probs
  are convolved with learned location filters and added to the energy,
  encouraging monotonic movement for ASR."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('location_filter_size', 5, 'Conv filter width (odd).')
    p.Define('location_num_filters', 8, 'Number of location filters.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    self.CreateVariable('location_filter_var', py_utils.WeightParams(
        [p.location_num_filters, 1, p.location_filter_size],
        p.params_init, p.dtype))
    self.CreateVariable('location_var', py_utils.WeightParams(
        [p.location_num_filters, p.hidden_dim], p.params_init, p.dtype))

  def ZeroAttentionState(self, source_len, batch, device=None,
                         dtype=torch.float32):
    # initial alignment: all mass on frame 0 (reference behavior)
    probs = torch.zeros(batch, source_len, device=device, dtype=dtype)
    probs[:, 0] = 1.0
    return NestedMap(atten_probs=probs)

  def ComputeContextVector(self, theta, packed, query_vec, state):
    p = self.p
    prev = state.atten_probs.unsqueeze(1)  # [B,1,S]
    loc = F.conv1d(prev.to(theta.location_filter_var.dtype),
                   theta.location_filter_var,
                   padding=p.location_filter_size // 2)  # [B,F,S]
    loc_term = torch.einsum('bfs,fh->bsh', loc, theta.location_var)
    q = torch.matmul(query_vec, theta.query_var).unsqueeze(1)
    energy = torch.einsum(
        'bsh,h->bs', torch.tanh(packed.projected + q + loc_term),
        theta.hidden_var.to(q.dtype))
    ctx, probs = self._Finalize(energy, packed)
    return ctx, probs, NestedMap(atten_probs=probs.to(state.atten_probs.dtype))


class MonotonicAttention(AdditiveAttention):
  """Soft monotonic alignment (reference attention.py:2900; Raffel et
  al. 2017). Training uses the parallel expected-alignment recurrence
    alpha_j = p_j * cp_j * cumsum_k<=j(alpha_prev_k / cp_k),
  cp = exclusive cumprod(1-p); eval uses the same soft form (the
  reference's hard mode is a decode-time option)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('pre_sigmoid_noise', 0.0, 'Training noise std on energy.')
    p.Define('hidden_bias_init', -1.0, 'Initial energy bias.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    self.CreateVariable('energy_bias_var', py_utils.WeightParams(
        [1], py_utils.WeightInit.Constant(p.hidden_bias_init), p.dtype))

  def ZeroAttentionState(self, source_len, batch, device=None,
                         dtype=torch.float32):
    alpha = torch.zeros(batch, source_len, device=device, dtype=dtype)
    alpha[:, 0] = 1.0  # alignment starts at the first frame
    return NestedMap(emit_probs=alpha)

  def ComputeContextVector(self, theta, packed, query_vec, state):
    p = self.p
    energy = self._Logits(theta, packed, query_vec) + \
        theta.energy_bias_var.float()
    if p.pre_sigmoid_noise and not self.do_eval:
      # approximate N(0,1) via sum of uniforms (graph-safe RNG)
      u = sum(py_utils.GraphSafeUniform(energy.shape, energy.device)
              for _ in range(4)) - 2.0
      energy = energy + p.pre_sigmoid_noise * u * math.sqrt(3.0)
    energy = energy.masked_fill(packed.source_padding > 0.5, -1e30)
    pchoose = torch.sigmoid(energy)
    cp = torch.cumprod(
        torch.cat([torch.ones_like(pchoose[:, :1]),
                   (1 - pchoose[:, :-1]).clamp_min(1e-10)], dim=1), dim=1)
    alpha = pchoose * cp * torch.cumsum(
        state.emit_probs.float() / cp.clamp_min(1e-10), dim=1)
    ctx = torch.bmm(alpha.unsqueeze(1).to(packed.source_contexts.dtype),
                    packed.source_contexts).squeeze(1)
    return ctx, alpha, NestedMap(emit_probs=alpha.to(state.emit_probs.dtype))


class GmmMonotonicAttention(BaseAttentionLayer):
  """GMM attention with forward-moving means (reference
  attention.py:3267; Graves 2013 window attention). An MLP on the query
  emits per-mixture (weight, delta>=0, scale); the means advance by
  delta each step, and probs are mixture densities over positions."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('num_mixtures', 5, 'GMM mixtures K.')
    p.Define('gmm_mlp_hidden_dim', 64, 'MLP hidden size.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    self.CreateVariable('mlp_w1', py_utils.WeightParams(
        [p.query_dim, p.gmm_mlp_hidden_dim], p.params_init, p.dtype))
    self.CreateVariable('mlp_b1', py_utils.WeightParams(
        [p.gmm_mlp_hidden_dim], py_utils.WeightInit.Constant(0.0), p.dtype))
    self.CreateVariable('mlp_w2', py_utils.WeightParams(
        [p.gmm_mlp_hidden_dim, 3 * p.num_mixtures], p.params_init, p.dtype))
    self.CreateVariable('mlp_b2', py_utils.WeightParams(
        [3 * p.num_mixtures], py_utils.WeightInit.Constant(0.0), p.dtype))

  def ZeroAttentionState(self, source_len, batch, device=None,
                         dtype=torch.float32):
    return NestedMap(position=torch.zeros(
        batch, self.p.num_mixtures, device=device, dtype=dtype))

  def ComputeContextVector(self, theta, packed, query_vec, state):
    p = self.p
    h = torch.tanh(py_utils.MatmulBias(query_vec, theta.mlp_w1,
                                       theta.mlp_b1))
    out = py_utils.MatmulBias(h, theta.mlp_w2, theta.mlp_b2).float()
    w, delta, scale = out.chunk(3, dim=-1)  # each [B,K]
    w = torch.softmax(w, dim=-1)
    mu = state.position.float() + F.softplus(delta)
    var = F.softplus(scale) + 1e-4
    s = packed.source_vecs.shape[1]
    pos = torch.arange(s, device=query_vec.device, dtype=torch.float32)
    # [B,K,S] gaussian densities
    dens = torch.exp(-0.5 * (pos[None, None] - mu[:, :, None]) ** 2
                     / var[:, :, None]) / torch.sqrt(
                         2 * math.pi * var[:, :, None])
    probs = torch.einsum('bk,bks->bs', w, dens)
    probs = probs.masked_fill(packed.source_padding > 0.5, 0.0)
    probs = probs / probs.sum(-1, keepdim=True).clamp_min(1e-8)
    ctx = torch.bmm(probs.unsqueeze(1).to(packed.source_contexts.dtype),
                    packed.source_contexts).squeeze(1)
    return ctx, probs, NestedMap(position=mu.to(state.position.dtype))


class MergerLayer(BaseLayer):
  """Merge a list of equally-shaped context vectors (reference
  attention.py:3608). merger_op: mean | sum | concat | weighted_sum |
  gated_avg."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('merger_op', 'mean', 'mean|sum|concat|weighted_sum|gated_avg')
    p.Define('num_sources', 0, 'Number of inputs (for weighted/gated).')
    p.Define('source_dim', 0, 'Per-source dim (for gated_avg).')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    if p.merger_op == 'weighted_sum':
      self.CreateVariable('sum_weight', py_utils.WeightParams(
          [p.num_sources], py_utils.WeightInit.Constant(0.0), p.dtype))
    elif p.merger_op == 'gated_avg':
      self.CreateVariable('gate_w', py_utils.WeightParams(
          [p.num_sources * p.source_dim, p.num_sources],
          p.params_init, p.dtype))

  def FProp(self, theta: NestedMap, inputs) -> torch.Tensor:
    p = self.p
    x = torch.stack(list(inputs), dim=-2)  # [..., N, D]
    if p.merger_op == 'mean':
      return x.mean(dim=-2)
    if p.merger_op == 'sum':
      return x.sum(dim=-2)
    if p.merger_op == 'concat':
      return torch.cat(list(inputs), dim=-1)
    if p.merger_op == 'weighted_sum':
      w = torch.softmax(theta.sum_weight.float(), dim=0).to(x.dtype)
      return torch.einsum('...nd,n->...d', x, w)
    if p.merger_op == 'gated_avg':
      flat = torch.cat(list(inputs), dim=-1)
      gates = torch.softmax(torch.matmul(flat, theta.gate_w), dim=-1)
      return torch.einsum('...nd,...n->...d', x, gates.to(x.dtype))
    raise ValueError(p.merger_op)


class MultiHeadedAttention(BaseAttentionLayer):
  """Per-step multi-headed wrapper around an inner attention
  (reference attention.py:1425): projects sources/queries/contexts per
  head, folds heads into the batch, runs `inner_atten` per head, and
  combines with an output projection."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('num_attention_heads', 4, 'Heads.')
    p.Define('inner_atten_tpl', DotProductAttention.Params(),
             'Per-head attention template.')
    p.Define('context_dim', 0, 'Value dim (defaults to source_dim).')
    p.Define('use_source_vec_as_attention_value', True,
             'Use keys as values (reference default).')
    p.Define('enable_query_proj', True, 'Project queries.')
    p.Define('enable_ctx_post_proj', True, 'Output projection.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    n = p.num_attention_heads
    assert p.hidden_dim % n == 0, 'hidden_dim % heads'
    ctx_dim = p.context_dim or p.source_dim
    self.CreateVariable('source_proj', py_utils.WeightParams(
        [p.source_dim, p.hidden_dim], p.params_init, p.dtype))
    if p.enable_query_proj:
      self.CreateVariable('query_proj', py_utils.WeightParams(
          [p.query_dim, p.hidden_dim], p.params_init, p.dtype))
    if not p.use_source_vec_as_attention_value:
      self.CreateVariable('ctx_proj', py_utils.WeightParams(
          [ctx_dim, p.hidden_dim], p.params_init, p.dtype))
    if p.enable_ctx_post_proj:
      self.CreateVariable('ctx_post_proj', py_utils.WeightParams(
          [p.hidden_dim, p.hidden_dim], p.params_init, p.dtype))
    inner = p.inner_atten_tpl.Copy().Set(
        name='inner', source_dim=p.hidden_dim // n,
        query_dim=p.hidden_dim // n, hidden_dim=p.hidden_dim // n,
        atten_dropout_prob=p.atten_dropout_prob)
    self.CreateChild('inner', inner)

  def InitForSourcePacked(self, theta, source_vecs, source_contexts,
                          source_padding):
    p = self.p
    n = p.num_attention_heads
    b, s, _ = source_vecs.shape
    h = p.hidden_dim // n
    keys = torch.matmul(source_vecs, theta.source_proj)
    if p.use_source_vec_as_attention_value or source_contexts is None:
      vals = keys
    else:
      vals = torch.matmul(source_contexts, theta.ctx_proj)
    # fold heads into batch: [B*N, S, H]
    keys = keys.reshape(b, s, n, h).permute(0, 2, 1, 3).reshape(
        b * n, s, h)
    vals = vals.reshape(b, s, n, h).permute(0, 2, 1, 3).reshape(
        b * n, s, h)
    pad = source_padding.repeat_interleave(n, dim=0)
    packed = self.inner.InitForSourcePacked(theta.inner, keys, vals, pad)
    packed.batch = b
    return packed

  def ZeroAttentionState(self, source_len, batch, device=None,
                         dtype=torch.float32):
    return self.inner.ZeroAttentionState(
        source_len, batch * self.p.num_attention_heads, device, dtype)

  def ComputeContextVector(self, theta, packed, query_vec, state):
    p = self.p
    n = p.num_attention_heads
    b = packed.batch
    h = p.hidden_dim // n
    q = query_vec
    if p.enable_query_proj:
      q = torch.matmul(q, theta.query_proj)
    q = q.reshape(b, n, h).reshape(b * n, h)
    ctx, probs, state = self.inner.ComputeContextVector(
        theta.inner, packed, q, state)
    ctx = ctx.reshape(b, n * h)
    if p.enable_ctx_post_proj:
      ctx = torch.matmul(ctx, theta.ctx_post_proj)
    # head-averaged probs (reference returns per-head; average for the
    # [B, S] contract used by decoder callbacks)
    probs = probs.reshape(b, n, -1).mean(dim=1)
    return ctx, probs, state


class MultiSourceAttention(BaseAttentionLayer):
  """Attention over multiple named source sets, merged
  (reference attention.py:3856): one child attention per source,
  combined by a MergerLayer."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('source_atten_tpls', [],
             'List of (name, attention Params).')
    p.Define('primary_source_key', '',
             'Source whose probs are returned (default: first).')
    p.Define('atten_merger_tpl', None,
             'MergerLayer params (default: mean).')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    self._names = [name for name, _ in p.source_atten_tpls]
    for name, tpl in p.source_atten_tpls:
      self.CreateChild(f'atten_{name}', tpl.Copy())
    if p.atten_merger_tpl is not None:
      self.CreateChild('merger', p.atten_merger_tpl.Copy())

  def InitForSourcePacked(self, theta, source_vecs, source_contexts,
                          source_padding):
    """source_vecs/contexts/padding: NestedMaps keyed by source name."""
    packed = NestedMap()
    for name in self._names:
      child = getattr(self, f'atten_{name}')
      packed[name] = child.InitForSourcePacked(
          theta[f'atten_{name}'], source_vecs[name],
          None if source_contexts is None else source_contexts[name],
          source_padding[name])
    return packed

  def ZeroAttentionState(self, source_len, batch, device=None,
                         dtype=torch.float32):
    return NestedMap()

  def ComputeContextVector(self, theta, packed, query_vec, state):
    p = self.p
    ctxs = []
    primary_probs = None
    primary = p.primary_source_key or self._names[0]
    for name in self._names:
      child = getattr(self, f'atten_{name}')
      ctx, probs, _ = child.ComputeContextVector(
          theta[f'atten_{name}'], packed[name], query_vec, NestedMap())
      ctxs.append(ctx)
      if name == primary:
        primary_probs = probs
    if p.atten_merger_tpl is not None:
      merged = self.merger.FProp(theta.merger, ctxs)
    else:
      merged = torch.stack(ctxs).mean(dim=0)
    return merged, primary_probs, state
