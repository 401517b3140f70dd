"""Batch-major multi-headed attention (reference
lingvo/core/batch_major_attention.py:481 MultiHeadedAttention).

QKV projections run as one fused [D, (2+G)NH] GEMM on hipBLASLt
(reference precedent `enable_qkv_proj_in_onestep`,
batch_major_attention.py:568-582); the attention core is the hand-written
gfx950 flash kernel with causal/local/padding masks, GQA
(num_kv_heads, reference :542) and clipped relative-position bias.
Decode path ExtendStep (reference :1617) keeps an in-place KV cache.
"""

from __future__ import annotations

import math
from typing import Optional, Tuple

import torch

from lingvo_amd.core import py_utils
from lingvo_amd.core.base_layer import BaseLayer
from lingvo_amd.core.nested_map import NestedMap
from lingvo_amd.ops import flash_attn


class MultiHeadedAttention(BaseLayer):

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('input_dim', 0, 'Query (and default key/value) dim.')
    p.Define('hidden_dim', 0, 'Total projected dim (N * H).')
    p.Define('num_heads', 1, 'Query heads N.')
    p.Define('num_kv_heads', 0, 'KV heads for GQA; 0 = num_heads.')
    p.Define('dim_per_head', 0, 'H; 0 derives hidden_dim/num_heads.')
    p.Define('atten_dropout_prob', 0.0, 'Attention prob dropout.')
    p.Define('use_bias', True, 'Bias on projections.')
    p.Define('enable_scaling_code_motion', True,
             'Scale logits in-kernel (always true here).')
    p.Define('rel_pos_bias', False,
             'Learned clipped relative-position bias per head.')
    p.Define('rel_pos_clip', 127, 'Max relative distance for the bias.')
    p.Define('left_context', -1, 'Local attention left window; -1 = inf.')
    p.Define('right_context', -1, 'Local attention right window; -1 = inf.')
    p.Define('causal', False, 'Causal (self-attention) masking.')
    p.Define('use_rope', False,
             'Rotary position embedding on Q/K (reference '
             'layers.py:3476 + batch_major_attention rope option).')
    p.Define('rope_tpl', None,
             'Optional RotaryPositionalEmbeddingLayer params override.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    n = p.num_heads
    nkv = p.num_kv_heads or n
    h = p.dim_per_head or (p.hidden_dim // n)
    self._n, self._nkv, self._h = n, nkv, h
    # The gfx950 flash kernel supports dim_per_head 64/128 (checked at
    # the kernel boundary); other sizes run only on the CPU reference.
    # Fused QKV for self-attention: [D, (N + 2*NKV) * H].
    self.CreateVariable('qkv_w', py_utils.WeightParams(
        [p.input_dim, (n + 2 * nkv) * h], p.params_init, p.dtype))
    if p.use_bias:
      self.CreateVariable('qkv_b', py_utils.WeightParams(
          [(n + 2 * nkv) * h], py_utils.WeightInit.Constant(0.0), p.dtype))
    self.CreateVariable('post_w', py_utils.WeightParams(
        [n * h, p.input_dim], p.params_init, p.dtype))
    if p.use_bias:
      self.CreateVariable('post_b', py_utils.WeightParams(
          [p.input_dim], py_utils.WeightInit.Constant(0.0), p.dtype))
    if p.rel_pos_bias:
      self.CreateVariable('rel_bias', py_utils.WeightParams(
          [n, 2 * p.rel_pos_clip + 1],
          py_utils.WeightInit.Constant(0.0), p.dtype))
    if p.use_rope:
      from lingvo_amd.layers import layers as lingvo_layers
      rope_p = (p.rope_tpl or
                lingvo_layers.RotaryPositionalEmbeddingLayer.Params())
      self.CreateChild('rope', rope_p.Copy().Set(embedding_dim=h))

  def _Project(self, theta: NestedMap, x: torch.Tensor):
    p = self.p
    n, nkv, h = self._n, self._nkv, self._h
    b, t = x.shape[0], x.shape[1]
    qkv = py_utils.MatmulBias(x, theta.qkv_w,
                              theta.qkv_b if p.use_bias else None)
    q, k, v = qkv.split([n * h, nkv * h, nkv * h], dim=-1)
    return (q.reshape(b, t, n, h), k.reshape(b, t, nkv, h),
            v.reshape(b, t, nkv, h))

  def FProp(self, theta: NestedMap, query_vec: torch.Tensor,
            paddings: Optional[torch.Tensor] = None,
            segment_ids: Optional[torch.Tensor] = None) -> torch.Tensor:
    """Self-attention over [B, T, D] with [B, T] paddings. Optional
    packed-input segment_ids [B, T] block cross-segment attention
    (reference packed-input segment mask, batch_major_attention.py)."""
    p = self.p
    q, k, v = self._Project(theta, query_vec)
    if p.use_rope:
      q = self.rope.FProp(theta.rope, q)
      k = self.rope.FProp(theta.rope, k)
    klen = None
    if paddings is not None and segment_ids is None:
      klen = py_utils.LengthsFromPaddings(paddings).to(torch.int32)
    bias = theta.rel_bias if p.rel_pos_bias else None
    win_r = 0 if p.causal else p.right_context
    out = flash_attn.flash_attention(
        q, k, v, klen, bias, p.left_context, win_r, p.rel_pos_clip,
        q_segment_ids=segment_ids, k_segment_ids=segment_ids)
    if p.atten_dropout_prob and not self.do_eval:
      # Dropout on the context vectors (prob-dropout approximation; the
      # reference drops attention probs, batch_major_attention.py:1045).
      out = py_utils.DeterministicDropout(out, 1.0 - p.atten_dropout_prob)
    b, t = out.shape[0], out.shape[1]
    ctx = out.reshape(b, t, self._n * self._h)
    post = py_utils.MatmulBias(ctx, theta.post_w,
                               theta.post_b if p.use_bias else None)
    if paddings is not None:
      post = py_utils.ApplyPadding(paddings, post)
    return post

  def _NoRope(self):
    if self.p.use_rope:
      raise NotImplementedError('RoPE is self-attention only')

  def FPropCross(self, theta: NestedMap, query_vec: torch.Tensor,
                 key_vec: torch.Tensor, value_vec: torch.Tensor,
                 source_paddings: Optional[torch.Tensor] = None
                 ) -> torch.Tensor:
    """Cross-attention: query [B,T,D], key/value source [B,S,D]."""
    self._NoRope()
    p = self.p
    n, nkv, h = self._n, self._nkv, self._h
    b, t = query_vec.shape[0], query_vec.shape[1]
    s = key_vec.shape[1]
    qkv_w = theta.qkv_w
    wq = qkv_w[:, :n * h]
    wk = qkv_w[:, n * h:(n + nkv) * h]
    wv = qkv_w[:, (n + nkv) * h:]
    q = torch.matmul(query_vec, wq).reshape(b, t, n, h)
    k = torch.matmul(key_vec, wk).reshape(b, s, nkv, h)
    v = torch.matmul(value_vec, wv).reshape(b, s, nkv, h)
    if p.use_bias:
      qb, kb, vb = theta.qkv_b.split([n * h, nkv * h, nkv * h])
      q = q + qb.reshape(n, h)
      k = k + kb.reshape(nkv, h)
      v = v + vb.reshape(nkv, h)
    klen = None
    if source_paddings is not None:
      klen = py_utils.LengthsFromPaddings(source_paddings).to(torch.int32)
    out = flash_attn.flash_attention(q, k, v, klen, None, -1, -1)
    ctx = out.reshape(b, t, n * h)
    post = torch.matmul(ctx, theta.post_w)
    if p.use_bias:
      post = post + theta.post_b
    return post

  # ---- incremental decoding (reference InitStates/ExtendStep) -----------
  def InitStates(self, theta: NestedMap, batch: int, max_len: int,
                 device, dtype=torch.bfloat16) -> NestedMap:
    nkv, h = self._nkv, self._h
    return NestedMap(
        key=torch.zeros(batch, max_len, nkv, h, device=device, dtype=dtype),
        value=torch.zeros(batch, max_len, nkv, h, device=device,
                          dtype=dtype),
        time_step=0)

  def StreamStep(self, theta: NestedMap, x_chunk: torch.Tensor,
                 paddings_chunk: torch.Tensor,
                 state: NestedMap) -> tuple:
    """Chunked streaming self-attention (reference StreamStep,
    batch_major_attention.py / conformer_layer.py:390): attends causally
    within the chunk and over up to `left_context` cached frames."""
    p = self.p
    n, nkv, h = self._n, self._nkv, self._h
    b, c = x_chunk.shape[0], x_chunk.shape[1]
    q, k, v = self._Project(theta, x_chunk)
    t0 = state.time_step
    if p.use_rope:
      pos = (t0 + torch.arange(c, device=x_chunk.device,
                               dtype=torch.float32)).expand(b, c)
      q = self.rope.FProp(theta.rope, q, position=pos)
      k = self.rope.FProp(theta.rope, k, position=pos)
    L = p.left_context if p.left_context >= 0 else state.key.shape[1]
    state.key[:, t0:t0 + c] = k.to(state.key.dtype)
    state.value[:, t0:t0 + c] = v.to(state.value.dtype)
    state.time_step = t0 + c
    lo = max(0, t0 - L)
    keys = state.key[:, lo:t0 + c]
    values = state.value[:, lo:t0 + c]
    group = n // nkv
    kf = keys.float()
    vf = values.float()
    if group > 1:
      kf = kf.repeat_interleave(group, dim=2)
      vf = vf.repeat_interleave(group, dim=2)
    logits = torch.einsum('bcnh,bsnh->bncs', q.float(), kf)
    logits = logits / math.sqrt(h)
    qpos = (t0 + torch.arange(c, device=x_chunk.device))[:, None]
    kpos = (lo + torch.arange(kf.shape[1], device=x_chunk.device))[None, :]
    mask = (kpos <= qpos) & (kpos >= qpos - L)
    if p.rel_pos_bias:
      d = (qpos - kpos).clamp(-p.rel_pos_clip, p.rel_pos_clip)           + p.rel_pos_clip
      logits = logits + theta.rel_bias.float()[None, :, d]
    logits = logits.masked_fill(~mask[None, None], -1e30)
    probs = torch.softmax(logits, dim=-1)
    ctx = torch.einsum('bncs,bsnh->bcnh', probs, vf)
    ctx = ctx.reshape(b, c, n * h).to(x_chunk.dtype)
    post = py_utils.MatmulBias(ctx, theta.post_w,
                               theta.post_b if p.use_bias else None)
    post = py_utils.ApplyPadding(paddings_chunk, post)
    return post, state

  def ExtendStep(self, theta: NestedMap, query_vec: torch.Tensor,
                 cached_states: NestedMap,
                 per_step_padding=None) -> Tuple[torch.Tensor, NestedMap]:
    """query_vec [B, 1, D]; appends to the KV cache and attends."""
    p = self.p
    n, nkv, h = self._n, self._nkv, self._h
    b = query_vec.shape[0]
    q, k, v = self._Project(theta, query_vec)
    t = cached_states.time_step
    if p.use_rope:
      pos = torch.full((b, 1), t, dtype=torch.float32,
                       device=query_vec.device)
      q = self.rope.FProp(theta.rope, q, position=pos)
      k = self.rope.FProp(theta.rope, k, position=pos)
    cached_states.key[:, t:t + 1] = k.to(cached_states.key.dtype)
    cached_states.value[:, t:t + 1] = v.to(cached_states.value.dtype)
    cached_states.time_step = t + 1
    keys = cached_states.key[:, :t + 1]
    values = cached_states.value[:, :t + 1]
    # One-step attention (GEMV-shaped); composed torch ops are fine here.
    group = n // nkv
    qf = q.reshape(b, n, h).float()
    kf = keys.float().repeat_interleave(group, dim=2) if group > 1 \
        else keys.float()
    vf = values.float().repeat_interleave(group, dim=2) if group > 1 \
        else values.float()
    logits = torch.einsum('bnh,bsnh->bns', qf, kf) / math.sqrt(h)
    if p.rel_pos_bias:
      d = (t - torch.arange(t + 1, device=q.device)).clamp(
          -p.rel_pos_clip, p.rel_pos_clip) + p.rel_pos_clip
      logits = logits + theta.rel_bias.float()[:, d]
    probs = torch.softmax(logits, dim=-1)
    ctx = torch.einsum('bns,bsnh->bnh', probs, vf)
    ctx = ctx.reshape(b, 1, n * h).to(query_vec.dtype)
    post = torch.matmul(ctx, theta.post_w)
    if p.use_bias:
      post = post + theta.post_b
    return post, cached_states


class TransformerXLAttention(MultiHeadedAttention):
  """Transformer-XL relative attention with segment memory (reference
  batch_major_attention.py:2233 MultiHeadedAttentionXL +
  attention_util.py:384 PositionalAttenLogits; Dai et al. 2019).

  logits_ij = (q_i + u) . k_j  +  (q_i + v) . r_{d(i,j)}
  with learned per-head biases u, v and sinusoidal relative embeddings
  r projected per head. `memory` (the previous segment's input hidden
  states, no grad) is prepended on the key/value side, giving XL-style
  segment recurrence. This composed-einsum form runs on CPU and GPU but
  materializes the [B, N, T, L] rel-logits; the in-kernel rel-shift
  variant (extra MFMA against a staged R tile) is a round-3 item
  (docs/ROADMAP.md #6).
  """

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('max_rel_dist', 512, 'Max relative distance embedded.')
    p.cls = cls
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    n, h = self._n, self._h
    assert self._nkv == n, 'XL attention uses full KV heads'
    self.CreateVariable('u_var', py_utils.WeightParams(
        [n, h], py_utils.WeightInit.Constant(0.0), p.dtype))
    self.CreateVariable('v_var', py_utils.WeightParams(
        [n, h], py_utils.WeightInit.Constant(0.0), p.dtype))
    self.CreateVariable('pos_proj', py_utils.WeightParams(
        [p.input_dim, n * h], p.params_init, p.dtype))

  def _RelEmb(self, dists: torch.Tensor, theta) -> torch.Tensor:
    """Sinusoidal embedding of signed distances -> [L, N, H]."""
    p = self.p
    d = dists.float().clamp(-p.max_rel_dist, p.max_rel_dist)
    half = p.input_dim // 2
    inv = torch.exp(torch.arange(half, device=d.device, dtype=torch.float32)
                    * -(math.log(10000.0) / max(1, half - 1)))
    ang = d[:, None] * inv[None, :]
    emb = torch.cat([torch.sin(ang), torch.cos(ang)], dim=-1)
    if p.input_dim % 2:
      emb = torch.nn.functional.pad(emb, (0, 1))
    r = torch.matmul(emb.to(theta.pos_proj.dtype), theta.pos_proj)
    return r.reshape(-1, self._n, self._h)

  def FProp(self, theta: NestedMap, query_vec: torch.Tensor,
            paddings: Optional[torch.Tensor] = None,
            memory: Optional[torch.Tensor] = None) -> torch.Tensor:
    """query_vec [B,T,D]; memory [B,M,D] previous-segment hidden states
    (detached inside). Returns [B,T,D]."""
    p = self.p
    n, h = self._n, self._h
    b, t, _ = query_vec.shape
    m = 0 if memory is None else memory.shape[1]
    kv_in = query_vec if memory is None else torch.cat(
        [memory.detach(), query_vec], dim=1)
    q, _, _ = self._Project(theta, query_vec)
    _, k, v = self._Project(theta, kv_in)
    qf, kf, vf = q.float(), k.float(), v.float()
    u = theta.u_var.float()[None, None]   # [1,1,N,H]
    vv = theta.v_var.float()[None, None]
    scale = 1.0 / math.sqrt(h)
    ac = torch.einsum('btnh,bsnh->bnts', qf + u, kf)
    # distances d(i, j) = (m + i) - j for key index j in [0, m+t)
    qpos = torch.arange(t, device=q.device) + m
    kpos = torch.arange(m + t, device=q.device)
    dmat = qpos[:, None] - kpos[None, :]                   # [T, S]
    uniq = torch.arange(-(m + t) + 1, m + t, device=q.device)
    r = self._RelEmb(uniq, theta).float()                  # [L,N,H]
    bd_all = torch.einsum('btnh,lnh->bntl', qf + vv, r)
    idx = (dmat + (m + t) - 1).reshape(-1)
    bd = bd_all.reshape(b, n, t, -1).gather(
        3, idx.reshape(1, 1, t, m + t).expand(b, n, t, m + t))
    logits = (ac + bd) * scale
    mask = kpos[None, :] <= qpos[:, None] if p.causal else \
        torch.ones(t, m + t, dtype=torch.bool, device=q.device)
    mask = mask[None, None].expand(b, 1, t, m + t).clone()
    if paddings is not None:
      kpad = paddings if memory is None else torch.cat(
          [torch.zeros(b, m, device=q.device), paddings], dim=1)
      mask = mask & (kpad[:, None, None, :] < 0.5)
    logits = logits.masked_fill(~mask, -1e30)
    probs = torch.softmax(logits, dim=-1)
    ctx = torch.einsum('bnts,bsnh->btnh', probs, vf)
    ctx = ctx.reshape(b, t, n * h).to(query_vec.dtype)
    post = py_utils.MatmulBias(ctx, theta.post_w,
                               theta.post_b if p.use_bias else None)
    if paddings is not None:
      post = py_utils.ApplyPadding(paddings, post)
    return post
