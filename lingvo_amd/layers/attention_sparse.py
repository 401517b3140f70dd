"""Efficient attention variants: chunkwise, routing (k-means sparse),
and Performer/FAVOR+ linear attention.

MI355X-native re-implementations of the reference's long-sequence
attention family (lingvo/core/batch_major_attention.py:
ChunkwiseSelfAttention:4008, RoutingAttention:4458 +
attention_util.py:656 KMeansClusteringForAtten; favor_attention.py
FAVOR+). All share MultiHeadedAttention's projections and FProp
surface, so they are drop-in `atten_tpl` choices.

These variants reshape the problem rather than the kernel: chunkwise
runs the dense flash kernel per chunk; routing gathers per-cluster key
subsets then runs dense attention; Performer replaces softmax with a
random-feature kernel so attention is two GEMMs (prefix sums when
causal) — all GEMM-shaped work lands on MFMA via hipBLASLt.
"""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn.functional as F

from lingvo_amd.core import py_utils
from lingvo_amd.core.nested_map import NestedMap
from lingvo_amd.layers import attention as attention_lib
from lingvo_amd.ops import flash_attn


class ChunkwiseSelfAttention(attention_lib.MultiHeadedAttention):
  """Attention within fixed chunks of the sequence (reference
  batch_major_attention.py:4008). Queries in chunk c attend to keys in
  chunks [c - left_chunks, c] (causal within the current chunk when
  p.causal). Memory/time O(S * W) instead of O(S^2)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('chunk_size', 128, 'Chunk width W.')
    p.Define('left_chunks', 0, 'Extra previous chunks visible.')
    p.cls = cls
    return p

  def FProp(self, theta: NestedMap, query_vec: torch.Tensor,
            paddings: Optional[torch.Tensor] = None,
            segment_ids: Optional[torch.Tensor] = None) -> torch.Tensor:
    """Runs the chunk mask NATIVELY inside the flash kernel (chunk_size/
    left_chunks args): no widened K/V copies, and key tiles outside the
    chunk window are skipped in-kernel — O(T*W) work on GPU and CPU."""
    p = self.p
    assert segment_ids is None, 'packed inputs: use the flash seg path'
    b, t, _ = query_vec.shape
    q, k, v = self._Project(theta, query_vec)
    if p.use_rope:
      q = self.rope.FProp(theta.rope, q)
      k = self.rope.FProp(theta.rope, k)
    klen = (py_utils.LengthsFromPaddings(paddings).to(torch.int32)
            if paddings is not None else None)
    out = flash_attn.flash_attention(
        q, k, v, klen=klen, win_l=-1, win_r=0 if p.causal else -1,
        chunk_size=p.chunk_size, left_chunks=p.left_chunks)
    out = out.reshape(b, t, self._n * self._h)
    post = py_utils.MatmulBias(out, theta.post_w,
                               theta.post_b if p.use_bias else None)
    if paddings is not None:
      post = py_utils.ApplyPadding(paddings, post)
    return post


class RoutingAttention(attention_lib.MultiHeadedAttention):
  """k-means routed sparse attention (reference
  batch_major_attention.py:4458): queries attend only to the
  `atten_window` keys nearest (by cluster) to them. Per-head centroids
  are EMA-updated buffers; queries/keys are routed to their nearest
  centroid and attention runs within each cluster's gathered key set."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('num_clusters', 4, 'k-means clusters per head.')
    p.Define('atten_window', 64, 'Keys gathered per cluster (capacity).')
    p.Define('decay', 0.999, 'Centroid EMA decay.')
    p.cls = cls
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    self.register_buffer('centroids', torch.randn(
        self._n, p.num_clusters, self._h) * (1.0 / math.sqrt(self._h)))

  def _Route(self, x):
    """x [B,T,N,H] -> nearest-centroid id [B,N,T] using cosine sim."""
    xn = F.normalize(x.float(), dim=-1)
    cn = F.normalize(self.centroids.float(), dim=-1)
    sim = torch.einsum('btnh,nch->bntc', xn, cn)
    return sim.argmax(-1), xn

  def FProp(self, theta: NestedMap, query_vec: torch.Tensor,
            paddings: Optional[torch.Tensor] = None,
            segment_ids: Optional[torch.Tensor] = None) -> torch.Tensor:
    p = self.p
    assert segment_ids is None
    b, t, _ = query_vec.shape
    n, h = self._n, self._h
    assert self._nkv == n, 'routing attention requires full KV heads'
    q, k, v = self._Project(theta, query_vec)
    q_cl, _ = self._Route(q)                  # [B,N,T]
    k_cl, kn = self._Route(k)
    if self.training:
      with torch.no_grad():
        # EMA centroid update toward the mean of assigned keys
        for c in range(p.num_clusters):
          msk = (k_cl == c).float().permute(0, 2, 1).unsqueeze(-1)
          denom = msk.sum((0, 1)).clamp_min(1.0)  # [N,1]
          mean = (kn.detach() * msk).sum((0, 1)) / denom
          alive = (msk.sum((0, 1)) > 0).float()
          target = alive * mean + (1 - alive) * self.centroids[:, c]
          self.centroids[:, c] = (p.decay * self.centroids[:, c] +
                                  (1 - p.decay) * target)
    wnd = min(p.atten_window, t)
    kpad = paddings if paddings is not None else torch.zeros(
        b, t, device=q.device)
    out = torch.zeros(b, t, n, h, dtype=torch.float32, device=q.device)
    qf, kf, vf = q.float(), k.float(), v.float()
    scale = 1.0 / math.sqrt(h)
    for c in range(p.num_clusters):
      # scores for membership: keys in cluster c, unpadded
      k_in = (k_cl == c) & (kpad.unsqueeze(1) < 0.5)      # [B,N,T]
      # gather up to wnd member keys; deterministic earliest-first
      # tiebreak (graph-safe: no RNG)
      tiebreak = torch.linspace(1e-3, 0.0, t, device=q.device)
      top = (k_in.float() + tiebreak).topk(wnd, dim=-1).indices
      kg = torch.gather(
          kf.permute(0, 2, 1, 3), 2,
          top.unsqueeze(-1).expand(-1, -1, -1, h))          # [B,N,wnd,H]
      vg = torch.gather(vf.permute(0, 2, 1, 3), 2,
                        top.unsqueeze(-1).expand(-1, -1, -1, h))
      valid = torch.gather(k_in, 2, top)                    # [B,N,wnd]
      logits = torch.einsum('btnh,bnwh->bntw',
                            qf, kg) * scale
      logits = logits.masked_fill(~valid.unsqueeze(2), -1e30)
      if p.causal:
        qpos = torch.arange(t, device=q.device)[None, None, :, None]
        kpos = top.unsqueeze(2)
        logits = logits.masked_fill(kpos > qpos, -1e30)
      probs = torch.softmax(logits, dim=-1)
      all_masked = (logits.max(-1, keepdim=True).values < -1e29)
      probs = probs.masked_fill(all_masked, 0.0)
      ctx = torch.einsum('bntw,bnwh->btnh', probs, vg)
      sel = (q_cl == c).permute(0, 2, 1).unsqueeze(-1).float()
      out = out + ctx * sel
    ctx = out.reshape(b, t, n * h).to(query_vec.dtype)
    post = py_utils.MatmulBias(ctx, theta.post_w,
                               theta.post_b if p.use_bias else None)
    if paddings is not None:
      post = py_utils.ApplyPadding(paddings, post)
    return post


class PerformerAttention(attention_lib.MultiHeadedAttention):
  """FAVOR+ linear attention (reference favor_attention.py; Choromanski
  et al. 2021). Positive random features phi(x) = exp(w.x - |x|^2/2)
  give an unbiased softmax estimate; attention becomes two GEMMs
  (prefix sums when causal): O(T * m * H)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('num_random_features', 128, 'Feature count m.')
    p.Define('redraw_each_step', False, 'Resample features every FProp.')
    p.cls = cls
    return p

  def __init__(self, params):
    super().__init__(params)
    g = torch.Generator().manual_seed(
        (self.p.random_seed or 1234) & 0x7FFFFFFF)
    self.register_buffer('proj_mat', self._Orthogonal(g))

  def _Orthogonal(self, g):
    p = self.p
    m, h = p.num_random_features, self._h
    blocks = []
    for _ in range((m + h - 1) // h):
      a = torch.randn(h, h, generator=g)
      qmat, _ = torch.linalg.qr(a)
      blocks.append(qmat * math.sqrt(h))
    return torch.cat(blocks, 0)[:m]  # [m, H]

  def _Phi(self, x, scale, is_key):
    # x [B,T,N,H] -> [B,T,N,m]; stabilized positive features. The
    # stabilizer must be CONSTANT across key positions (else it
    # reweights the softmax), so keys share one per-(B,N) max; a
    # per-query stabilizer cancels in the num/den ratio and is fine.
    xs = x.float() * math.sqrt(scale)
    wx = torch.einsum('btnh,mh->btnm', xs, self.proj_mat.float())
    z = wx - 0.5 * (xs * xs).sum(-1, keepdim=True)
    if is_key:
      stab = z.amax(dim=(1, 3), keepdim=True).detach()
    else:
      stab = z.amax(dim=-1, keepdim=True).detach()
    return torch.exp(z - stab) / math.sqrt(
        self.p.num_random_features) + 1e-6

  def FProp(self, theta: NestedMap, query_vec: torch.Tensor,
            paddings: Optional[torch.Tensor] = None,
            segment_ids: Optional[torch.Tensor] = None) -> torch.Tensor:
    p = self.p
    assert segment_ids is None
    if p.redraw_each_step and self.training:
      g = torch.Generator().manual_seed(
          int(py_utils.GenerateStepSeedPair()[0]) & 0x7FFFFFFF)
      self.proj_mat = self._Orthogonal(g).to(self.proj_mat.device)
    b, t, _ = query_vec.shape
    n, h = self._n, self._h
    assert self._nkv == n, 'Performer requires full KV heads'
    q, k, v = self._Project(theta, query_vec)
    scale = 1.0 / math.sqrt(h)
    qp = self._Phi(q, scale, is_key=False)   # [B,T,N,m]
    kp = self._Phi(k, scale, is_key=True)
    if paddings is not None:
      kp = kp * (1.0 - paddings)[:, :, None, None]
    vf = v.float()
    if p.causal:
      # prefix sums over time
      kv = torch.einsum('btnm,btnh->btnmh', kp, vf).cumsum(1)
      z = kp.cumsum(1)
      num = torch.einsum('btnm,btnmh->btnh', qp, kv)
      den = torch.einsum('btnm,btnm->btn', qp, z).clamp_min(1e-6)
    else:
      kv = torch.einsum('btnm,btnh->bnmh', kp, vf)
      z = kp.sum(1)  # [B,N,m]
      num = torch.einsum('btnm,bnmh->btnh', qp, kv)
      den = torch.einsum('btnm,bnm->btn', qp, z).clamp_min(1e-6)
    ctx = (num / den.unsqueeze(-1)).reshape(b, t, n * h).to(
        query_vec.dtype)
    post = py_utils.MatmulBias(ctx, theta.post_w,
                               theta.post_b if p.use_bias else None)
    if paddings is not None:
      post = py_utils.ApplyPadding(paddings, post)
    return post


class BlockSparseAttention(attention_lib.MultiHeadedAttention):
  """Attention under a fixed block-level visibility pattern (reference
  self_attention_layer.py:30 BlockSparseAttention): queries in block i
  attend keys in block j iff block_mask[i][j]. Patterns like
  local+global land as small boolean matrices; on GPU the masked
  blocks are skipped by the flash kernel's tile loop (round 2), the
  dense-mask fp32 form below is the oracle."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('block_size', 64, 'Block width in tokens.')
    p.Define('block_mask', None,
             'Nested list / tensor [nq_blocks, nk_blocks] of 0/1; '
             'None = full attention.')
    p.cls = cls
    return p

  def FProp(self, theta: NestedMap, query_vec: torch.Tensor,
            paddings: Optional[torch.Tensor] = None,
            segment_ids: Optional[torch.Tensor] = None) -> torch.Tensor:
    p = self.p
    assert segment_ids is None
    b, t, _ = query_vec.shape
    n, h = self._n, self._h
    assert self._nkv == n
    q, k, v = self._Project(theta, query_vec)
    scale = 1.0 / math.sqrt(h)
    logits = torch.einsum('btnh,bsnh->bnts', q.float(),
                          k.float()) * scale
    tpos = torch.arange(t, device=q.device)
    mask = torch.ones(t, t, dtype=torch.bool, device=q.device)
    if p.block_mask is not None:
      bm = torch.as_tensor(p.block_mask, dtype=torch.bool,
                           device=q.device)
      qb = (tpos // p.block_size).clamp(max=bm.shape[0] - 1)
      kb = (tpos // p.block_size).clamp(max=bm.shape[1] - 1)
      mask = bm[qb[:, None], kb[None, :]]
    if p.causal:
      mask = mask & (tpos[None, :] <= tpos[:, None])
    mask4 = mask[None, None].expand(b, 1, t, t).clone()
    if paddings is not None:
      mask4 = mask4 & (paddings[:, None, None, :] < 0.5)
    logits = logits.masked_fill(~mask4, -1e30)
    probs = torch.softmax(logits, dim=-1)
    probs = torch.where(mask4.any(-1, keepdim=True), probs,
                        torch.zeros_like(probs))
    ctx = torch.einsum('bnts,bsnh->btnh', probs, v.float())
    ctx = ctx.reshape(b, t, n * h).to(query_vec.dtype)
    post = py_utils.MatmulBias(ctx, theta.post_w,
                               theta.post_b if p.use_bias else None)
    if paddings is not None:
      post = py_utils.ApplyPadding(paddings, post)
    return post


def LocalGlobalBlockMask(num_blocks: int, num_global: int = 1,
                         local_width: int = 1) -> list:
  """Common block pattern: the first `num_global` blocks see/are seen
  by everything; other blocks see +-local_width neighbors."""
  m = [[0] * num_blocks for _ in range(num_blocks)]
  for i in range(num_blocks):
    for j in range(num_blocks):
      if i < num_global or j < num_global or abs(i - j) <= local_width:
        m[i][j] = 1
  return m
