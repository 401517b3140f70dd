"""Ring-attention context parallelism (CP) over RCCL/xGMI send-recv.

The reference has NO context parallelism — it scales sequence length with
banded/local attention (batch_major_attention.py:2656), chunkwise
attention (4008) and sequence packing. On MI355X the natural long-context
strategy is ring attention: shard the sequence across the CP group, keep
Q local, and rotate the K/V shards around the xGMI ring, merging partial
attention results with log-sum-exp accumulation (exact, not approximate).

Each of the cp ring steps overlaps the next K/V hop with the current
block's attention math; on 8 GPUs the per-hop payload is 2*B*(S/cp)*NKV*H
bf16 over one 153 GB/s xGMI link.

Exactness: softmax(QK^T)V over the full sequence equals the LSE-weighted
combination of per-block partial attentions; gradients flow back through
the reverse ring automatically (each `_RingExchange.backward` sends the
K/V gradient back to the rank that produced the shard).
"""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.distributed as dist

from lingvo_amd.core import py_utils
from lingvo_amd.core.nested_map import NestedMap
from lingvo_amd.layers import attention as attention_lib


def _GlobalRank(group, group_rank: int) -> int:
  if group is None or group is dist.group.WORLD:
    return group_rank
  return dist.get_global_rank(group, group_rank)


class _RingExchange(torch.autograd.Function):
  """Rotate a tensor one hop around the CP ring (rank -> rank+1).

  Backward rotates the gradient the opposite way, so dK/dV land on the
  rank that owns that K/V shard.
  """

  @staticmethod
  def forward(ctx, x, group):
    rank = dist.get_rank(group)
    world = dist.get_world_size(group)
    nxt = _GlobalRank(group, (rank + 1) % world)
    prv = _GlobalRank(group, (rank - 1) % world)
    ctx.group, ctx.nxt, ctx.prv = group, nxt, prv
    x = x.contiguous()
    buf = torch.empty_like(x)
    reqs = [dist.irecv(buf, prv, group=group),
            dist.isend(x, nxt, group=group)]
    for r in reqs:
      r.wait()
    return buf

  @staticmethod
  def backward(ctx, g):
    g = g.contiguous()  # recv buffer must be contiguous (empty_like
    buf = torch.empty_like(g)  # copies the input's strides)
    reqs = [dist.irecv(buf, ctx.nxt, group=ctx.group),
            dist.isend(g, ctx.prv, group=ctx.group)]
    for r in reqs:
      r.wait()
    return buf, None


def _BlockAttention(q, k, v, q_off, k_off, causal, scale, blk_klen):
  """Partial attention of local Q against one K/V block.

  q: [B, Tq, N, H] fp32; k, v: [B, Sk, NKV, H] fp32.
  Returns (out [B, Tq, N, H], lse [B, N, Tq]) where out is the
  softmax-normalized context WITHIN the block and lse the block's
  log-sum-exp, so blocks combine exactly.
  """
  b, tq, n, h = q.shape
  sk, nkv = k.shape[1], k.shape[2]
  qf = q.permute(0, 2, 1, 3)  # [B,N,T,H]
  kf = k.permute(0, 2, 1, 3)
  vf = v.permute(0, 2, 1, 3)
  if n != nkv:
    kf = kf.repeat_interleave(n // nkv, dim=1)
    vf = vf.repeat_interleave(n // nkv, dim=1)
  logits = torch.einsum('bnth,bnsh->bnts', qf, kf) * scale
  kpos = torch.arange(sk, device=q.device)[None, :]
  mask = torch.ones(tq, sk, dtype=torch.bool, device=q.device)
  if causal:
    qpos = torch.arange(tq, device=q.device)[:, None] + q_off
    mask &= (kpos + k_off) <= qpos
  mask = mask[None, None]
  if blk_klen is not None:
    mask = mask & (kpos[None, None] < blk_klen[:, None, None, None])
  logits = logits.masked_fill(~mask, -1e30)
  lse = torch.logsumexp(logits, dim=-1)  # [B,N,T]
  probs = torch.exp(logits - lse.unsqueeze(-1))
  out = torch.einsum('bnts,bnsh->bnth', probs, vf)
  return out.permute(0, 2, 1, 3), lse


def RingAttention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                  klen: Optional[torch.Tensor] = None,
                  causal: bool = False,
                  scale: Optional[float] = None,
                  group=None) -> torch.Tensor:
  """Exact attention over a sequence sharded across the CP group.

  q, k, v: LOCAL shards [B, S/cp, N|NKV, H] — rank r holds global
  positions [r*S/cp, (r+1)*S/cp). klen: GLOBAL sequence lengths [B]
  (trailing padding in the global sequence). Returns the local output
  shard [B, S/cp, N, H].
  """
  if scale is None:
    scale = 1.0 / math.sqrt(q.shape[-1])
  world = dist.get_world_size(group) if dist.is_initialized() else 1
  if world == 1:
    from lingvo_amd.ops import flash_attn
    return flash_attn.flash_attention(
        q, k, v, None if klen is None else klen.to(torch.int32),
        None, -1, 0 if causal else -1)
  rank = dist.get_rank(group)
  tq = q.shape[1]
  q_off = rank * tq
  qf = q.float()
  kv = torch.stack([k.float(), v.float()])
  outs, lses = [], []
  for i in range(world):
    src = (rank - i) % world
    k_off = src * tq
    # NOTE: fully-masked causal blocks are still computed (their LSE
    # weight underflows to 0) so every rank's autograd graph contains
    # the same ring-exchange sequence — skipping would orphan the
    # exchange on some ranks and deadlock the reverse ring. Zigzag
    # (load-balanced) sharding is the round-2 fix for the wasted math.
    blk_klen = None if klen is None else (klen - k_off).clamp(0, tq)
    out_b, lse_b = _BlockAttention(qf, kv[0], kv[1], q_off, k_off,
                                   causal, scale, blk_klen)
    outs.append(out_b)
    lses.append(lse_b)
    if i < world - 1:
      kv = _RingExchange.apply(kv, group)
  lse_all = torch.stack(lses, dim=-1)            # [B,N,T,nblk]
  lse_tot = torch.logsumexp(lse_all, dim=-1)     # [B,N,T]
  w = torch.exp(lse_all - lse_tot.unsqueeze(-1))  # [B,N,T,nblk]
  out = sum(w[..., i, None] * o.permute(0, 2, 1, 3)
            for i, o in enumerate(outs))
  # fully-padded query rows (lse ~ -1e30) -> 0.
  out = torch.where(lse_tot.unsqueeze(-1) > -1e29, out,
                    torch.zeros_like(out))
  return out.permute(0, 2, 1, 3).to(q.dtype)


class CpMultiHeadedAttention(attention_lib.MultiHeadedAttention):
  """MultiHeadedAttention whose sequence dim is sharded over a CP group.

  Drop-in for the base layer when activations are sequence-sharded:
  projections are per-position (replicated weights act on the local
  shard); the attention itself runs as ring attention. Weight gradients
  must be all-reduced over the CP group after backward (the DP GradSync
  bucket does this when the CP group is folded into the DP group).
  """

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('cp_group', None, 'torch.distributed group; None = WORLD.')
    p.cls = cls
    return p

  def FProp(self, theta: NestedMap, query_vec: torch.Tensor,
            paddings: Optional[torch.Tensor] = None,
            segment_ids: Optional[torch.Tensor] = None) -> torch.Tensor:
    p = self.p
    if segment_ids is not None:
      raise NotImplementedError('packed inputs under CP')
    if p.rel_pos_bias or p.left_context >= 0 or p.right_context >= 0:
      raise NotImplementedError(
          'ring attention supports full or causal masks')
    q, k, v = self._Project(theta, query_vec)
    klen = None
    if paddings is not None:
      # local non-pad counts -> global lengths (padding is trailing in
      # the GLOBAL sequence, so summing local counts is exact).
      klen = py_utils.LengthsFromPaddings(paddings)
      if dist.is_initialized() and dist.get_world_size(p.cp_group) > 1:
        klen = klen.clone()
        dist.all_reduce(klen, group=p.cp_group)
    out = RingAttention(q, k, v, klen=klen, causal=p.causal,
                        group=p.cp_group)
    if p.atten_dropout_prob and not self.do_eval:
      out = py_utils.DeterministicDropout(out, 1.0 - p.atten_dropout_prob)
    b, t = out.shape[0], out.shape[1]
    ctx = out.reshape(b, t, self._n * self._h)
    post = py_utils.MatmulBias(ctx, theta.post_w,
                               theta.post_b if p.use_bias else None)
    if paddings is not None:
      post = py_utils.ApplyPadding(paddings, post)
    return post


def ShardSequence(x: torch.Tensor, rank: int, world: int,
                  dim: int = 1) -> torch.Tensor:
  """Slice a [.., S, ..] tensor into this rank's contiguous CP shard."""
  s = x.shape[dim]
  assert s % world == 0, f'seq len {s} not divisible by cp={world}'
  return x.narrow(dim, rank * (s // world), s // world)


class _SeqHeadAllToAll(torch.autograd.Function):
  """All-to-all along dim 0; backward is the inverse all-to-all."""

  @staticmethod
  def forward(ctx, x, group):
    ctx.group = group
    x = x.contiguous()
    out = torch.empty_like(x)
    dist.all_to_all_single(out, x, group=group)
    return out

  @staticmethod
  def backward(ctx, g):
    g = g.contiguous()
    out = torch.empty_like(g)
    dist.all_to_all_single(out, g, group=ctx.group)
    return out, None


def _ScatterHeadsGatherSeq(x: torch.Tensor, world: int, group):
  """[B, s, N, H] seq-sharded -> [B, s*world, N/world, H]: each rank
  gets the FULL sequence for its N/world heads (Ulysses layout)."""
  b, s, n, h = x.shape
  nl = n // world
  xs = x.reshape(b, s, world, nl, h).permute(2, 0, 1, 3, 4).contiguous()
  recv = _SeqHeadAllToAll.apply(xs, group)          # [P, B, s, nl, H]
  return recv.permute(1, 0, 2, 3, 4).reshape(b, world * s, nl, h)


def _GatherHeadsScatterSeq(x: torch.Tensor, world: int, group):
  """Inverse of _ScatterHeadsGatherSeq: [B, S, nl, H] -> [B, S/world,
  nl*world, H]."""
  b, stot, nl, h = x.shape
  s = stot // world
  xs = x.reshape(b, world, s, nl, h).permute(1, 0, 2, 3, 4).contiguous()
  recv = _SeqHeadAllToAll.apply(xs, group)          # [P, B, s, nl, H]
  return recv.permute(1, 2, 0, 3, 4).reshape(b, s, world * nl, h)


class UlyssesMultiHeadedAttention(attention_lib.MultiHeadedAttention):
  """DeepSpeed-Ulysses-style sequence parallelism: instead of rotating
  K/V around a ring, ONE all-to-all re-shards activations from
  sequence-split to head-split, the full-sequence attention runs
  locally on N/P heads (the dense flash kernel, unchanged), and a
  second all-to-all restores the sequence split. Two a2a pairs per
  layer vs cp ring hops — on 8-way xGMI the a2a moves each byte once
  over one link, so Ulysses wins when N >= P and the ring wins for
  GQA-heavy configs (N/P < 1). Both live behind the same atten_tpl
  surface."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('cp_group', None, 'Sequence-parallel group; None = WORLD.')
    p.cls = cls
    return p

  def FProp(self, theta: NestedMap, query_vec: torch.Tensor,
            paddings: Optional[torch.Tensor] = None,
            segment_ids: Optional[torch.Tensor] = None) -> torch.Tensor:
    p = self.p
    if segment_ids is not None:
      raise NotImplementedError('packed inputs under Ulysses SP')
    world = dist.get_world_size(p.cp_group) if dist.is_initialized() \
        else 1
    if world == 1:
      return super().FProp(theta, query_vec, paddings)
    assert self._n % world == 0, (self._n, world)
    assert self._nkv % world == 0, 'GQA heads must divide the SP degree'
    q, k, v = self._Project(theta, query_vec)
    if p.use_rope:
      # positions are GLOBAL: offset by this rank's sequence start
      rank = dist.get_rank(p.cp_group)
      s = q.shape[1]
      pos = (rank * s + torch.arange(
          s, device=q.device, dtype=torch.float32)).expand(q.shape[0], s)
      q = self.rope.FProp(theta.rope, q, position=pos)
      k = self.rope.FProp(theta.rope, k, position=pos)
    qg = _ScatterHeadsGatherSeq(q, world, p.cp_group)
    kg = _ScatterHeadsGatherSeq(k, world, p.cp_group)
    vg = _ScatterHeadsGatherSeq(v, world, p.cp_group)
    klen = None
    if paddings is not None:
      klen = py_utils.LengthsFromPaddings(paddings).clone()
      dist.all_reduce(klen, group=p.cp_group)
      klen = klen.to(torch.int32)
    bias = None
    if p.rel_pos_bias:
      # local head slice of the bias table (chunk r lands on rank r)
      rank = dist.get_rank(p.cp_group)
      nl = self._n // world
      bias = theta.rel_bias[rank * nl:(rank + 1) * nl]
    from lingvo_amd.ops import flash_attn
    out = flash_attn.flash_attention(
        qg, kg, vg, klen, bias,
        p.left_context, 0 if p.causal else p.right_context,
        p.rel_pos_clip)
    out = _GatherHeadsScatterSeq(out, world, p.cp_group)
    if p.atten_dropout_prob and not self.do_eval:
      out = py_utils.DeterministicDropout(out, 1.0 - p.atten_dropout_prob)
    b, t = out.shape[0], out.shape[1]
    ctx = out.reshape(b, t, self._n * self._h)
    post = py_utils.MatmulBias(ctx, theta.post_w,
                               theta.post_b if p.use_bias else None)
    if paddings is not None:
      post = py_utils.ApplyPadding(paddings, post)
    return post


def ZigzagShard(x: torch.Tensor, rank: int, world: int,
                dim: int = 1) -> torch.Tensor:
  """Zigzag CP sharding for causal load balance: the sequence splits
  into 2P chunks and rank r holds chunks (r, 2P-1-r), so every rank
  sees the same mix of early (cheap) and late (expensive) causal
  positions. Returns concat of the two chunks along `dim`."""
  s = x.shape[dim]
  assert s % (2 * world) == 0, (s, world)
  L = s // (2 * world)
  lo = x.narrow(dim, rank * L, L)
  hi = x.narrow(dim, (2 * world - 1 - rank) * L, L)
  return torch.cat([lo, hi], dim=dim)


def ZigzagPositions(rank: int, world: int, total_len: int,
                    device=None) -> torch.Tensor:
  """Global positions of rank r's zigzag shard, [total_len/P]."""
  L = total_len // (2 * world)
  lo = torch.arange(rank * L, (rank + 1) * L, device=device)
  hi = torch.arange((2 * world - 1 - rank) * L,
                    (2 * world - rank) * L, device=device)
  return torch.cat([lo, hi])


def _BlockAttentionPos(q, k, v, q_pos, k_pos, causal, scale, klen):
  """_BlockAttention with explicit (possibly non-contiguous) global
  positions per query/key — the zigzag form."""
  b, tq, n, h = q.shape
  sk, nkv = k.shape[1], k.shape[2]
  qf = q.permute(0, 2, 1, 3)
  kf = k.permute(0, 2, 1, 3)
  vf = v.permute(0, 2, 1, 3)
  if n != nkv:
    kf = kf.repeat_interleave(n // nkv, dim=1)
    vf = vf.repeat_interleave(n // nkv, dim=1)
  logits = torch.einsum('bnth,bnsh->bnts', qf, kf) * scale
  mask = torch.ones(tq, sk, dtype=torch.bool, device=q.device)
  if causal:
    mask &= k_pos[None, :] <= q_pos[:, None]
  mask = mask[None, None]
  if klen is not None:
    mask = mask & (k_pos[None, None, None, :] <
                   klen[:, None, None, None])
  logits = logits.masked_fill(~mask, -1e30)
  lse = torch.logsumexp(logits, dim=-1)
  probs = torch.exp(logits - lse.unsqueeze(-1))
  out = torch.einsum('bnts,bnsh->bnth', probs, vf)
  return out.permute(0, 2, 1, 3), lse


def RingAttentionZigzag(q: torch.Tensor, k: torch.Tensor,
                        v: torch.Tensor,
                        klen: Optional[torch.Tensor] = None,
                        causal: bool = True,
                        scale: Optional[float] = None,
                        group=None) -> torch.Tensor:
  """Ring attention over ZigzagShard-ed q/k/v: same exact LSE merge as
  RingAttention, but every rank does ~the same causal work per block
  (the round-2 balance fix, landed early since it is exactness-testable
  on CPU). Inputs/outputs are in zigzag-local layout."""
  if scale is None:
    scale = 1.0 / math.sqrt(q.shape[-1])
  world = dist.get_world_size(group) if dist.is_initialized() else 1
  rank = dist.get_rank(group) if world > 1 else 0
  t_local = q.shape[1]
  total = t_local * world
  q_pos = ZigzagPositions(rank, world, total, device=q.device)
  qf = q.float()
  # kv travels with its source rank id so positions can be derived
  kv = torch.stack([k.float(), v.float()])
  outs, lses = [], []
  for i in range(world):
    src = (rank - i) % world
    k_pos = ZigzagPositions(src, world, total, device=q.device)
    out_b, lse_b = _BlockAttentionPos(qf, kv[0], kv[1], q_pos, k_pos,
                                      causal, scale, klen)
    outs.append(out_b)
    lses.append(lse_b)
    if i < world - 1:
      kv = _RingExchange.apply(kv, group)
  lse_all = torch.stack(lses, dim=-1)
  lse_tot = torch.logsumexp(lse_all, dim=-1)
  w = torch.exp(lse_all - lse_tot.unsqueeze(-1))
  out = sum(w[..., i, None] * o.permute(0, 2, 1, 3)
            for i, o in enumerate(outs))
  out = torch.where(lse_tot.unsqueeze(-1) > -1e29, out,
                    torch.zeros_like(out))
  return out.permute(0, 2, 1, 3).to(q.dtype)
