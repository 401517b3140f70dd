"""Flash-attention autograd wrapper.

`flash_attention(q, k, v, ...)` with [B, T, N, H] tensors. On GPU it runs
the hand-written gfx950 kernel; on CPU a composed fp32 torch reference
with identical mask semantics (used as the numerics oracle in tests).

Mask semantics: key k visible from query q iff
  k < klen[b] AND (win_l < 0 or k >= q - win_l)
              AND (win_r < 0 or k <= q + win_r)
Causal attention = win_r=0. LocalSelfAttention = finite (win_l, win_r)
(reference batch_major_attention.py:2656). Optional clipped relative
position bias table [N, 2*bias_clip+1] indexed by clamp(q-k) + clip.
"""

from __future__ import annotations

import math
from typing import Optional, Tuple

import torch

from lingvo_amd.ops import _loader


def _ref_attention(q, k, v, klen, bias, win_l, win_r, bias_clip, scale,
                   q_segment_ids=None, k_segment_ids=None,
                   chunk_size=0, left_chunks=0):
  """fp32 reference with identical semantics (any backend)."""
  B, T, N, H = q.shape
  S, NKV = k.shape[1], k.shape[2]
  group = N // NKV
  qf = q.float().permute(0, 2, 1, 3)  # [B,N,T,H]
  kf = k.float().permute(0, 2, 1, 3)
  vf = v.float().permute(0, 2, 1, 3)
  if group > 1:
    kf = kf.repeat_interleave(group, dim=1)
    vf = vf.repeat_interleave(group, dim=1)
  logits = torch.einsum('bnth,bnsh->bnts', qf, kf) * scale
  qpos = torch.arange(T, device=q.device)[:, None]
  kpos = torch.arange(S, device=q.device)[None, :]
  if bias is not None:
    d = (qpos - kpos).clamp(-bias_clip, bias_clip) + bias_clip
    logits = logits + bias.float()[:, d]  # [N,T,S] broadcast over B
  mask = torch.ones(T, S, dtype=torch.bool, device=q.device)
  if win_l >= 0:
    mask &= kpos >= qpos - win_l
  if win_r >= 0:
    mask &= kpos <= qpos + win_r
  if chunk_size > 0:
    qc, kc = qpos // chunk_size, kpos // chunk_size
    mask &= (kc <= qc) & (kc >= qc - left_chunks)
  mask = mask[None, None]
  if klen is not None:
    mask = mask & (kpos[None, None] < klen[:, None, None, None])
  if q_segment_ids is not None:
    seg = (q_segment_ids[:, :, None] == k_segment_ids[:, None, :])
    mask = mask & seg[:, None]
  logits = logits.masked_fill(~mask, -1e30)
  probs = torch.softmax(logits, dim=-1)
  # fully-masked rows -> 0
  probs = torch.where(mask.any(-1, keepdim=True), probs,
                      torch.zeros_like(probs))
  out = torch.einsum('bnts,bnsh->bnth', probs, vf)
  return out.permute(0, 2, 1, 3)


class _FlashAttnFn(torch.autograd.Function):

  @staticmethod
  def forward(ctx, q, k, v, klen, bias, qseg, kseg, win_l, win_r,
              bias_clip, scale, chunk_size, left_chunks):
    ext = _loader.get_ext(required=True)
    bias_b = None if bias is None else bias.to(torch.bfloat16).contiguous()
    o, lse = ext.fa_fwd(q, k, v, klen, bias_b, qseg, kseg, win_l, win_r,
                        bias_clip, scale, chunk_size, left_chunks)
    empty = torch.empty(0)
    ctx.save_for_backward(q, k, v, o, lse,
                          klen if klen is not None else empty,
                          bias_b if bias_b is not None else empty,
                          qseg if qseg is not None else empty,
                          kseg if kseg is not None else empty)
    ctx.cfg = (win_l, win_r, bias_clip, scale, bias is not None and
               bias.requires_grad, None if bias is None else bias.dtype,
               chunk_size, left_chunks)
    return o

  @staticmethod
  def backward(ctx, dout):
    ext = _loader.get_ext(required=True)
    q, k, v, o, lse, klen, bias_b, qseg, kseg = ctx.saved_tensors
    (win_l, win_r, bias_clip, scale, bias_grad, bias_dtype, chunk_size,
     left_chunks) = ctx.cfg
    klen = klen if klen.numel() else None
    bias_b = bias_b if bias_b.numel() else None
    qseg = qseg if qseg.numel() else None
    kseg = kseg if kseg.numel() else None
    dq, dk, dv, dbias = ext.fa_bwd(
        dout.contiguous(), q, k, v, o, lse, klen, bias_b, qseg, kseg,
        bias_grad, win_l, win_r, bias_clip, scale, chunk_size,
        left_chunks)
    dbias_out = dbias.to(bias_dtype) if bias_grad else None
    return (dq, dk, dv, None, dbias_out, None, None, None, None, None,
            None, None, None)


def flash_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                    klen: Optional[torch.Tensor] = None,
                    bias: Optional[torch.Tensor] = None,
                    win_l: int = -1, win_r: int = -1,
                    bias_clip: int = 127,
                    scale: Optional[float] = None,
                    q_segment_ids: Optional[torch.Tensor] = None,
                    k_segment_ids: Optional[torch.Tensor] = None,
                    chunk_size: int = 0, left_chunks: int = 0
                    ) -> torch.Tensor:
  """q [B,T,N,H], k/v [B,S,NKV,H] -> [B,T,N,H]. Optional packed-input
  segment ids [B,T]/[B,S]: attention is blocked across segments
  (reference PackSequences segment_ids consumed by the mask path)."""
  if scale is None:
    scale = 1.0 / math.sqrt(q.shape[-1])
  if klen is not None:
    klen = klen.to(torch.int32).contiguous()
  if q_segment_ids is not None:
    q_segment_ids = q_segment_ids.to(torch.int32).contiguous()
    k_segment_ids = k_segment_ids.to(torch.int32).contiguous()
  if q.is_cuda:
    orig = q.dtype
    out = _FlashAttnFn.apply(
        q.to(torch.bfloat16).contiguous(), k.to(torch.bfloat16).contiguous(),
        v.to(torch.bfloat16).contiguous(), klen, bias, q_segment_ids,
        k_segment_ids, win_l, win_r, bias_clip, scale, chunk_size,
        left_chunks)
    return out.to(orig) if orig != torch.bfloat16 else out
  out = _ref_attention(q, k, v, klen, bias, win_l, win_r, bias_clip, scale,
                       q_segment_ids, k_segment_ids, chunk_size,
                       left_chunks)
  return out.to(q.dtype)
