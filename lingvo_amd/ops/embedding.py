"""Embedding gather/scatter-add custom op (reference
lingvo/core/layers.py:2679 SimpleEmbeddingLayer gather mode).

GPU bf16 fast path via embedding.hip: vectorized gather forward,
deterministic sorted-segment scatter-add backward (no fp32 atomics, so
gradients are bitwise reproducible run-to-run)."""

from __future__ import annotations

import torch

from lingvo_amd.ops import _loader


class _EmbeddingFn(torch.autograd.Function):

  @staticmethod
  def forward(ctx, table, ids, scale):
    ext = _loader.get_ext(required=True)
    ctx.save_for_backward(ids)
    ctx.vocab = table.shape[0]
    ctx.scale = scale
    return ext.emb_gather(table, ids, scale)

  @staticmethod
  def backward(ctx, dy):
    ext = _loader.get_ext(required=True)
    (ids,) = ctx.saved_tensors
    if torch.cuda.is_current_stream_capturing():
      # The deterministic sorted-segment path synchronizes (dynamic
      # unique counts) and cannot be captured in a hipGraph; fall back
      # to capture-safe atomic index_add (nondeterministic summation
      # order, standard embedding-backward semantics).
      d = dy.shape[-1]
      dtable = torch.zeros(ctx.vocab, d, dtype=torch.float32,
                           device=dy.device)
      dtable.index_add_(0, ids.reshape(-1),
                        dy.float().reshape(-1, d) * ctx.scale)
      return dtable.to(dy.dtype), None, None
    dtable = ext.emb_scatter_add(dy.to(torch.bfloat16).contiguous(), ids,
                                 ctx.vocab, ctx.scale)
    return dtable, None, None


def embedding_lookup(table: torch.Tensor, ids: torch.Tensor,
                     scale: float = 1.0) -> torch.Tensor:
  """table [V, D], ids [...] long -> [..., D]. GPU bf16 fast path;
  torch fallback elsewhere."""
  if table.is_cuda and table.dtype == torch.bfloat16 and \
      table.shape[1] % 8 == 0:
    return _EmbeddingFn.apply(table.contiguous(), ids.long(), scale)
  out = torch.nn.functional.embedding(ids.long(), table)
  return out * scale if scale != 1.0 else out
