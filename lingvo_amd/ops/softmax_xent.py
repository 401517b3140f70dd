"""Fused logits-GEMM + softmax cross-entropy (K8).

logits = x @ w + b runs on hipBLASLt; the softmax/xent fwd+bwd run in one
HIP pass each without materializing fp32 log-probs.
"""

from __future__ import annotations

import torch

from lingvo_amd.ops import _loader


class _LogitsXentFn(torch.autograd.Function):

  @staticmethod
  def forward(ctx, x, w, b, labels):
    ext = _loader.get_ext(required=True)
    logits = (torch.matmul(x, w) + b).contiguous()
    loss, lse = ext.xent_fwd(logits, labels)
    ctx.save_for_backward(x, w, logits, labels, lse)
    return loss

  @staticmethod
  def backward(ctx, gout):
    ext = _loader.get_ext(required=True)
    x, w, logits, labels, lse = ctx.saved_tensors
    dlogits = ext.xent_bwd(logits, labels, lse,
                           gout.contiguous().float())
    dx = torch.matmul(dlogits, w.t())
    dw = torch.matmul(x.t(), dlogits)
    db = dlogits.sum(0)
    return dx, dw, db, None


def logits_xent(x: torch.Tensor, w: torch.Tensor, b: torch.Tensor,
                labels: torch.Tensor) -> torch.Tensor:
  """x [R, D] bf16, w [D, V], b [V], labels [R] -> per-example xent [R] f32."""
  return _LogitsXentFn.apply(
      x.to(torch.bfloat16), w.to(torch.bfloat16), b.to(torch.bfloat16),
      labels.long().contiguous())
