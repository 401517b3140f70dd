"""Fused LSTM gate math (K11) autograd wrapper."""

from __future__ import annotations

import torch

from lingvo_amd.ops import _loader


class _LstmGatesFn(torch.autograd.Function):

  @staticmethod
  def forward(ctx, gates, c0, fgb, cap):
    ext = _loader.get_ext(required=True)
    c1, m1 = ext.lstm_gates_fwd(gates, c0, fgb, cap)
    ctx.save_for_backward(gates, c0, c1)
    ctx.cfg = (fgb, cap)
    return c1, m1

  @staticmethod
  def backward(ctx, dc1, dm1):
    ext = _loader.get_ext(required=True)
    gates, c0, c1 = ctx.saved_tensors
    fgb, cap = ctx.cfg
    if dm1 is None:
      dm1 = torch.zeros_like(c0)
    dc1_opt = None if dc1 is None else dc1.contiguous()
    dgates, dc0 = ext.lstm_gates_bwd(gates, c0, c1, dm1.contiguous(),
                                     dc1_opt, fgb, cap)
    return dgates, dc0, None, None


def lstm_gates(gates: torch.Tensor, c0: torch.Tensor,
               forget_gate_bias: float = 0.0,
               cell_value_cap: float = 0.0):
  """gates [B, 4H] bf16, c0 [B, H] -> (c1, m1). cap<=0 disables clamp."""
  return _LstmGatesFn.apply(gates.contiguous(), c0.contiguous(),
                            forget_gate_bias, cell_value_cap)
