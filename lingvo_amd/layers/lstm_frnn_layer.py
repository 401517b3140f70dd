"""Functional LSTM layers with the input projection hoisted out of the
recurrence (reference lingvo/core/lstm_frnn_layer.py:27 LSTMCellExt,
:123 LSTMCellSimpleExt).

The per-step LSTM matmul over [x_t, m_{t-1}] splits into (a) ONE
whole-sequence GEMM `acts @ W_x` computed up front and (b) a small
per-step `m @ W_m` inside the scan. On GPU the gate nonlinearity +
state carry run in the fused K11 kernel, so each scan step is one
small GEMM + one fused kernel instead of ~10 elementwise launches —
this is what makes the LAS biLSTM stack viable without hipGraph
capture (HIP rejects the full-scan graph; see docs/KERNEL_NOTES.md).
"""

from __future__ import annotations

from typing import Optional

import torch

from lingvo_amd.core import py_utils, recurrent
from lingvo_amd.core.base_layer import BaseLayer
from lingvo_amd.core.nested_map import NestedMap
from lingvo_amd.layers import rnn_cell


class LSTMCellSimpleExt(rnn_cell.LSTMCellSimple):
  """LSTMCellSimple + whole-sequence input projection
  (reference lstm_frnn_layer.py:123)."""

  def ProjectInputSequence(self, theta: NestedMap,
                           acts: torch.Tensor) -> torch.Tensor:
    """acts [T, B, D] -> projected gate inputs [T, B, G*H] via one GEMM
    against the input rows of wm (reference :34)."""
    d_in = self.p.num_input_nodes
    return torch.matmul(acts, theta.wm[:d_in])

  def FPropWithProjectedInput(self, theta: NestedMap, state0: NestedMap,
                              inputs: NestedMap) -> NestedMap:
    """Like FProp but inputs.proj already holds acts @ W_x
    (reference :73)."""
    p = self.p
    d_in = p.num_input_nodes
    gates = inputs.proj + torch.matmul(state0.m, theta.wm[d_in:])
    if p.enable_lstm_bias:
      gates = gates + theta.b
    h = self._hidden
    pad = inputs.Get('padding')
    if (gates.is_cuda and gates.dtype == torch.bfloat16 and
        not p.couple_input_forget_gates and not self._proj and
        p.output_nonlinearity and p.zo_prob == 0.0):
      from lingvo_amd.ops import lstm_gates as gate_ops
      c1, m1 = gate_ops.lstm_gates(
          gates, state0.c, p.forget_gate_bias,
          p.cell_value_cap if p.cell_value_cap is not None else 0.0)
    else:
      if p.couple_input_forget_gates:
        i_g, f_gbase, o_g = gates.split([h, h, h], dim=-1)
        c_cand = torch.tanh(i_g)
        f_gate = torch.sigmoid(f_gbase + p.forget_gate_bias)
        i_gate = 1.0 - f_gate
        o_gate = torch.sigmoid(o_g)
      else:
        i_i, i_g, f_g, o_g = gates.split([h, h, h, h], dim=-1)
        c_cand = torch.tanh(i_i)
        i_gate = torch.sigmoid(i_g)
        f_gate = torch.sigmoid(f_g + p.forget_gate_bias)
        o_gate = torch.sigmoid(o_g)
      c1 = f_gate * state0.c + i_gate * c_cand
      if p.cell_value_cap is not None:
        c1 = torch.clamp(c1, -p.cell_value_cap, p.cell_value_cap)
      m1 = o_gate * (torch.tanh(c1) if p.output_nonlinearity else c1)
      if self._proj:
        m1 = torch.matmul(m1, theta.w_proj)
    if pad is not None:
      c1 = c1 * (1 - pad) + state0.c * pad
      m1 = m1 * (1 - pad) + state0.m * pad
    return NestedMap(c=c1, m=m1)


class LstmFRNN(BaseLayer):
  """FRNN over LSTMCellSimpleExt with hoisted input projection
  (reference lstm_frnn_layer.py LSTMFRNN). API mirrors
  rnn_layers.FRNN."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('cell', LSTMCellSimpleExt.Params(), 'Cell params.')
    p.Define('reverse', False, 'Scan right-to-left.')
    p.Define('remat', False, 'Recompute cells in backward.')
    return p

  def __init__(self, params):
    super().__init__(params)
    cp = self.p.cell
    if not issubclass(cp.cls, LSTMCellSimpleExt):
      cp = cp.Copy()
      cp.cls = LSTMCellSimpleExt
    self.CreateChild('cell', cp)

  def FProp(self, theta: NestedMap, inputs: torch.Tensor,
            paddings: Optional[torch.Tensor] = None,
            state0: Optional[NestedMap] = None):
    p = self.p
    b, t, _ = inputs.shape
    x = inputs.transpose(0, 1)
    pad = (paddings.transpose(0, 1).unsqueeze(-1).to(inputs.dtype)
           if paddings is not None else torch.zeros(
               t, b, 1, dtype=inputs.dtype, device=inputs.device))
    if p.reverse:
      x = x.flip(0)
      pad = pad.flip(0)
    if state0 is None:
      state0 = self.cell.InitState(b, inputs.device, inputs.dtype)
    # The big GEMM, once for the whole sequence.
    proj = self.cell.ProjectInputSequence(theta.cell, x)

    def cell_fn(th, state, inp):
      return self.cell.FPropWithProjectedInput(th, state, inp), \
          NestedMap()

    acc, final = recurrent.Recurrent(
        theta.cell, state0, NestedMap(proj=proj, padding=pad), cell_fn,
        remat=p.remat)
    out = acc.m
    if p.reverse:
      out = out.flip(0)
    return out.transpose(0, 1), final


class BidirectionalLstmFRNN(BaseLayer):
  """Concat of forward/backward LstmFRNNs (drop-in for
  rnn_layers.BidirectionalFRNN on LSTM cells)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('fwd', LSTMCellSimpleExt.Params(), 'Forward cell.')
    p.Define('bak', LSTMCellSimpleExt.Params(), 'Backward cell.')
    return p

  def __init__(self, params):
    super().__init__(params)
    self.CreateChild('fwd_rnn', LstmFRNN.Params().Set(cell=self.p.fwd))
    self.CreateChild('bak_rnn', LstmFRNN.Params().Set(cell=self.p.bak,
                                                      reverse=True))

  def FProp(self, theta, inputs, paddings=None):
    out_f, _ = self.fwd_rnn.FProp(theta.fwd_rnn, inputs, paddings)
    out_b, _ = self.bak_rnn.FProp(theta.bak_rnn, inputs, paddings)
    return torch.cat([out_f, out_b], dim=-1)
