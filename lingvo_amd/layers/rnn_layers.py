"""RNN layers built on Recurrent (reference lingvo/core/rnn_layers.py:
RNN/FRNN:69,365, BidirectionalFRNN:487, StackedFRNNLayerByLayer)."""

from __future__ import annotations

from typing import Optional

import torch

from lingvo_amd.core import recurrent
from lingvo_amd.core.base_layer import BaseLayer
from lingvo_amd.core.nested_map import NestedMap
from lingvo_amd.layers import rnn_cell


class FRNN(BaseLayer):
  """Functional unidirectional RNN over [B, T, D] batch-major IO."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('cell', rnn_cell.LSTMCellSimple.Params(), 'Cell params.')
    p.Define('reverse', False, 'Scan right-to-left.')
    p.Define('remat', False, 'Recompute cell in backward.')
    return p

  def __init__(self, params):
    super().__init__(params)
    self.CreateChild('cell', self.p.cell)

  def FProp(self, theta: NestedMap, inputs: torch.Tensor,
            paddings: Optional[torch.Tensor] = None,
            state0: Optional[NestedMap] = None):
    """inputs [B,T,D], paddings [B,T] -> (outputs [B,T,H], final_state)."""
    p = self.p
    b, t, _ = inputs.shape
    x = inputs.transpose(0, 1)  # time-major for the scan
    # Cast paddings to the activation dtype: the cells' padded-state
    # carry (c1*(1-pad) + c0*pad) would otherwise promote the whole
    # recurrent state to fp32 after the first step.
    pad = (paddings.transpose(0, 1).unsqueeze(-1).to(inputs.dtype)
           if paddings is not None else torch.zeros(
               t, b, 1, dtype=inputs.dtype, device=inputs.device))
    if p.reverse:
      x = x.flip(0)
      pad = pad.flip(0)
    if state0 is None:
      state0 = self.cell.InitState(b, inputs.device, inputs.dtype)

    def cell_fn(th, state, inp):
      return self.cell.FProp(th, state, inp), NestedMap()

    acc, final = recurrent.Recurrent(
        theta.cell, state0, NestedMap(act=x, padding=pad), cell_fn,
        remat=p.remat)
    out = acc.m
    if p.reverse:
      out = out.flip(0)
    return out.transpose(0, 1), final


class BidirectionalFRNN(BaseLayer):
  """Concat of forward and backward FRNNs (reference rnn_layers.py:487)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('fwd', rnn_cell.LSTMCellSimple.Params(), 'Forward cell.')
    p.Define('bak', rnn_cell.LSTMCellSimple.Params(), 'Backward cell.')
    return p

  def __init__(self, params):
    super().__init__(params)
    self.CreateChild('fwd_rnn', FRNN.Params().Set(cell=self.p.fwd))
    self.CreateChild('bak_rnn', FRNN.Params().Set(cell=self.p.bak,
                                                  reverse=True))

  def FProp(self, theta, inputs, paddings=None):
    out_f, _ = self.fwd_rnn.FProp(theta.fwd_rnn, inputs, paddings)
    out_b, _ = self.bak_rnn.FProp(theta.bak_rnn, inputs, paddings)
    return torch.cat([out_f, out_b], dim=-1)


class StackedFRNNLayerByLayer(BaseLayer):
  """Stack of FRNNs with optional residual (reference rnn_layers.py)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('cell_tpl', [], 'List of cell params, one per layer.')
    p.Define('skip_start', 2, 'Residual connections from this layer on.')
    return p

  def __init__(self, params):
    super().__init__(params)
    self.CreateChildren('rnn', [
        FRNN.Params().Set(cell=c) for c in self.p.cell_tpl])

  def FProp(self, theta, inputs, paddings=None):
    x = inputs
    for i, layer in enumerate(self.rnn):
      out, _ = layer.FProp(theta.rnn[i], x, paddings)
      if i >= self.p.skip_start and out.shape == x.shape:
        out = out + x
      x = out
    return x
