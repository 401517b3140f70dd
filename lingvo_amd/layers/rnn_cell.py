"""RNN cells (reference lingvo/core/rnn_cell.py: LSTMCellSimple:213,
LayerNormalizedLSTMCellLean:1495, GRUCell:2683).

Cell contract (reference rnn_cell.py): `state1 = cell.FProp(theta, state0,
inputs)` where inputs = NestedMap(act=[B, D_in], padding=[B, 1]); padded
steps carry state0 through unchanged.
"""

from __future__ import annotations

from typing import Optional

import torch

from lingvo_amd.core import py_utils
from lingvo_amd.core.base_layer import BaseLayer
from lingvo_amd.core.nested_map import NestedMap
from lingvo_amd.layers import layers as lingvo_layers


class RNNCell(BaseLayer):

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('num_input_nodes', 0, 'Input dim.')
    p.Define('num_output_nodes', 0, 'Output (m) dim.')
    p.Define('reset_cell_state', False, 'Reset state on padding (unused).')
    return p

  def InitState(self, batch: int, device, dtype) -> NestedMap:
    raise NotImplementedError

  def GetOutput(self, state: NestedMap) -> torch.Tensor:
    return state.m


class LSTMCellSimple(RNNCell):
  """LSTM with optional projection, CIFG and forget-gate bias
  (reference rnn_cell.py:213)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('num_hidden_nodes', 0,
             'Hidden (c) dim; 0 = num_output_nodes (no projection).')
    p.Define('forget_gate_bias', 0.0, 'Forget gate bias init.')
    p.Define('couple_input_forget_gates', False, 'CIFG.')
    p.Define('output_nonlinearity', True, 'tanh on output.')
    p.Define('cell_value_cap', 10.0, 'Clip |c| to this (None disables).')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    self._hidden = p.num_hidden_nodes or p.num_output_nodes
    self._proj = p.num_hidden_nodes > 0 and \
        p.num_hidden_nodes != p.num_output_nodes
    num_gates = 3 if p.couple_input_forget_gates else 4
    self._num_gates = num_gates
    in_dim = p.num_input_nodes + p.num_output_nodes
    self.CreateVariable('wm', py_utils.WeightParams(
        [in_dim, num_gates * self._hidden], p.params_init, p.dtype))
    self.CreateVariable('b', py_utils.WeightParams(
        [num_gates * self._hidden], py_utils.WeightInit.Constant(0.0),
        p.dtype))
    if self._proj:
      self.CreateVariable('w_proj', py_utils.WeightParams(
          [self._hidden, p.num_output_nodes], p.params_init, p.dtype))

  def InitState(self, batch, device, dtype) -> NestedMap:
    return NestedMap(
        c=torch.zeros(batch, self._hidden, device=device, dtype=dtype),
        m=torch.zeros(batch, self.p.num_output_nodes, device=device,
                      dtype=dtype))

  def FProp(self, theta: NestedMap, state0: NestedMap,
            inputs: NestedMap) -> NestedMap:
    p = self.p
    xm = torch.cat([inputs.act, state0.m], dim=-1)
    gates = torch.matmul(xm, theta.wm) + theta.b
    h = self._hidden
    if p.couple_input_forget_gates:
      i_g, f_gbase, o_g = gates.split([h, h, h], dim=-1)
      c_candidate = torch.tanh(i_g)
      f_gate = torch.sigmoid(f_gbase + p.forget_gate_bias)
      i_gate = 1.0 - f_gate
      o_gate = torch.sigmoid(o_g)
      c1 = f_gate * state0.c + i_gate * c_candidate
    else:
      i_i, i_g, f_g, o_g = gates.split([h, h, h, h], dim=-1)
      c_candidate = torch.tanh(i_i)
      i_gate = torch.sigmoid(i_g)
      f_gate = torch.sigmoid(f_g + p.forget_gate_bias)
      o_gate = torch.sigmoid(o_g)
      c1 = f_gate * state0.c + i_gate * c_candidate
    if p.cell_value_cap is not None:
      c1 = torch.clamp(c1, -p.cell_value_cap, p.cell_value_cap)
    m1 = o_gate * (torch.tanh(c1) if p.output_nonlinearity else c1)
    if self._proj:
      m1 = torch.matmul(m1, theta.w_proj)
    if 'padding' in inputs and inputs.padding is not None:
      pad = inputs.padding
      c1 = c1 * (1 - pad) + state0.c * pad
      m1 = m1 * (1 - pad) + state0.m * pad
    return NestedMap(c=c1, m=m1)


class LayerNormalizedLSTMCellLean(LSTMCellSimple):
  """LSTM with layer-normalized gates (reference rnn_cell.py:1495)."""

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    self.CreateVariable('ln_scale', py_utils.WeightParams(
        [self._num_gates * self._hidden],
        py_utils.WeightInit.Constant(0.0), p.dtype))

  def FProp(self, theta, state0, inputs):
    p = self.p
    xm = torch.cat([inputs.act, state0.m], dim=-1)
    gates = torch.matmul(xm, theta.wm)
    # Per-gate layer norm.
    g = gates.float().reshape(gates.shape[0], self._num_gates, self._hidden)
    mean = g.mean(-1, keepdim=True)
    var = g.var(-1, unbiased=False, keepdim=True)
    g = (g - mean) * torch.rsqrt(var + 1e-6)
    g = g.reshape(gates.shape[0], -1) * (1.0 + theta.ln_scale.float())
    gates = g.to(gates.dtype) + theta.b
    h = self._hidden
    i_i, i_g, f_g, o_g = gates.split([h, h, h, h], dim=-1)
    c1 = torch.sigmoid(f_g + p.forget_gate_bias) * state0.c + \
        torch.sigmoid(i_g) * torch.tanh(i_i)
    if p.cell_value_cap is not None:
      c1 = torch.clamp(c1, -p.cell_value_cap, p.cell_value_cap)
    m1 = torch.sigmoid(o_g) * torch.tanh(c1)
    if self._proj:
      m1 = torch.matmul(m1, theta.w_proj)
    if 'padding' in inputs and inputs.padding is not None:
      pad = inputs.padding
      c1 = c1 * (1 - pad) + state0.c * pad
      m1 = m1 * (1 - pad) + state0.m * pad
    return NestedMap(c=c1, m=m1)


class GRUCell(RNNCell):
  """GRU (reference rnn_cell.py:2683)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    d_in = p.num_input_nodes + p.num_output_nodes
    h = p.num_output_nodes
    self.CreateVariable('w_rz', py_utils.WeightParams(
        [d_in, 2 * h], p.params_init, p.dtype))
    self.CreateVariable('b_rz', py_utils.WeightParams(
        [2 * h], py_utils.WeightInit.Constant(0.0), p.dtype))
    self.CreateVariable('w_h', py_utils.WeightParams(
        [d_in, h], p.params_init, p.dtype))
    self.CreateVariable('b_h', py_utils.WeightParams(
        [h], py_utils.WeightInit.Constant(0.0), p.dtype))

  def InitState(self, batch, device, dtype) -> NestedMap:
    return NestedMap(m=torch.zeros(batch, self.p.num_output_nodes,
                                   device=device, dtype=dtype))

  def FProp(self, theta, state0, inputs):
    xm = torch.cat([inputs.act, state0.m], dim=-1)
    rz = torch.sigmoid(torch.matmul(xm, theta.w_rz) + theta.b_rz)
    r, z = rz.chunk(2, dim=-1)
    xh = torch.cat([inputs.act, r * state0.m], dim=-1)
    h_cand = torch.tanh(torch.matmul(xh, theta.w_h) + theta.b_h)
    m1 = (1 - z) * h_cand + z * state0.m
    if 'padding' in inputs and inputs.padding is not None:
      m1 = m1 * (1 - inputs.padding) + state0.m * inputs.padding
    return NestedMap(m=m1)


class SRUCell(RNNCell):
  """Simple Recurrent Unit (reference rnn_cell.py:2174): the matmuls
  depend only on the input, so they batch across time; only cheap
  elementwise recurrences remain sequential."""

  @classmethod
  def Params(cls):
    p = super().Params()
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    h = p.num_output_nodes
    self.CreateVariable('w', py_utils.WeightParams(
        [p.num_input_nodes, 4 * h], p.params_init, p.dtype))
    self.CreateVariable('b', py_utils.WeightParams(
        [4 * h], py_utils.WeightInit.Constant(0.0), p.dtype))

  def InitState(self, batch, device, dtype) -> NestedMap:
    h = self.p.num_output_nodes
    return NestedMap(c=torch.zeros(batch, h, device=device, dtype=dtype),
                     m=torch.zeros(batch, h, device=device, dtype=dtype))

  def FProp(self, theta: NestedMap, state0: NestedMap,
            inputs: NestedMap) -> NestedMap:
    h = self.p.num_output_nodes
    proj = torch.matmul(inputs.act, theta.w) + theta.b
    x_t, f_t, r_t, x2_t = proj.split([h, h, h, h], dim=-1)
    f = torch.sigmoid(f_t)
    r = torch.sigmoid(r_t)
    c1 = f * state0.c + (1 - f) * x_t
    m1 = r * torch.tanh(c1) + (1 - r) * x2_t
    if 'padding' in inputs and inputs.padding is not None:
      pad = inputs.padding
      c1 = c1 * (1 - pad) + state0.c * pad
      m1 = m1 * (1 - pad) + state0.m * pad
    return NestedMap(c=c1, m=m1)
