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

  def _ZoneOut(self, prev_v: torch.Tensor, cur_v: torch.Tensor,
               padding: Optional[torch.Tensor], zo_prob: float,
               is_eval: bool) -> torch.Tensor:
    """ZoneOut regularization (reference rnn_cell.py:140): with prob
    zo_prob keep the previous value; at eval use the expectation.
    Padded steps always carry the previous value."""
    if zo_prob == 0.0:
      mixed = cur_v
    elif is_eval:
      mixed = zo_prob * prev_v + (1.0 - zo_prob) * cur_v
    else:
      u = py_utils.GraphSafeUniform(cur_v.shape, cur_v.device)
      zo = (u < zo_prob).to(cur_v.dtype)
      mixed = zo * prev_v + (1.0 - zo) * cur_v
    if padding is not None:
      mixed = mixed * (1.0 - padding) + prev_v * padding
    return mixed


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
    p.Define('zo_prob', 0.0, 'ZoneOut prob on c and m (reference :249).')
    p.Define('enable_lstm_bias', True, 'Use the gate bias (reference :251).')
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
    if p.enable_lstm_bias:
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
    gates = torch.matmul(xm, theta.wm)
    if p.enable_lstm_bias:
      gates = gates + theta.b
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
    pad = inputs.Get('padding')
    if p.zo_prob > 0.0:
      c1 = self._ZoneOut(state0.c, c1, pad, p.zo_prob, self.do_eval)
      m1 = self._ZoneOut(state0.m, m1, pad, p.zo_prob, self.do_eval)
    elif pad is not None:
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
    gates = g.to(gates.dtype)
    if self.p.enable_lstm_bias:
      gates = gates + theta.b
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


class WeightNormalizedLSTMCellSimple(LSTMCellSimple):
  """LSTMCellSimple with weight-normalized gate matrix
  (reference rnn_cell.py:1377): wm column j is g_j * v_j / ||v_j||."""

  def __init__(self, params):
    super().__init__(params)
    self.CreateVariable('wm_g', py_utils.WeightParams(
        [self._num_gates * self._hidden],
        py_utils.WeightInit.Constant(1.0), self.p.dtype))

  def FProp(self, theta, state0, inputs):
    theta = theta.DeepCopy()
    w = theta.wm.float()
    norm = w.norm(dim=0, keepdim=True).clamp_min(1e-12)
    theta.wm = (w / norm * theta.wm_g.float()).to(theta.wm.dtype)
    return super().FProp(theta, state0, inputs)


class LSTMCellGrouped(RNNCell):
  """Grouped LSTM (reference rnn_cell.py:735; "Factorization tricks for
  LSTM networks" + ShuffleNet-style shard shuffling). Input and state
  are split into num_groups sub-cells; outputs are concatenated after
  an optional cross-group shard shuffle."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('child_lstm_tpl', LSTMCellSimple.Params(),
             'Template of child LSTM cells.')
    p.Define('num_hidden_nodes', 0, 'Total hidden nodes across groups.')
    p.Define('split_inputs', True,
             'Split inputs across groups (False: each group sees all).')
    p.Define('num_groups', 0, 'Number of groups.')
    p.Define('num_shuffle_shards', 1,
             'If > 1, shards for cross-group output shuffling.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    assert p.num_groups > 0
    assert p.num_output_nodes % p.num_groups == 0
    assert p.num_hidden_nodes % p.num_groups == 0
    if p.split_inputs:
      assert p.num_input_nodes % p.num_groups == 0
    child_in = (p.num_input_nodes // p.num_groups
                if p.split_inputs else p.num_input_nodes)
    out_g = p.num_output_nodes // p.num_groups
    assert out_g % p.num_shuffle_shards == 0
    cells = []
    for i in range(p.num_groups):
      cp = p.child_lstm_tpl.Copy().Set(
          name='group_%d' % i, num_input_nodes=child_in,
          num_output_nodes=out_g,
          num_hidden_nodes=p.num_hidden_nodes // p.num_groups)
      cells.append(cp)
    self.CreateChildren('groups', cells)

  def InitState(self, batch, device, dtype) -> NestedMap:
    return NestedMap(
        groups=[c.InitState(batch, device, dtype) for c in self.groups])

  def GetOutput(self, state: NestedMap) -> torch.Tensor:
    p = self.p
    outs = [c.GetOutput(s) for c, s in zip(self.groups, state.groups)]
    if p.num_shuffle_shards > 1:
      shards = []
      for o in outs:
        shards.extend(o.chunk(p.num_shuffle_shards, dim=-1))
      shards = self._ShuffleShards(shards)
      outs = [torch.cat(shards[i * p.num_shuffle_shards:
                               (i + 1) * p.num_shuffle_shards], dim=-1)
              for i in range(p.num_groups)]
    return torch.cat(outs, dim=-1)

  def _ShuffleShards(self, shards):
    """Reference rnn_cell.py:838 shuffle: output group g takes shard s
    from input group (g + s) % num_groups."""
    p = self.p
    assert len(shards) == p.num_shuffle_shards * p.num_groups
    return [shards[((g + s) % p.num_groups) * p.num_shuffle_shards + s]
            for g in range(p.num_groups)
            for s in range(p.num_shuffle_shards)]

  def FProp(self, theta: NestedMap, state0: NestedMap,
            inputs: NestedMap) -> NestedMap:
    p = self.p
    if p.split_inputs:
      acts = inputs.act.chunk(p.num_groups, dim=-1)
    else:
      acts = [inputs.act] * p.num_groups
    state1 = NestedMap(groups=[])
    for cell, th, s0, act in zip(self.groups, theta.groups, state0.groups,
                                 acts):
      child_inputs = NestedMap(act=act, padding=inputs.Get('padding'))
      state1.groups.append(cell.FProp(th, s0, child_inputs))
    return state1


class DoubleProjectionLSTMCell(RNNCell):
  """Layer-normalized LSTM with input AND output projections
  (reference rnn_cell.py:1838). Per-gate weight variables from the
  projected input; no bias (LN handles shift via ln_scale only)."""

  GATES = ('i_i', 'i_g', 'f_g', 'o_g')

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('num_input_hidden_nodes', 0,
             'Dim of the input projection (must be > 0).')
    p.Define('num_hidden_nodes', 0, 'Hidden (c) dim.')
    p.Define('cell_value_cap', 10.0, 'Clip |c|.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    assert p.num_input_hidden_nodes > 0
    assert p.num_hidden_nodes > 0
    self._hidden = p.num_hidden_nodes
    d_in = p.num_input_nodes + p.num_output_nodes
    self.CreateVariable('w_input_proj', py_utils.WeightParams(
        [d_in, p.num_input_hidden_nodes], p.params_init, p.dtype))
    for g in self.GATES:
      self.CreateVariable('wm_%s' % g, py_utils.WeightParams(
          [p.num_input_hidden_nodes, self._hidden], p.params_init,
          p.dtype))
      self.CreateVariable('ln_scale_%s' % g, py_utils.WeightParams(
          [self._hidden], py_utils.WeightInit.Constant(0.0), p.dtype))
    self.CreateVariable('w_output_proj', py_utils.WeightParams(
        [self._hidden, p.num_output_nodes], p.params_init, p.dtype))

  def InitState(self, batch, device, dtype) -> NestedMap:
    return NestedMap(
        c=torch.zeros(batch, self._hidden, device=device, dtype=dtype),
        m=torch.zeros(batch, self.p.num_output_nodes, device=device,
                      dtype=dtype))

  @staticmethod
  def _LN(x, scale):
    xf = x.float()
    mean = xf.mean(-1, keepdim=True)
    var = xf.var(-1, unbiased=False, keepdim=True)
    return ((xf - mean) * torch.rsqrt(var + 1e-6) *
            (1.0 + scale.float())).to(x.dtype)

  def FProp(self, theta, state0, inputs):
    p = self.p
    xm = torch.cat([inputs.act, state0.m], dim=-1)
    proj = torch.matmul(xm, theta.w_input_proj)
    gates = {g: self._LN(torch.matmul(proj, theta.Get('wm_%s' % g)),
                         theta.Get('ln_scale_%s' % g))
             for g in self.GATES}
    c1 = (torch.sigmoid(gates['f_g']) * state0.c +
          torch.sigmoid(gates['i_g']) * torch.tanh(gates['i_i']))
    if p.cell_value_cap is not None:
      c1 = torch.clamp(c1, -p.cell_value_cap, p.cell_value_cap)
    m1 = torch.sigmoid(gates['o_g']) * torch.tanh(c1)
    m1 = torch.matmul(m1, theta.w_output_proj)
    pad = inputs.Get('padding')
    if pad is not None:
      c1 = c1 * (1 - pad) + state0.c * pad
      m1 = m1 * (1 - pad) + state0.m * pad
    return NestedMap(c=c1, m=m1)


class ConvLSTMCell(RNNCell):
  """Convolutional LSTM (reference rnn_cell.py:2015). State m/c are
  [B, H, W, C_cell]; gates come from a 2D same-pad conv over
  concat([act, m]) along channels."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('inputs_shape', [None, None, None, None],
             '[batch, height, width, channels] of the input.')
    p.Define('cell_shape', [None, None, None, None],
             '[batch, height, width, channels] of the cell state.')
    p.Define('filter_shape', [None, None], 'Conv filter (h, w).')
    p.Define('cell_value_cap', 10.0, 'Clip |c|.')
    p.Define('output_nonlinearity', True, 'tanh on output.')
    p.Define('zo_prob', 0.0, 'ZoneOut prob.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    assert p.inputs_shape[1] == p.cell_shape[1]
    assert p.inputs_shape[2] == p.cell_shape[2]
    in_ch = p.inputs_shape[3] + p.cell_shape[3]
    out_ch = p.cell_shape[3]
    self.CreateVariable('wm', py_utils.WeightParams(
        [p.filter_shape[0], p.filter_shape[1], in_ch, 4 * out_ch],
        p.params_init, p.dtype))
    self.CreateVariable('b', py_utils.WeightParams(
        [4 * out_ch], py_utils.WeightInit.Constant(0.0), p.dtype))

  def InitState(self, batch, device, dtype) -> NestedMap:
    p = self.p
    shape = [batch, p.cell_shape[1], p.cell_shape[2], p.cell_shape[3]]
    return NestedMap(c=torch.zeros(*shape, device=device, dtype=dtype),
                     m=torch.zeros(*shape, device=device, dtype=dtype))

  def FProp(self, theta, state0, inputs):
    p = self.p
    import torch.nn.functional as F
    xm = torch.cat([inputs.act, state0.m], dim=-1)  # [B,H,W,Cin+Cc]
    x = xm.permute(0, 3, 1, 2)
    w = theta.wm.permute(3, 2, 0, 1)  # [4C, Cin+Cc, kh, kw]
    kh, kw = p.filter_shape
    pad = (kh // 2, kw // 2)
    gates = F.conv2d(x, w, theta.b, padding=pad).permute(0, 2, 3, 1)
    i_i, i_g, f_g, o_g = gates.chunk(4, dim=-1)
    c1 = torch.sigmoid(f_g) * state0.c + \
        torch.sigmoid(i_g) * torch.tanh(i_i)
    if p.cell_value_cap is not None:
      c1 = torch.clamp(c1, -p.cell_value_cap, p.cell_value_cap)
    m1 = torch.sigmoid(o_g) * (torch.tanh(c1) if p.output_nonlinearity
                               else c1)
    pad_t = inputs.Get('padding')
    if pad_t is not None:
      pad_t = pad_t.reshape(-1, 1, 1, 1).to(c1.dtype)
    if p.zo_prob > 0.0:
      c1 = self._ZoneOut(state0.c, c1, pad_t, p.zo_prob, self.do_eval)
      m1 = self._ZoneOut(state0.m, m1, pad_t, p.zo_prob, self.do_eval)
    elif pad_t is not None:
      c1 = c1 * (1 - pad_t) + state0.c * pad_t
      m1 = m1 * (1 - pad_t) + state0.m * pad_t
    return NestedMap(c=c1, m=m1)


class QuantizedLSTMCell(RNNCell):
  """Quantization-friendly LSTM (reference rnn_cell.py:900): no bias,
  no forget-gate bias, no output nonlinearity; the cell state is
  clipped by a LinearClippingCapSchedule that narrows over training."""

  @classmethod
  def Params(cls):
    from lingvo_amd.core import quant_utils
    p = super().Params()
    p.Define('cc_schedule',
             quant_utils.LinearClippingCapSchedule.Params(),
             'Clipping cap schedule.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    self.CreateChild('cc_schedule', p.cc_schedule)
    self.CreateVariable('wm', py_utils.WeightParams(
        [p.num_input_nodes + p.num_output_nodes,
         4 * p.num_output_nodes], p.params_init, p.dtype))

  def PostTrainingStepUpdate(self, global_step: int) -> None:
    self.cc_schedule.SetStep(global_step)

  def InitState(self, batch, device, dtype) -> NestedMap:
    h = self.p.num_output_nodes
    return NestedMap(c=torch.zeros(batch, h, device=device, dtype=dtype),
                     m=torch.zeros(batch, h, device=device, dtype=dtype))

  def FProp(self, theta, state0, inputs):
    xm = torch.cat([inputs.act, state0.m], dim=-1)
    gates = torch.matmul(xm, theta.wm)
    i_i, i_g, f_g, o_g = gates.chunk(4, dim=-1)
    c1 = torch.sigmoid(f_g) * state0.c + \
        torch.sigmoid(i_g) * torch.tanh(i_i)
    c1 = self.cc_schedule.ApplyClipping(theta.cc_schedule, c1)
    m1 = torch.sigmoid(o_g) * c1
    pad = inputs.Get('padding')
    if pad is not None:
      c1 = c1 * (1 - pad) + state0.c * pad
      m1 = m1 * (1 - pad) + state0.m * pad
    return NestedMap(c=c1, m=m1)
