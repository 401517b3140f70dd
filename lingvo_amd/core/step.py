"""Step API: single-timestep processing interface
(reference lingvo/core/step.py:40 Step, StackStep, IteratorStep;
core/steps/rnn_steps.py RnnStep/RnnStackStep)."""

from __future__ import annotations

from typing import List, Optional

import torch

from lingvo_amd.core.base_layer import BaseLayer
from lingvo_amd.core.nested_map import NestedMap


class Step(BaseLayer):
  """Contract: ZeroState -> (PrepareExternalInputs) -> FProp per step."""

  def PrepareExternalInputs(self, theta: NestedMap,
                            external_inputs: NestedMap) -> NestedMap:
    return external_inputs or NestedMap()

  def ZeroState(self, theta: NestedMap, prepared: NestedMap,
                batch: int, device, dtype) -> NestedMap:
    return NestedMap()

  def FProp(self, theta: NestedMap, prepared: NestedMap,
            step_inputs: NestedMap, padding: torch.Tensor,
            state0: NestedMap):
    """Returns (output NestedMap, state1)."""
    raise NotImplementedError


class StackStep(Step):
  """Composes sub-steps sequentially (reference step.py StackStep)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('sub', [], 'Sub-step params.')
    p.Define('residual_start', -1,
             'Add residuals from this sub-step on (-1 disables).')
    return p

  def __init__(self, params):
    super().__init__(params)
    self.CreateChildren('steps', [sp.Copy() for sp in self.p.sub])

  def PrepareExternalInputs(self, theta, external_inputs):
    return NestedMap(sub=[
        s.PrepareExternalInputs(theta.steps[i], external_inputs)
        for i, s in enumerate(self.steps)
    ])

  def ZeroState(self, theta, prepared, batch, device, dtype):
    return NestedMap(sub=[
        s.ZeroState(theta.steps[i], prepared.sub[i], batch, device, dtype)
        for i, s in enumerate(self.steps)
    ])

  def FProp(self, theta, prepared, step_inputs, padding, state0):
    state1 = NestedMap(sub=[])
    inp = step_inputs
    for i, s in enumerate(self.steps):
      out, st = s.FProp(theta.steps[i], prepared.sub[i], inp, padding,
                        state0.sub[i])
      if self.p.residual_start >= 0 and i >= self.p.residual_start and \
          'output' in inp and out.output.shape == inp.output.shape:
        out.output = out.output + inp.output
      state1.sub.append(st)
      inp = out
    return inp, state1


class RnnStep(Step):
  """Wraps an RNNCell as a Step (reference steps/rnn_steps.py:21)."""

  @classmethod
  def Params(cls):
    from lingvo_amd.layers import rnn_cell
    p = super().Params()
    p.Define('cell', rnn_cell.LSTMCellSimple.Params(), 'Cell params.')
    return p

  def __init__(self, params):
    super().__init__(params)
    self.CreateChild('cell', self.p.cell)

  def ZeroState(self, theta, prepared, batch, device, dtype):
    return self.cell.InitState(batch, device, dtype)

  def FProp(self, theta, prepared, step_inputs, padding, state0):
    state1 = self.cell.FProp(theta.cell, state0,
                             NestedMap(act=step_inputs.output,
                                       padding=padding))
    return NestedMap(output=state1.m), state1


class IteratorStep(Step):
  """Yields one [B, D] slice of a [B, T, D] tensor per call
  (reference step.py IteratorStep)."""

  def PrepareExternalInputs(self, theta, external_inputs):
    return external_inputs

  def ZeroState(self, theta, prepared, batch, device, dtype):
    return NestedMap(t=0)

  def FProp(self, theta, prepared, step_inputs, padding, state0):
    t = state0.t
    out = prepared.inputs[:, t]
    return NestedMap(output=out), NestedMap(t=t + 1)


class EmbeddingStep(Step):
  """Embeds one id per step (reference steps/embedding_steps.py:23)."""

  @classmethod
  def Params(cls):
    from lingvo_amd.layers import layers as lingvo_layers
    p = super().Params()
    p.Define('emb', lingvo_layers.EmbeddingLayer.Params(),
             'Embedding params.')
    return p

  def __init__(self, params):
    super().__init__(params)
    self.CreateChild('emb', self.p.emb)

  def FProp(self, theta, prepared, step_inputs, padding, state0):
    ids = step_inputs.inputs
    out = self.emb.EmbLookup(theta.emb, ids.long())
    return NestedMap(output=out), state0


class AttentionStep(Step):
  """Computes one attention context per step (reference
  steps/attention_steps.py:23 AttentionStep).

  external_inputs: NestedMap(src=[B,S,D], padding=[B,S],
  context=[B,S,Dc] optional). step_inputs.inputs: query [B, Q].
  Output: NestedMap(context=..., probs=...).
  """

  @classmethod
  def Params(cls):
    from lingvo_amd.layers import attention_legacy
    p = super().Params()
    p.Define('atten', attention_legacy.AdditiveAttention.Params(),
             'Per-step attention params.')
    return p

  def __init__(self, params):
    super().__init__(params)
    self.CreateChild('atten', self.p.atten)

  def PrepareExternalInputs(self, theta, external_inputs):
    packed = self.atten.InitForSourcePacked(
        theta.atten, external_inputs.src,
        external_inputs.Get('context'), external_inputs.padding)
    return NestedMap(packed=packed,
                     source_len=external_inputs.src.shape[1])

  def ZeroState(self, theta, prepared, batch, device, dtype):
    return NestedMap(atten=self.atten.ZeroAttentionState(
        prepared.source_len, batch, device, dtype))

  def FProp(self, theta, prepared, step_inputs, padding, state0):
    ctx, probs, atten_state = self.atten.ComputeContextVector(
        theta.atten, prepared.packed, step_inputs.inputs, state0.atten)
    return (NestedMap(output=ctx, context=ctx, probs=probs),
            NestedMap(atten=atten_state))


class RnnStackStep(Step):
  """Stack of RNN cells stepped jointly, with optional residuals and
  an optional attention context appended to every layer's input
  (reference steps/rnn_steps.py:99 RnnStackStep)."""

  @classmethod
  def Params(cls):
    from lingvo_amd.layers import rnn_cell
    p = super().Params()
    p.Define('cell_tpls', [], 'List of RNN cell params (one per layer).')
    p.Define('residual_start', -1, 'Residual-add from this layer on.')
    return p

  def __init__(self, params):
    super().__init__(params)
    self.CreateChildren('cells', [cp.Copy() for cp in self.p.cell_tpls])

  def ZeroState(self, theta, prepared, batch, device, dtype):
    return NestedMap(cells=[c.InitState(batch, device, dtype)
                            for c in self.cells])

  def FProp(self, theta, prepared, step_inputs, padding, state0):
    x = step_inputs.output
    extra = step_inputs.Get('context')
    state1 = NestedMap(cells=[])
    for i, cell in enumerate(self.cells):
      inp = x if extra is None else torch.cat([x, extra], dim=-1)
      st = cell.FProp(theta.cells[i], state0.cells[i],
                      NestedMap(act=inp, padding=padding))
      out = st.m
      if self.p.residual_start >= 0 and i >= self.p.residual_start and \
          out.shape == x.shape:
        out = out + x
      state1.cells.append(st)
      x = out
    return NestedMap(output=x), state1


class AttentionBlockStep(Step):
  """Attention + query-generating RNN stack, stepped together
  (reference steps/attention_steps.py:171 AttentionBlockStep): the
  previous step's context feeds the RNN stack, whose output queries the
  attention for this step's context."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('atten_step', AttentionStep.Params(), 'Attention step.')
    p.Define('query_step', RnnStackStep.Params(), 'Query generator.')
    return p

  def __init__(self, params):
    super().__init__(params)
    self.CreateChild('atten_step', self.p.atten_step)
    self.CreateChild('query_step', self.p.query_step)

  def PrepareExternalInputs(self, theta, external_inputs):
    return NestedMap(
        atten=self.atten_step.PrepareExternalInputs(
            theta.atten_step, external_inputs))

  def ZeroState(self, theta, prepared, batch, device, dtype):
    ctx_dim = self.atten_step.atten.p.source_dim
    return NestedMap(
        atten=self.atten_step.ZeroState(theta.atten_step, prepared.atten,
                                        batch, device, dtype),
        query=self.query_step.ZeroState(theta.query_step, NestedMap(),
                                        batch, device, dtype),
        context=torch.zeros(batch, ctx_dim, device=device, dtype=dtype))

  def FProp(self, theta, prepared, step_inputs, padding, state0):
    q_out, q_state = self.query_step.FProp(
        theta.query_step, NestedMap(),
        NestedMap(output=step_inputs.output, context=state0.context),
        padding, state0.query)
    a_out, a_state = self.atten_step.FProp(
        theta.atten_step, prepared.atten,
        NestedMap(inputs=q_out.output), padding, state0.atten)
    return (NestedMap(output=a_out.context, query=q_out.output,
                      probs=a_out.probs),
            NestedMap(atten=a_state, query=q_state,
                      context=a_out.context))
