"""Learning-rate schedules (reference lingvo/core/schedule.py)."""

from __future__ import annotations

import math

from lingvo_amd.core.base_layer import BaseLayer
from lingvo_amd.core.hyperparams import InstantiableParams


class BaseSchedule(BaseLayer):

  def Value(self, step: int) -> float:
    raise NotImplementedError

  def FProp(self, theta, step):
    return self.Value(step)


class Constant(BaseSchedule):

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('value', 1.0, 'Constant multiplier.')
    return p

  def Value(self, step: int) -> float:
    return self.p.value


class ContinuousSchedule(BaseSchedule):
  """Exponential decay after a start step (reference schedule.py)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('initial_value', 1.0, 'Initial multiplier.')
    p.Define('start_step', 400_000, 'Decay start step.')
    p.Define('half_life_steps', 100_000, 'Halving period in steps.')
    p.Define('min', 0.01, 'Floor multiplier.')
    return p

  def Value(self, step: int) -> float:
    p = self.p
    decayed = p.initial_value * 0.5 ** (
        max(0, step - p.start_step) / p.half_life_steps)
    return max(p.min, decayed)


class TransformerSchedule(BaseSchedule):
  """warmup * rsqrt decay (reference schedule.py; used by the 1BW LM at
  one_billion_wds.py:256)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('warmup_steps', 4000, 'Linear warmup steps.')
    p.Define('model_dim', 512, 'Model dimension.')
    p.Define('worker_replicas', 1, 'Number of DP replicas.')
    p.Define('decay_end', None, 'If set, freeze the value after this step.')
    return p

  def Value(self, step: int) -> float:
    p = self.p
    if p.decay_end is not None:
      step = min(step, p.decay_end)
    step = max(step, 1)
    return (p.model_dim ** -0.5) * min(
        step ** -0.5, step * p.warmup_steps ** -1.5) / p.worker_replicas


class LinearRampupExponentialDecay(BaseSchedule):

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('warmup', 100, 'Linear warmup steps.')
    p.Define('decay_start', 1000, 'Decay start step.')
    p.Define('decay_end', 10000, 'Decay end step.')
    p.Define('min', 0.01, 'Final multiplier.')
    p.Define('max', 1.0, 'Peak multiplier.')
    return p

  def Value(self, step: int) -> float:
    p = self.p
    if step < p.warmup:
      return p.max * step / max(1, p.warmup)
    if step < p.decay_start:
      return p.max
    if step >= p.decay_end:
      return p.min
    frac = (step - p.decay_start) / max(1, p.decay_end - p.decay_start)
    return p.max * (p.min / p.max) ** frac


class CosineSchedule(BaseSchedule):

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('initial_value', 1.0, 'Initial multiplier.')
    p.Define('final_value', 0.0, 'Final multiplier.')
    p.Define('total_steps', 100_000, 'Steps to reach final value.')
    return p

  def Value(self, step: int) -> float:
    p = self.p
    frac = min(1.0, step / max(1, p.total_steps))
    return p.final_value + 0.5 * (p.initial_value - p.final_value) * (
        1 + math.cos(math.pi * frac))


class PiecewiseConstantSchedule(BaseSchedule):

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('boundaries', [], 'Step boundaries.')
    p.Define('values', [1.0], 'len(boundaries)+1 values.')
    return p

  def Value(self, step: int) -> float:
    p = self.p
    for b, v in zip(p.boundaries, p.values):
      if step < b:
        return v
    return p.values[len(p.boundaries)]


class PolynomialSchedule(BaseSchedule):

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('power', 1.0, 'Polynomial power.')
    p.Define('start', (0, 0.0), '(step, value) at start.')
    p.Define('limit', (1, 1.0), '(step, value) at end.')
    return p

  def Value(self, step: int) -> float:
    p = self.p
    s0, v0 = p.start
    s1, v1 = p.limit
    if step <= s0:
      return v0
    if step >= s1:
      return v1
    frac = (step - s0) / (s1 - s0)
    return v0 + (v1 - v0) * frac ** p.power


class DevBasedSchedule(BaseSchedule):
  """Plateau-based decay (reference schedule.py DevBasedSchedule): the
  eval job reports the dev metric via ReportMetric; the multiplier decays
  by `factor` when the metric has not improved for `window` steps."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('factor', 0.5, 'Decay factor on plateau.')
    p.Define('window', 5000, 'Steps without improvement before decay.')
    p.Define('min_factor', 0.01, 'Multiplier floor.')
    p.Define('metric_minimize', True, 'Lower metric is better.')
    return p

  def __init__(self, params):
    super().__init__(params)
    self._mult = 1.0
    self._best = None
    self._best_step = 0

  def ReportMetric(self, value: float, step: int) -> None:
    p = self.p
    better = (self._best is None or
              (value < self._best if p.metric_minimize
               else value > self._best))
    if better:
      self._best = value
      self._best_step = step
    elif step - self._best_step > p.window:
      self._mult = max(p.min_factor, self._mult * p.factor)
      self._best_step = step

  def Value(self, step: int) -> float:
    return self._mult
