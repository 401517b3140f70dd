"""Multi-task sampling schedules (reference lingvo/core/task_scheduler.py)."""

from __future__ import annotations

import random
from typing import List, Tuple

from lingvo_amd.core.base_layer import BaseLayer


class TaskScheduler(BaseLayer):

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('task_probs', [], 'List of (task_name, prob_or_schedule).')
    return p

  def Sample(self, global_step: int) -> str:
    raise NotImplementedError


class ConstantScheduler(TaskScheduler):
  """Samples tasks with fixed probabilities."""

  def __init__(self, params):
    super().__init__(params)
    seed = self.p.random_seed or 1234
    self._rng = random.Random(seed)

  def Sample(self, global_step: int) -> str:
    names = [n for n, _ in self.p.task_probs]
    probs = [float(w) for _, w in self.p.task_probs]
    return self._rng.choices(names, weights=probs, k=1)[0]


class RoundRobinScheduler(TaskScheduler):

  def __init__(self, params):
    super().__init__(params)
    self._i = 0

  def Sample(self, global_step: int) -> str:
    names = sorted(n for n, _ in self.p.task_probs)
    name = names[self._i % len(names)]
    self._i += 1
    return name


class SequentialScheduler(TaskScheduler):
  """Trains task k until its step budget, then moves on. task_probs is a
  list of (task_name, num_steps)."""

  def Sample(self, global_step: int) -> str:
    acc = 0
    for name, steps in self.p.task_probs:
      acc += int(steps)
      if global_step < acc:
        return name
    return self.p.task_probs[-1][0]


class ExponentialScheduler(TaskScheduler):
  """Interpolates task probabilities exponentially over exp_steps."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('alpha_start', [], 'List of (task, prob) at step 0.')
    p.Define('alpha_end', [], 'List of (task, prob) at exp_steps.')
    p.Define('exp_steps', 10000, 'Annealing horizon.')
    return p

  def __init__(self, params):
    super().__init__(params)
    self._rng = random.Random(self.p.random_seed or 1234)

  def Sample(self, global_step: int) -> str:
    p = self.p
    frac = 1.0 - 0.5 ** (global_step / p.exp_steps)
    names = [n for n, _ in p.alpha_start]
    w0 = {n: v for n, v in p.alpha_start}
    w1 = {n: v for n, v in p.alpha_end}
    probs = [w0[n] + (w1[n] - w0[n]) * frac for n in names]
    return self._rng.choices(names, weights=probs, k=1)[0]
