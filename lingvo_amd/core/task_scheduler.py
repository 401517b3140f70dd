"""Multi-task sampling schedules (reference lingvo/core/task_scheduler.py)."""

from __future__ import annotations

import random
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


class AdaptiveScheduler(TaskScheduler):
  """Samples tasks proportionally to how far each is from its target
  metric (reference task_scheduler.py AdaptiveScheduler): tasks report
  progress via ReportMetric; probability ~ (gap/target)^pow."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('tasks', [], 'Task names.')
    p.Define('targets', [], 'Target metric value per task.')
    p.Define('pow', 1.0, 'Sharpness exponent.')
    p.Define('minimize', True, 'Metrics improve downward (loss-like).')
    return p

  def __init__(self, params):
    super().__init__(params)
    self._rng = random.Random(self.p.random_seed or 1234)
    self._current = {n: None for n in self.p.tasks}

  def ReportMetric(self, task: str, value: float) -> None:
    self._current[task] = float(value)

  def Sample(self, global_step: int) -> str:
    p = self.p
    weights = []
    for name, target in zip(p.tasks, p.targets):
      cur = self._current.get(name)
      if cur is None:
        gap = 1.0  # unreported tasks sample at full weight
      elif p.minimize:
        gap = max(0.0, (cur - target) / max(1e-8, abs(target)))
      else:
        gap = max(0.0, (target - cur) / max(1e-8, abs(target)))
      weights.append(max(1e-6, gap) ** p.pow)
    return self._rng.choices(p.tasks, weights=weights, k=1)[0]


class PieceWiseScheduler(TaskScheduler):
  """Different sub-scheduler per step range (reference
  task_scheduler.py PieceWiseScheduler)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('schedule_steps', [],
             'List of (sub_scheduler_params, last_step).')
    return p

  def __init__(self, params):
    super().__init__(params)
    self._subs = [(sp.Instantiate(), until)
                  for sp, until in self.p.schedule_steps]

  def Sample(self, global_step: int) -> str:
    for sub, until in self._subs:
      if global_step <= until:
        return sub.Sample(global_step)
    return self._subs[-1][0].Sample(global_step)
