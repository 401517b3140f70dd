"""Early stopping on tracked metrics (reference lingvo/core/early_stop.py:
MetricHistory:24, EarlyStop:126; BestStep C++ op
best_step_op_kernels.cc)."""

from __future__ import annotations

import json
import os
from typing import Optional

from lingvo_amd.core.hyperparams import Params


class MetricHistory:
  """Appends (step, value) records to a jsonl file and finds the best
  step within a tolerance."""

  def __init__(self, logdir: str, jobname: str, metric: str,
               minimize: bool = True, tolerance: float = 0.0):
    self.metric = metric
    self.minimize = minimize
    self.tolerance = tolerance
    self._path = os.path.join(logdir, f'{jobname}.history.jsonl')
    os.makedirs(logdir, exist_ok=True)

  def ConditionalAppend(self, step: int, value: float) -> None:
    with open(self._path, 'a') as f:
      f.write(json.dumps({'step': step, self.metric: value}) + '\n')

  def _Records(self):
    if not os.path.exists(self._path):
      return []
    with open(self._path) as f:
      return [json.loads(l) for l in f if l.strip()]

  def BestStep(self) -> int:
    """Earliest step whose value is within tolerance of the optimum."""
    recs = self._Records()
    if not recs:
      return 0
    vals = [r[self.metric] for r in recs]
    best = min(vals) if self.minimize else max(vals)
    for r in recs:
      v = r[self.metric]
      if (self.minimize and v <= best + self.tolerance) or \
          (not self.minimize and v >= best - self.tolerance):
        return r['step']
    return recs[-1]['step']


class EarlyStop:
  """Stops when no improvement for `window` steps after min_steps."""

  @classmethod
  def Params(cls) -> Params:
    p = Params()
    p.Define('metric_history', None, 'MetricHistory instance or None.')
    p.Define('metric_name', 'loss',
             'Eval metric the runners track into MetricHistory when '
             'wiring EarlyStop from task params.')
    p.Define('minimize', True, 'Whether lower metric is better.')
    p.Define('window', 10000, 'Steps without improvement to stop.')
    p.Define('min_steps', 1000, 'Never stop before this step.')
    p.Define('tolerance', 0.0, 'Improvement tolerance.')
    p.Define('verbose', False, 'Log decisions.')
    return p

  def __init__(self, params: Params):
    self.p = params
    self._stopped = False

  def Stop(self, current_step: int) -> bool:
    p = self.p
    mh: Optional[MetricHistory] = p.metric_history
    if self._stopped or mh is None or current_step < p.min_steps:
      return self._stopped
    best = mh.BestStep()
    if current_step - best > p.window:
      self._stopped = True
    return self._stopped
