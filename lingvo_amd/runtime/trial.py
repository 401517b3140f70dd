"""Hyperparameter-trial interface (reference lingvo/base_trial.py).

A Trial receives eval measures and can request early termination — the
hook Vizier-style tuners use. Runners accept any object with this
interface; NoOpTrial is the default."""

from __future__ import annotations

import json
import os
import time
from typing import Dict, Optional


class Trial:
  """Interface (reference base_trial.py Trial)."""

  def Name(self) -> str:
    raise NotImplementedError

  def ReportEvalMeasure(self, global_step: int,
                        metrics: Dict[str, float],
                        checkpoint_path: str = '') -> bool:
    """Reports metrics at a step; returns True if the trial should
    stop early (infeasible / pruned)."""
    raise NotImplementedError

  def ReportDone(self, infeasible: bool = False,
                 infeasible_reason: str = '') -> None:
    raise NotImplementedError

  def ShouldStop(self) -> bool:
    raise NotImplementedError

  def ShouldStopAndMaybeReport(self, global_step: int,
                               metrics: Optional[Dict[str, float]]
                               ) -> bool:
    """The runners' single entry point (reference base_trial.py)."""
    if metrics:
      if self.ReportEvalMeasure(global_step, metrics):
        return True
    return self.ShouldStop()


class NoOpTrial(Trial):
  """Training without a tuner (reference base_trial.py NoOpTrial)."""

  def Name(self) -> str:
    return ''

  def ReportEvalMeasure(self, global_step, metrics, checkpoint_path=''):
    return False

  def ReportDone(self, infeasible=False, infeasible_reason=''):
    pass

  def ShouldStop(self) -> bool:
    return False


class FileTrial(Trial):
  """File-backed trial: measures append to <dir>/trial_measures.jsonl;
  an external tuner requests a stop by creating <dir>/trial_stop. The
  process-boundary protocol a sidecar tuner (the reference's Vizier
  integration) drives."""

  def __init__(self, trial_dir: str, name: str = 'trial'):
    self._dir = trial_dir
    self._name = name
    os.makedirs(trial_dir, exist_ok=True)
    self._measures = os.path.join(trial_dir, 'trial_measures.jsonl')
    self._stopfile = os.path.join(trial_dir, 'trial_stop')
    self._donefile = os.path.join(trial_dir, 'trial_done.json')

  def Name(self) -> str:
    return self._name

  def ReportEvalMeasure(self, global_step, metrics, checkpoint_path=''):
    with open(self._measures, 'a') as f:
      f.write(json.dumps({'step': int(global_step),
                          'wall_time': time.time(),
                          'checkpoint': checkpoint_path,
                          **{k: float(v) for k, v in metrics.items()}})
              + '\n')
    return self.ShouldStop()

  def ReportDone(self, infeasible=False, infeasible_reason=''):
    with open(self._donefile, 'w') as f:
      json.dump({'infeasible': infeasible,
                 'reason': infeasible_reason}, f)

  def ShouldStop(self) -> bool:
    return os.path.exists(self._stopfile)
