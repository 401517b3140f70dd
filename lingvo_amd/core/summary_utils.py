"""Summary / observability utilities (reference lingvo/core/summary_utils.py:
scalar/histogram writers :42-95, StepRateTracker :393, ModelAnalysis :432).

Events are jsonl records per job directory (the MI355X framework's
TB-events equivalent; metrics.jsonl is what runners/programs consume).
"""

from __future__ import annotations

import json
import os
import time
from typing import Dict, Optional

import torch


class SummaryWriter:
  """Append-only jsonl scalar/histogram/text writer."""

  def __init__(self, logdir: str, name: str = 'events'):
    os.makedirs(logdir, exist_ok=True)
    self._path = os.path.join(logdir, f'{name}.jsonl')

  def _Write(self, rec: dict) -> None:
    rec['wall_time'] = time.time()
    with open(self._path, 'a') as f:
      f.write(json.dumps(rec) + '\n')

  def scalar(self, tag: str, value, step: int) -> None:
    if isinstance(value, torch.Tensor):
      value = float(value.detach().cpu())
    self._Write({'kind': 'scalar', 'tag': tag, 'value': value,
                 'step': step})

  def histogram(self, tag: str, values: torch.Tensor, step: int,
                bins: int = 30) -> None:
    v = values.detach().float().cpu().flatten()
    hist = torch.histc(v, bins=bins)
    self._Write({'kind': 'histogram', 'tag': tag, 'step': step,
                 'min': float(v.min()) if v.numel() else 0.0,
                 'max': float(v.max()) if v.numel() else 0.0,
                 'counts': hist.tolist()})

  def text(self, tag: str, value: str, step: int) -> None:
    self._Write({'kind': 'text', 'tag': tag, 'value': value,
                 'step': step})


class StepRateTracker:
  """EMA steps/sec + examples/sec (reference summary_utils.py:393)."""

  def __init__(self):
    self._last_time = None
    self._last_step = None
    self.steps_per_sec = 0.0
    self.examples_per_sec = 0.0

  def Update(self, step: int, examples_per_step: float = 0.0) -> None:
    now = time.perf_counter()
    if self._last_time is not None and step > self._last_step:
      rate = (step - self._last_step) / max(now - self._last_time, 1e-9)
      alpha = 0.9 if self.steps_per_sec else 0.0
      self.steps_per_sec = alpha * self.steps_per_sec + (1 - alpha) * rate
      self.examples_per_sec = self.steps_per_sec * examples_per_step
    self._last_time = now
    self._last_step = step


def ModelAnalysis(model: torch.nn.Module) -> str:
  """Parameter-count report (reference summary_utils.py:432; written to
  model_analysis.txt by the Controller)."""
  lines = []
  total = 0
  for name, prm in model.named_parameters():
    n = prm.numel()
    total += n
    lines.append(f'{name} {tuple(prm.shape)} {n}')
  lines.append(f'total #params: {total}')
  return '\n'.join(lines) + '\n'
