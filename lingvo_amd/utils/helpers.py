"""Leaf utilities (reference lingvo/core/input_generator_helper.py:21-90
ComputeSplits/SplitTensors, batch_utils.py:21-84 scaling,
gradient_combiner.py:44, ml_perf_log.py:80 mlperf_print,
base_trial.py Trial)."""

from __future__ import annotations

import json
import sys
import time
from typing import List, Optional, Sequence

import torch

from lingvo_amd.core.nested_map import NestedMap


def ComputeSplits(batch_size: int, num_splits: int) -> List[int]:
  """Evenly splits batch_size into num_splits parts
  (reference input_generator_helper.py:21)."""
  assert num_splits >= 1
  base = batch_size // num_splits
  rem = batch_size % num_splits
  return [base + (1 if i < rem else 0) for i in range(num_splits)]


def SplitTensors(tensors: Sequence[torch.Tensor],
                 num_splits: int) -> List[List[torch.Tensor]]:
  """Splits each tensor along dim 0 into num_splits pieces; returns one
  list per split (reference input_generator_helper.py:45)."""
  b = tensors[0].shape[0]
  sizes = ComputeSplits(b, num_splits)
  per_tensor = [list(torch.split(t, sizes)) for t in tensors]
  return [[pt[i] for pt in per_tensor] for i in range(num_splits)]


def SplitNestedMap(batch: NestedMap, num_splits: int) -> List[NestedMap]:
  flat = batch.Flatten()
  splits = SplitTensors([t for t in flat], num_splits)
  return [batch.Pack(s) for s in splits]


def ScaleInfeedToGlobal(infeed_batch_size: int,
                        num_replicas: int) -> int:
  """Per-replica -> global batch (reference batch_utils.py)."""
  return infeed_batch_size * num_replicas


def ScaleGlobalToInfeed(global_batch_size: int, num_replicas: int) -> int:
  assert global_batch_size % num_replicas == 0
  return global_batch_size // num_replicas


class GradientCombiner:
  """Multi-loss gradient combination interface
  (reference gradient_combiner.py:44). Default: weighted sum."""

  def __init__(self, weights: Optional[Sequence[float]] = None):
    self._weights = weights

  def Combine(self, losses: Sequence[torch.Tensor]) -> torch.Tensor:
    w = self._weights or [1.0] * len(losses)
    total = None
    for wi, li in zip(w, losses):
      term = wi * li
      total = term if total is None else total + term
    return total


def mlperf_print(key: str, value=None, metadata: Optional[dict] = None,
                 stream=None) -> None:
  """MLPerf structured log line (reference ml_perf_log.py:80)."""
  rec = {
      'namespace': 'lingvo_amd',
      'time_ms': int(time.time() * 1000),
      'event_type': 'POINT_IN_TIME',
      'key': key,
      'value': value,
      'metadata': metadata or {},
  }
  print(':::MLLOG ' + json.dumps(rec), file=stream or sys.stdout)


class Trial:
  """Hyperparameter-search trial interface (reference base_trial.py).
  The no-op trial never requests stopping; tuner integrations subclass."""

  def Name(self) -> str:
    return 'base_trial'

  def OverrideModelParams(self, model_params):
    return model_params

  def ShouldStop(self) -> bool:
    return False

  def ReportEvalMeasure(self, global_step: int, metrics: dict,
                        checkpoint_path: str) -> bool:
    """Returns True if the trial should stop early."""
    return False

  def ReportDone(self, infeasible: bool = False, reason: str = '') -> bool:
    return False


class NoOpTrial(Trial):
  pass
