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


# Canonical implementations live in core (richer APIs there):
# core/input_generator_helper.py (ComputeSplits/SplitTensors/
# SplitDictOfTensors/SplitNestedMap), core/batch_utils.py
# (world-size-aware scaling), core/gradient_combiner.py (layer-style
# Sum + PCGrad combiners on grads). These wrappers keep the original
# helper names working.
from lingvo_amd.core.input_generator_helper import (  # noqa: F401
    ComputeSplits, SplitNestedMap, SplitTensors)


def ScaleInfeedToGlobal(infeed_batch_size: int,
                        num_replicas: int) -> int:
  """Per-replica -> global batch (reference batch_utils.py)."""
  return infeed_batch_size * num_replicas


def ScaleGlobalToInfeed(global_batch_size: int, num_replicas: int) -> int:
  assert global_batch_size % num_replicas == 0
  return global_batch_size // num_replicas


class GradientCombiner:
  """Loss-level weighted-sum combiner. For gradient-level combination
  (sum / PCGrad on per-loss grads) use core/gradient_combiner.py."""

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
