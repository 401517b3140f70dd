"""Magnitude pruning with a polynomial sparsity schedule.

MI355X-native re-implementation of the reference's pruning hooks
(lingvo/core/pruning_utils.py + model_pruning, wired via
base_model.py:1105 _GetMaskUpdateOp): selected weights get persistent
0/1 masks that are re-thresholded by |w| on a schedule and applied
multiplicatively every step.
"""

from __future__ import annotations

import re
from typing import Dict, List, Optional

import torch


def PolynomialSparsity(step: int, initial_sparsity: float,
                       final_sparsity: float, begin_step: int,
                       end_step: int) -> float:
  """Cubic ramp from initial to final sparsity over [begin, end]
  (reference model_pruning polynomial_decay schedule)."""
  if step <= begin_step:
    return initial_sparsity if step == begin_step else 0.0
  if step >= end_step:
    return final_sparsity
  frac = (step - begin_step) / float(end_step - begin_step)
  return final_sparsity + (initial_sparsity - final_sparsity) * \
      (1.0 - frac) ** 3


class MagnitudePruner:
  """Attaches magnitude-pruning masks to a model's weights.

  usage:
    pruner = MagnitudePruner(model, weight_regex='.*linear.*/w',
                             final_sparsity=0.9, begin_step=100,
                             end_step=1000, frequency=10)
    each train step, after the optimizer update:
      pruner.Prune(global_step)   # updates masks on schedule + applies
  """

  def __init__(self, model: torch.nn.Module, weight_regex: str = '.*',
               initial_sparsity: float = 0.0,
               final_sparsity: float = 0.5,
               begin_step: int = 0, end_step: int = 1000,
               frequency: int = 10, min_numel: int = 256):
    self.cfg = dict(initial_sparsity=initial_sparsity,
                    final_sparsity=final_sparsity,
                    begin_step=begin_step, end_step=end_step)
    self.frequency = max(1, frequency)
    self.begin_step = begin_step
    self._targets: List[tuple] = []
    pat = re.compile(weight_regex)
    for name, param in model.named_parameters():
      if param.requires_grad and param.dim() >= 2 and \
          param.numel() >= min_numel and pat.search(name):
        self._targets.append((name, param,
                              torch.ones_like(param, dtype=torch.bool)))

  @property
  def masks(self) -> Dict[str, torch.Tensor]:
    return {name: mask for name, _, mask in self._targets}

  def CurrentSparsity(self, step: int) -> float:
    return PolynomialSparsity(step, **self.cfg)

  @torch.no_grad()
  def UpdateMasks(self, step: int) -> float:
    """Re-threshold each target's mask at the scheduled sparsity."""
    s = self.CurrentSparsity(step)
    for _, param, mask in self._targets:
      k = int(param.numel() * s)
      if k <= 0:
        mask.fill_(True)
        continue
      flat = param.detach().abs().flatten()
      thresh = flat.kthvalue(k).values
      mask.copy_((flat > thresh).reshape_as(mask))
    return s

  @torch.no_grad()
  def ApplyMasks(self) -> None:
    for _, param, mask in self._targets:
      param.mul_(mask)

  @torch.no_grad()
  def Prune(self, step: int) -> Optional[float]:
    """Per-step entry point: update masks on schedule, always apply."""
    s = None
    if step >= self.begin_step and \
        (step - self.begin_step) % self.frequency == 0:
      s = self.UpdateMasks(step)
    self.ApplyMasks()
    return s

  def MeasuredSparsity(self) -> float:
    zeros = total = 0
    for _, param, _ in self._targets:
      zeros += int((param.detach() == 0).sum())
      total += param.numel()
    return zeros / max(1, total)
