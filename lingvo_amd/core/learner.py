"""Learner: one training program = loss -> gradients -> clipped update.

Reference: lingvo/core/learner.py:31 (`Apply` at :177). Owns the optimizer,
LR schedule, grad clipping, L1/L2 regularization and NaN/grad-norm step
skipping (GradNormTracker, reference layers.py:5590).
"""

from __future__ import annotations

import re
from typing import Callable, List, Optional, Tuple

import torch

from lingvo_amd.core import optimizer as optimizer_lib
from lingvo_amd.core import py_utils
from lingvo_amd.core import schedule as schedule_lib
from lingvo_amd.core.base_layer import BaseLayer
from lingvo_amd.core.nested_map import NestedMap


class Learner(BaseLayer):

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('learning_rate', 1e-3, 'Base learning rate.')
    p.Define('lr_schedule', schedule_lib.Constant.Params(),
             'LR multiplier schedule.')
    p.Define('optimizer', optimizer_lib.Adam.Params(), 'Optimizer params.')
    p.Define('loss_name', 'loss', 'Metric name used as training loss.')
    p.Define('clip_gradient_norm_to_value', 0.0,
             'Global-norm clip; 0 disables.')
    p.Define('clip_gradient_single_norm_to_value', 0.0,
             'Per-tensor norm clip; 0 disables.')
    p.Define('grad_norm_to_clip_to_zero', 0.0,
             'Skip the step entirely if global grad norm exceeds this.')
    p.Define('l2_regularizer_weight', None, 'L2 regularization weight.')
    p.Define('l1_regularizer_weight', None, 'L1 regularization weight.')
    p.Define('bprop_variable_filter', None,
             'Regex: only train matching variable names.')
    p.Define('bprop_variable_exclusion', None,
             'Regex: exclude matching variable names.')
    p.Define('skip_step_on_non_finite', True,
             'Skip update when grads contain NaN/Inf.')
    return p

  def __init__(self, params):
    super().__init__(params)
    self.CreateChild('lr_schedule', self.p.lr_schedule)
    self.CreateChild('opt', self.p.optimizer)
    self._torch_opt: Optional[torch.optim.Optimizer] = None
    self._trainable: List[Tuple[str, torch.nn.Parameter]] = []

  def _SelectVariables(self, task) -> List[Tuple[str, torch.nn.Parameter]]:
    p = self.p
    out = []
    for name, param in task.named_parameters():
      if not param.requires_grad:
        continue
      if p.bprop_variable_filter and not re.search(p.bprop_variable_filter,
                                                   name):
        continue
      if p.bprop_variable_exclusion and re.search(p.bprop_variable_exclusion,
                                                  name):
        continue
      out.append((name, param))
    return out

  def EnsureOptimizer(self, task) -> torch.optim.Optimizer:
    if self._torch_opt is None:
      self._trainable = self._SelectVariables(task)
      self._torch_opt = self.opt.CreateTorchOptimizer(
          [prm for _, prm in self._trainable], self.p.learning_rate)
    return self._torch_opt

  def LearningRate(self, step: int) -> float:
    return self.p.learning_rate * self.lr_schedule.Value(step)

  def Apply(self, task, loss: torch.Tensor, global_step: int,
            grad_sync_finalize: Optional[Callable[[], None]] = None,
            retain_graph: bool = False) -> NestedMap:
    """Backward + clip + update. Returns eval metrics for this learner.

    grad_sync_finalize: callback run after backward, before clipping — the
    DP hook point where bucketed RCCL all-reduce completes.
    """
    p = self.p
    opt = self.EnsureOptimizer(task)
    params = [prm for _, prm in self._trainable]

    if p.l2_regularizer_weight:
      reg = sum((prm.float() ** 2).sum() for name, prm in self._trainable
                if not getattr(prm, '_skip_lp_regularization', False))
      loss = loss + 0.5 * p.l2_regularizer_weight * reg
    if p.l1_regularizer_weight:
      reg = sum(prm.float().abs().sum() for name, prm in self._trainable
                if not getattr(prm, '_skip_lp_regularization', False))
      loss = loss + p.l1_regularizer_weight * reg

    opt.zero_grad(set_to_none=True)
    # retain_graph: with multiple learners over one FProp, every backward
    # but the last must keep the autograd graph alive.
    loss.backward(retain_graph=retain_graph)
    if grad_sync_finalize is not None:
      grad_sync_finalize()

    grads = [prm.grad for prm in params if prm.grad is not None]
    metrics = NestedMap()
    if not grads:
      return metrics
    grad_norm = py_utils.GlobalGradNorm(grads)
    metrics.grad_norm = (grad_norm.detach(), torch.ones(()))

    skip = False
    if p.skip_step_on_non_finite and not bool(torch.isfinite(grad_norm)):
      skip = True
    if p.grad_norm_to_clip_to_zero and float(grad_norm) > \
        p.grad_norm_to_clip_to_zero:
      skip = True
    if skip:
      opt.zero_grad(set_to_none=True)
      metrics.step_skipped = (torch.ones(()), torch.ones(()))
      return metrics

    if p.clip_gradient_norm_to_value:
      scale = p.clip_gradient_norm_to_value / grad_norm.clamp_min(
          p.clip_gradient_norm_to_value)
      torch._foreach_mul_(grads, scale)
    if p.clip_gradient_single_norm_to_value:
      norms = torch._foreach_norm(grads, 2)
      scales = [(p.clip_gradient_single_norm_to_value /
                 n.float().clamp_min(
                     p.clip_gradient_single_norm_to_value)).to(g.dtype)
                for n, g in zip(norms, grads)]
      torch._foreach_mul_(grads, scales)

    lr = self.LearningRate(global_step)
    for group in opt.param_groups:
      group['lr'] = lr
    opt.step()
    metrics.learning_rate = (torch.tensor(lr), torch.ones(()))
    return metrics
