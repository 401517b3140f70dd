"""BaseTask / BaseModel / SingleTaskModel / MultiTaskModel.

Reference: lingvo/core/base_model.py (BaseTask:116 with
ComputePredictions:465 / ComputeLoss:486 / FPropTower:544 / BProp:718,
SingleTaskModel:1379, MultiTaskModel:1480). The train step here is eager
torch: FProp builds the loss, Learner.Apply does backward+update; the
runner (lingvo_amd/runtime) drives the loop and hipGraph capture.
"""

from __future__ import annotations

import contextlib

from typing import Dict, List, Optional, Tuple

import torch

from lingvo_amd.core import learner as learner_lib
from lingvo_amd.core import py_utils
from lingvo_amd.core.base_layer import BaseLayer
from lingvo_amd.core.hyperparams import InstantiableParams, Params
from lingvo_amd.core.nested_map import NestedMap


class ExponentialMovingAverage:
  """EMA of trainable variables (reference base_model.py:77-115)."""

  def __init__(self, decay: float):
    self.decay = decay
    self._shadow: Dict[str, torch.Tensor] = {}

  @torch.no_grad()
  def Update(self, named_params) -> None:
    for name, prm in named_params:
      if not prm.requires_grad:
        continue
      if name not in self._shadow:
        self._shadow[name] = prm.detach().clone()
      else:
        self._shadow[name].mul_(self.decay).add_(prm.detach(),
                                                 alpha=1 - self.decay)

  def StateDict(self) -> Dict[str, torch.Tensor]:
    return self._shadow

  def LoadStateDict(self, sd: Dict[str, torch.Tensor]) -> None:
    self._shadow = dict(sd)

  @torch.no_grad()
  def CopyTo(self, module: torch.nn.Module) -> None:
    for name, prm in module.named_parameters():
      if name in self._shadow:
        prm.copy_(self._shadow[name])

  @contextlib.contextmanager
  def AsWeights(self, module: torch.nn.Module):
    """Temporarily swap the EMA shadows into the module (the
    reference's eval/decode-under-EMA behavior), restoring the live
    weights afterwards."""
    saved = {n: p.detach().clone()
             for n, p in module.named_parameters() if n in self._shadow}
    self.CopyTo(module)
    try:
      yield
    finally:
      with torch.no_grad():
        for n, p in module.named_parameters():
          if n in saved:
            p.copy_(saved[n])


class BaseTask(BaseLayer):
  """One trainable task: input + model layers + learner(s)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('input', None, 'Input generator params.')
    tp = Params()
    tp.Define('learner', learner_lib.Learner.Params(),
              'Learner params (or list of).')
    tp.Define('max_steps', None, 'Stop training after this step.')
    tp.Define('ema_decay', 0.0, 'If >0, maintain EMA of weights.')
    tp.Define('bf16_weights', False,
              'Keep model weights in bf16 with fp32 masters inside the '
              'optimizer (MasterAdamW): theta casts become no-ops and '
              'DP all-reduce runs natively on bf16 grads.')
    tp.Define('start_up_delay_steps', 200, 'Unused on MI355X; kept for '
              'config parity.')
    tp.Define('vn_std', 0.0, 'Variational noise std (0 disables).')
    tp.Define('pruner_hparams', None,
              'dict of MagnitudePruner kwargs (reference '
              'base_model.py:1105 _GetMaskUpdateOp model_pruning hook); '
              'None disables magnitude pruning.')
    tp.Define('early_stop', None,
              'EarlyStop params (core/early_stop.py) with '
              'p.metric_name set: the trainer stops when the eval '
              'metric history shows no improvement for p.window steps '
              '(reference early_stop.py:126 wired through runners).')
    p.Define('train', tp, 'Training hyperparameters subtree.')
    ep = Params()
    ep.Define('samples_per_summary', 1000, 'Eval samples per summary.')
    ep.Define('decoder_samples', 0, 'Decode sample count (0 = all).')
    p.Define('eval', ep, 'Eval hyperparameters subtree.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    self.register_buffer('global_step_var',
                         torch.zeros((), dtype=torch.long), persistent=True)
    learners = p.train.learner
    if not isinstance(learners, (list, tuple)):
      learners = [learners]
    self.CreateChildren('learners', list(learners))
    self._ema = (ExponentialMovingAverage(p.train.ema_decay)
                 if p.train.ema_decay else None)
    self.input_generator = None
    if p.input is not None:
      self.input_generator = p.input.Instantiate()
    self._bf16_converted = False

  @property
  def global_step(self) -> int:
    return int(self.global_step_var.item())

  @property
  def ema(self) -> Optional[ExponentialMovingAverage]:
    return self._ema

  # ---- subclass contract -------------------------------------------------
  def ComputePredictions(self, theta: NestedMap,
                         input_batch: NestedMap) -> NestedMap:
    raise NotImplementedError

  def ComputeLoss(self, theta: NestedMap, predictions: NestedMap,
                  input_batch: NestedMap
                  ) -> Tuple[NestedMap, NestedMap]:
    """Returns (metrics {name: (value, weight)}, per_example)."""
    raise NotImplementedError

  def FPropTower(self, theta: NestedMap, input_batch: NestedMap):
    predictions = self.ComputePredictions(theta, input_batch)
    return self.ComputeLoss(theta, predictions, input_batch)

  def FProp(self, theta: NestedMap, input_batch: NestedMap):
    return self.FPropTower(theta, input_batch)

  # ---- training ---------------------------------------------------------
  def MaybeConvertBf16Weights(self) -> None:
    """Converts fp32 params to bf16 in place (p.train.bf16_weights);
    must run before the optimizer is created. Idempotent."""
    if self._bf16_converted or not self.p.train.bf16_weights:
      return
    if self.fprop_dtype != torch.bfloat16:
      return  # tests override fprop_dtype to fp32; keep fp32 weights
    for prm in self.parameters():
      if prm.dtype == torch.float32:
        prm.data = prm.data.to(torch.bfloat16)
    self._bf16_converted = True

  def TrainStep(self, input_batch: NestedMap,
                grad_sync_finalize=None) -> NestedMap:
    """One full train step: forward, backward, update, bookkeeping."""
    self.MaybeConvertBf16Weights()
    step = self.global_step
    with py_utils.StepSeedScope(self.p.random_seed or 1234, step):
      theta = self.theta
      if self.p.train.vn_std:
        theta = py_utils.AddVn(theta, self.p.train.vn_std)
      metrics, _ = self.FProp(theta, input_batch)
      loss_name = self.learners[0].p.loss_name
      loss = metrics[loss_name][0]
      for i, lrn in enumerate(self.learners):
        ln = lrn.p.loss_name
        lmetrics = lrn.Apply(self, metrics[ln][0] if i else loss, step,
                             grad_sync_finalize=grad_sync_finalize,
                             retain_graph=i < len(self.learners) - 1)
        for k, v in lmetrics.items():
          metrics[k if i == 0 else f'{k}_{i}'] = v
    self.global_step_var += 1
    if self.p.train.pruner_hparams is not None:
      if not hasattr(self, '_pruner'):
        from lingvo_amd.core.pruning_utils import MagnitudePruner
        self._pruner = MagnitudePruner(self,
                                       **self.p.train.pruner_hparams)
      self._pruner.Prune(int(self.global_step))
    self.PostTrainingStepUpdate(self.global_step)
    if self._ema is not None:
      self._ema.Update(self.named_parameters())
    return metrics

  def EvalStep(self, input_batch: NestedMap) -> NestedMap:
    was_training = self.training
    self.eval()
    with torch.no_grad():
      with py_utils.StepSeedScope(self.p.random_seed or 1234,
                                  self.global_step):
        metrics, _ = self.FProp(self.theta, input_batch)
    if was_training:
      self.train()
    return metrics

  # ---- decode -----------------------------------------------------------
  def Decode(self, input_batch: NestedMap) -> NestedMap:
    """Returns decode outputs (subclass-specific)."""
    raise NotImplementedError

  def CreateDecoderMetrics(self) -> NestedMap:
    from lingvo_amd.core import metrics as metrics_lib
    return NestedMap(num_samples_in_batch=metrics_lib.AverageMetric())

  def PostProcessDecodeOut(self, decode_out: NestedMap,
                           decode_metrics: NestedMap) -> None:
    if 'num_samples_in_batch' in decode_metrics:
      first = next(iter(decode_out.values())) if decode_out else None
      n = first.shape[0] if isinstance(first, torch.Tensor) else 1
      decode_metrics.num_samples_in_batch.Update(float(n), 1.0)

  def GetInputBatch(self) -> NestedMap:
    assert self.input_generator is not None, 'Task has no input generator'
    return self.input_generator.GetPreprocessedInputBatch()


class BaseModel(BaseLayer):
  """A trainable model: a collection of tasks."""

  @property
  def tasks(self) -> List[BaseTask]:
    raise NotImplementedError

  def GetTask(self, task_name: Optional[str] = None) -> BaseTask:
    raise NotImplementedError

  @property
  def global_step(self) -> int:
    return self.GetTask().global_step


class SingleTaskModel(BaseModel):
  """Model with exactly one task (reference base_model.py:1379)."""

  @classmethod
  def Params(cls, task_params: Optional[InstantiableParams] = None):
    p = super().Params()
    p.Define('task', task_params, 'Task params.')
    p.Define('input', None, 'Input params (propagated to task).')
    p.Define('model_key', '', 'Registry key this model came from.')
    if task_params is not None and not p.name:
      p.name = task_params.name or 'model'
    return p

  def __init__(self, params):
    if params.input is not None and params.task.input is None:
      params = params.Copy()
      params.task.input = params.input
    super().__init__(params)
    self.CreateChild('_task', self.p.task)

  @property
  def tasks(self):
    return [self._task]

  def GetTask(self, task_name: Optional[str] = None) -> BaseTask:
    return self._task


class MultiTaskModel(BaseModel):
  """Multiple tasks with a sampling schedule (reference base_model.py:1480)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('task_params', Params(), 'Params with one sub-Params per task.')
    p.Define('task_probs', Params(), 'Params with one float per task.')
    p.Define('task_schedule', None, 'Task sampling schedule params.')
    p.Define('model_key', '', 'Registry key.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    self._task_names = sorted(name for name, _ in p.task_params.IterParams())
    for name in self._task_names:
      self.CreateChild(f'task_{name}', p.task_params.Get(name))
    from lingvo_amd.core import task_scheduler
    sched_p = p.task_schedule
    if sched_p is None:
      probs = [(n, p.task_probs.Get(n) if n in p.task_probs else 1.0)
               for n in self._task_names]
      sched_p = task_scheduler.ConstantScheduler.Params().Set(
          task_probs=probs)
    self.CreateChild('task_schedule', sched_p)

  @property
  def task_names(self):
    return list(self._task_names)

  @property
  def tasks(self):
    return [getattr(self, f'task_{n}') for n in self._task_names]

  def GetTask(self, task_name: Optional[str] = None) -> BaseTask:
    if task_name is None:
      task_name = self.SampleTask()
    return getattr(self, f'task_{task_name}')

  def SampleTask(self) -> str:
    return self.task_schedule.Sample(self.global_step)

  @property
  def global_step(self) -> int:
    return self.tasks[0].global_step
