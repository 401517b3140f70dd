"""Programs + program schedules + executor
(reference lingvo/core/program.py: TrainProgram:441, EvalProgram:995,
DecodeProgram:1229, SimpleProgramSchedule:2329; lingvo/executor.py:161
ExecutorTpu._Loop).

The MI355X executor is a single-process multi-program driver over one
shared model: train K steps (hipGraph-captured when shapes are static),
then run eval/decode programs, checkpoint on cadence, repeat.
"""

from __future__ import annotations

import json
import os
import time
from typing import List, Optional

import torch

from lingvo_amd.core import py_utils
from lingvo_amd.core.base_layer import BaseLayer
from lingvo_amd.core.checkpointer import Checkpointer
from lingvo_amd.core.hyperparams import InstantiableParams, Params
from lingvo_amd.core.nested_map import NestedMap


class BaseProgram:

  @classmethod
  def Params(cls) -> Params:
    p = Params()
    p.Define('name', '', 'Program name.')
    p.Define('steps_per_loop', 100, 'Steps per Run() call.')
    p.Define('dataset_name', 'Train', 'Input dataset.')
    p.Define('ml_perf_log', False,
             'Emit MLPerf structured log lines (reference '
             'ml_perf_log.py via program.py mlperf hooks).')
    return p

  def __init__(self, params: Params, task, logdir: str, device: str):
    self.p = params
    self.task = task
    self.logdir = logdir
    self.device = device
    self._out = os.path.join(logdir, params.name or type(self).__name__)
    os.makedirs(self._out, exist_ok=True)

  def Run(self) -> NestedMap:
    raise NotImplementedError

  def _Log(self, record: dict) -> None:
    with open(os.path.join(self._out, 'metrics.jsonl'), 'a') as f:
      f.write(json.dumps(record) + '\n')


class TrainProgram(BaseProgram):
  """steps_per_loop train steps (reference program.py:441; the on-device
  tpu_training_loop.repeat maps to hipGraph replay of the step)."""

  def __init__(self, params, task, logdir, device, grad_sync=None):
    super().__init__(params, task, logdir, device)
    self._grad_sync = grad_sync
    self._graphed = None

  def _MaybeGraph(self, batch):
    if self._graphed is None and batch is not None and \
        torch.cuda.is_available():
      try:
        from lingvo_amd.runtime.graph_step import GraphedTrainStep
        self._graphed = GraphedTrainStep(self.task, batch,
                                         grad_sync=self._grad_sync)
      except Exception:
        self._graphed = False
    return self._graphed

  def Run(self) -> NestedMap:
    task = self.task
    task.train()
    finalize = self._grad_sync.Finalize if self._grad_sync else None
    metrics = NestedMap()
    for _ in range(self.p.steps_per_loop):
      batch = task.GetInputBatch()
      batch = task.input_generator.ToDevice(batch, self.device)
      graphed = self._MaybeGraph(batch)
      if graphed:
        metrics = graphed.Step(batch)
      else:
        metrics = task.TrainStep(batch, grad_sync_finalize=finalize)
    loss = py_utils.ToScalar(metrics[task.learners[0].p.loss_name][0])
    self._Log({'step': task.global_step, 'loss': loss})
    if self.p.ml_perf_log:
      from lingvo_amd.utils import helpers
      helpers.mlperf_print('block_stop', metadata={
          'first_step': task.global_step - self.p.steps_per_loop + 1,
          'step': task.global_step, 'loss': loss})
    return NestedMap(loss=loss, step=task.global_step)


class EvalProgram(BaseProgram):

  def Run(self) -> NestedMap:
    task = self.task
    task.eval()
    agg: List[NestedMap] = []
    for _ in range(self.p.steps_per_loop):
      batch = task.GetInputBatch()
      batch = task.input_generator.ToDevice(batch, self.device)
      agg.append(task.EvalStep(batch))
    task.train()
    avg = py_utils.WeightedAvgOfMetrics(agg)
    rec = {'step': task.global_step,
           **{k: py_utils.ToScalar(v[0]) for k, v in avg.items()}}
    self._Log(rec)
    return NestedMap(**{k: py_utils.ToScalar(v[0])
                        for k, v in avg.items()})


class DecodeProgram(BaseProgram):

  def Run(self) -> NestedMap:
    task = self.task
    task.eval()
    dec_metrics = task.CreateDecoderMetrics()
    for _ in range(self.p.steps_per_loop):
      batch = task.GetInputBatch()
      batch = task.input_generator.ToDevice(batch, self.device)
      out = task.Decode(batch)
      task.PostProcessDecodeOut(out, dec_metrics)
    task.train()
    rec = {'step': task.global_step,
           **{k: v.value for k, v in dec_metrics.items()}}
    self._Log(rec)
    return NestedMap(**{k: v.value for k, v in dec_metrics.items()})


class SimpleProgramSchedule:
  """train_executions_per_eval x TrainProgram, then eval/decode programs
  (reference program.py:2329)."""

  @classmethod
  def Params(cls) -> Params:
    p = Params()
    p.Define('train_program', TrainProgram.Params(), 'Train program.')
    p.Define('eval_programs', [], 'List of eval/decode program params.')
    p.Define('train_executions_per_eval', 1, 'Train runs per eval round.')
    return p

  def __init__(self, params, task, logdir, device, grad_sync=None):
    self.p = params
    self.train_program = TrainProgram(params.train_program, task, logdir,
                                      device, grad_sync)
    self.eval_programs = []
    for ep in params.eval_programs:
      cls = ep.Get('cls') if 'cls' in ep else EvalProgram
      self.eval_programs.append(cls(ep, task, logdir, device))

  def Run(self) -> NestedMap:
    out = NestedMap()
    for _ in range(self.p.train_executions_per_eval):
      out.train = self.train_program.Run()
    for prog in self.eval_programs:
      out[prog.p.name or 'eval'] = prog.Run()
    self.last_result = out
    return out


class Executor:
  """Single-process multi-program loop (reference executor.py:495):
  checkpoint-save check -> program_schedule.Run -> stop check."""

  def __init__(self, model_params: InstantiableParams, logdir: str,
               schedule_params: Optional[Params] = None,
               device: Optional[str] = None, max_steps: Optional[int]
               = None, grad_sync=None, trial=None):
    """trial: optional utils.helpers.Trial for hyperparameter search —
    eval results are reported and its ShouldStop() ends the run early
    (reference base_trial.py hooks in runners/executor)."""
    self.device = device or ('cuda:0' if torch.cuda.is_available()
                             else 'cpu')
    self.model = model_params.Instantiate().to(self.device)
    self.task = self.model.GetTask()
    self.logdir = logdir
    self.max_steps = max_steps
    sched_p = schedule_params or SimpleProgramSchedule.Params()
    self.schedule = SimpleProgramSchedule(sched_p, self.task, logdir,
                                          self.device, grad_sync)
    from lingvo_amd.utils import helpers
    self.trial = trial or helpers.NoOpTrial()
    self.ckpt = Checkpointer(
        Checkpointer.Params().Set(save_interval_seconds=600),
        os.path.join(logdir, 'train'), self.model,
        [l.EnsureOptimizer(self.task) for l in self.task.learners])

  def Start(self) -> None:
    if self.schedule.p.train_program.ml_perf_log:
      from lingvo_amd.utils import helpers
      helpers.mlperf_print('init_start')
      helpers.mlperf_print('run_start')
    restored = self.ckpt.Restore()
    while True:
      t0 = time.perf_counter()
      self.schedule.Run()
      cycle_secs = time.perf_counter() - t0
      self.ckpt.MaybeSave()
      step = self.task.global_step
      # per-cycle wall time (reference executor_cycle_secs export,
      # executor.py:584-590)
      with open(os.path.join(self.logdir, 'executor_metrics.jsonl'),
                'a') as f:
        f.write(json.dumps({'step': int(step),
                            'executor_cycle_secs': cycle_secs}) + '\n')
      out = getattr(self.schedule, 'last_result', None)
      if out is not None and 'train' in out:
        self.trial.ReportEvalMeasure(step,
                                     {'loss': float(out.train.loss)}, '')
      if self.trial.ShouldStop():
        break
      limit = self.max_steps or self.task.p.train.max_steps
      if limit is not None and step >= limit:
        break
    self.ckpt.Save()
    self.ckpt.Sync()
    if self.schedule.p.train_program.ml_perf_log:
      from lingvo_amd.utils import helpers
      helpers.mlperf_print('run_stop',
                           metadata={'step': self.task.global_step})


class InputBenchmark(BaseProgram):
  """Times the input pipeline alone (reference program.py:2249
  InputBenchmark): steps_per_loop GetInputBatch+ToDevice calls,
  reporting batches/sec."""

  def Run(self) -> NestedMap:
    task = self.task
    t0 = time.perf_counter()
    n = 0
    for _ in range(self.p.steps_per_loop):
      batch = task.GetInputBatch()
      task.input_generator.ToDevice(batch, self.device)
      n += 1
    dt = time.perf_counter() - t0
    rate = n / max(1e-9, dt)
    self._Log({'batches_per_sec': rate, 'steps': n})
    return NestedMap(batches_per_sec=rate)


class MultiTaskProgramSchedule:
  """Per-step task sampling over a MultiTaskModel (reference
  program.py:2319): each Run samples a task via the model's
  task_scheduler and executes that task's train program."""

  def __init__(self, model, logdir: str, device: str,
               steps_per_loop: int = 1, grad_syncs=None):
    self.model = model
    self.programs = {}
    for name in model.task_names:
      tp = TrainProgram.Params().Set(name=f'train_{name}',
                                     steps_per_loop=steps_per_loop)
      self.programs[name] = TrainProgram(
          tp, model.GetTask(name), logdir, device,
          (grad_syncs or {}).get(name))

  def Run(self) -> NestedMap:
    name = self.model.SampleTask()
    out = self.programs[name].Run()
    return NestedMap(task=name, **out)
