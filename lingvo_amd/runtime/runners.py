"""Runners: Trainer / Evaler / Decoder / Controller loops.

Reference: lingvo/base_runner.py:39 (`_RunLoop` retry policy :399-527),
lingvo/runners.py (Trainer:192, Evaler:860, Decoder:1105, Controller:70).
Single process per role; on GPU boxes the trainer is launched one process
per GPU via torch.distributed (see lingvo_amd/parallel/ddp.py).
"""

from __future__ import annotations

import json
import os
import time
import traceback
from typing import Callable, List, Optional

import torch

from lingvo_amd.core import py_utils
from lingvo_amd.core.checkpointer import (Checkpointer, LatestCheckpoint,
                                          StepFromPath)
from lingvo_amd.core.nested_map import NestedMap

TRANSIENT_ERRORS = (ConnectionError, TimeoutError, BrokenPipeError, OSError)

# RuntimeError substrings that indicate a transient comm/runtime fault
# (the reference's Aborted/FailedPrecondition/DataLoss/Cancelled taxonomy,
# base_runner.py:399-527, mapped to the RCCL/HIP world).
TRANSIENT_RUNTIME_MARKERS = (
    'NCCL', 'RCCL', 'Connection reset', 'Socket Timeout', 'EOF',
    'Watchdog caught', 'Broken pipe', 'transport error',
)
# Fatal marker strings: retrying cannot help (the reference's
# compile-error class -> os._exit(1) in daemon mode).
FATAL_RUNTIME_MARKERS = (
    'out of memory', 'HIP error: invalid device function',
    'no kernel image',
)


def ClassifyError(e: BaseException) -> str:
  """'transient' | 'fatal' | 'oor' (out-of-range: clean end of data)."""
  if isinstance(e, StopIteration):
    return 'oor'
  if isinstance(e, FloatingPointError):
    return 'fatal'
  msg = str(e)
  if isinstance(e, RuntimeError):
    if any(m.lower() in msg.lower() for m in FATAL_RUNTIME_MARKERS):
      return 'fatal'
    if any(m.lower() in msg.lower() for m in TRANSIENT_RUNTIME_MARKERS):
      return 'transient'
    return 'fatal'
  if isinstance(e, TRANSIENT_ERRORS):
    return 'transient'
  return 'fatal'


class Watchdog:
  """Per-rank hang detector (SURVEY SS5 failure detection): if Pet() is
  not called within `timeout_s`, dumps all python stacks, tears down the
  torch.distributed process group (so RCCL peers unblock instead of
  hanging the collective), and hard-exits with code 42 for an external
  scheduler to restart (the reference daemon-mode exit-code protocol,
  base_runner.py:413-420)."""

  def __init__(self, timeout_s: float, tag: str = 'trainer',
               exit_fn=None):
    import threading
    self._timeout = timeout_s
    self._tag = tag
    self._last = time.monotonic()
    self._stop = False
    self._exit_fn = exit_fn or (lambda code: os._exit(code))
    self._fired = False
    self._thread = threading.Thread(target=self._Loop, daemon=True)
    self._thread.start()

  def Pet(self) -> None:
    self._last = time.monotonic()

  def Stop(self) -> None:
    self._stop = True

  def _Loop(self) -> None:
    import faulthandler
    import sys
    while not self._stop:
      time.sleep(min(5.0, self._timeout / 4))
      if self._stop:
        return
      if time.monotonic() - self._last > self._timeout:
        self._fired = True
        sys.stderr.write(
            f'[watchdog:{self._tag}] no progress for '
            f'{self._timeout:.0f}s; dumping stacks and aborting\n')
        try:
          faulthandler.dump_traceback(file=sys.stderr)
        except Exception:
          pass
        try:
          import torch.distributed as dist
          if dist.is_available() and dist.is_initialized():
            dist.destroy_process_group()
        except Exception:
          pass
        self._exit_fn(42)
        return


class StepRateTracker:
  """steps/sec + examples/sec EMA (reference summary_utils.py:393)."""

  def __init__(self):
    self._last_time = None
    self._last_step = None
    self.steps_per_sec = 0.0
    self.examples_per_sec = 0.0

  def Update(self, step: int, examples: float) -> None:
    now = time.perf_counter()
    if self._last_time is not None and step > self._last_step:
      dt = now - self._last_time
      rate = (step - self._last_step) / max(dt, 1e-9)
      alpha = 0.9 if self.steps_per_sec else 0.0
      self.steps_per_sec = alpha * self.steps_per_sec + (1 - alpha) * rate
      self.examples_per_sec = self.steps_per_sec * examples
    self._last_time = now
    self._last_step = step


class BaseRunner:
  """Owns model/checkpointer; _RunLoop retries transient failures."""

  def __init__(self, model_params, logdir: str, job_name: str,
               device: Optional[str] = None, max_retries: int = 10,
               trial=None):
    self._params = model_params
    self._logdir = logdir
    self._job = job_name
    self._max_retries = max_retries
    self._trial = trial  # lingvo_amd.runtime.trial.Trial or None
    self._device = device or (
        'cuda:0' if torch.cuda.is_available() else 'cpu')
    self._train_dir = os.path.join(logdir, 'train')
    os.makedirs(self._train_dir, exist_ok=True)
    self._model = None
    self._status_path = os.path.join(logdir, f'{job_name}_status.txt')
    # Pluggable metric export (reference base_runner.py:174
    # _ExportMetrics): callable(step=int, **metrics) or None.
    self._export_metrics_fn = None

  def SetExportMetricsFn(self, fn) -> None:
    self._export_metrics_fn = fn

  def _ExportMetrics(self, step: int, **metrics) -> None:
    if self._export_metrics_fn is not None:
      self._export_metrics_fn(step=step, **metrics)

  @property
  def model(self):
    if self._model is None:
      self._model = self._params.Instantiate()
      self._model.to(self._device)
    return self._model

  def _SetStatusMessage(self, msg: str) -> None:
    try:
      with open(self._status_path, 'w') as f:
        f.write(f'{time.strftime("%F %T")} {msg}\n')
    except OSError:
      pass

  def _RunLoop(self, loop_fn: Callable[[], None]) -> None:
    """Retry policy (reference base_runner.py:399-527): transient errors
    retry with backoff; programming errors re-raise immediately."""
    retries = 0
    while True:
      try:
        loop_fn()
        return
      except Exception as e:  # noqa: BLE001 - classified below
        kind = ClassifyError(e)
        if kind == 'oor':
          # End of data in eval/decode: clean finish (reference treats
          # OutOfRange in eval jobs as loop completion).
          self._SetStatusMessage('end of data')
          return
        if kind == 'fatal':
          raise
        retries += 1
        if retries > self._max_retries:
          raise
        wait = min(60.0, 2.0 ** retries)
        self._SetStatusMessage(
            f'transient error, retry {retries} in {wait:.0f}s: {e}')
        time.sleep(wait)

  def Start(self) -> None:
    raise NotImplementedError


class Trainer(BaseRunner):
  """Synchronous training loop (reference runners.py:192)."""

  def __init__(self, model_params, logdir: str, max_steps: Optional[int]
               = None, grad_sync=None,
               watchdog_timeout: Optional[float] = None,
               detect_anomaly: bool = False, **kwargs):
    super().__init__(model_params, logdir, 'trainer', **kwargs)
    self._max_steps = max_steps
    if detect_anomaly:
      # autograd anomaly mode (SURVEY §5 sanitizer mapping): NaN
      # sources raise with the producing op's forward stack.
      torch.autograd.set_detect_anomaly(True)
    self._watchdog = (Watchdog(watchdog_timeout, 'trainer')
                      if watchdog_timeout else None)
    if grad_sync is None:
      # Auto-enable DP grad sync when launched under torchrun.
      import torch.distributed as dist
      if dist.is_available() and dist.is_initialized() and \
          dist.get_world_size() > 1:
        from lingvo_amd.parallel.ddp import GradSync
        grad_sync = GradSync(self.model.GetTask())
    self._grad_sync = grad_sync
    self._tracker = StepRateTracker()
    self._metrics_log = os.path.join(logdir, 'train', 'metrics.jsonl')
    self._tb = None  # lazily created TensorBoard events writer
    self._early_stop = None  # lazily built from p.train.early_stop

  def Start(self) -> None:
    self._RunLoop(self._Loop)

  def _ShouldStop(self, task) -> bool:
    limit = self._max_steps or task.p.train.max_steps
    if limit is not None and task.global_step >= limit:
      return True
    if self._trial is not None and self._trial.ShouldStop():
      self._SetStatusMessage('trial requested stop')
      return True
    es_p = task.p.train.early_stop
    if es_p is not None:
      if self._early_stop is None:
        from lingvo_amd.core.early_stop import EarlyStop, MetricHistory
        es = es_p.Copy()
        es.metric_history = MetricHistory(
            self._logdir, 'eval', es_p.metric_name,
            minimize=es_p.minimize)
        self._early_stop = EarlyStop(es)
      if self._early_stop.Stop(int(task.global_step)):
        self._SetStatusMessage(
            f'early stop at step {int(task.global_step)}')
        return True
    return False

  def _Loop(self) -> None:
    model = self.model
    task = model.GetTask()
    ckpt = Checkpointer(Checkpointer.Params().Set(save_interval_seconds=600),
                        self._train_dir, model,
                        [l.EnsureOptimizer(task) for l in task.learners])
    restored = ckpt.Restore()
    if restored is not None:
      self._SetStatusMessage(f'restored step {restored}')
    model.train()
    finalize = self._grad_sync.Finalize if self._grad_sync else None
    while not self._ShouldStop(task):
      ig = task.input_generator
      if ig is not None and hasattr(ig, 'SetGlobalStep'):
        # Curriculum-style inputs switch stages on the global step.
        ig.SetGlobalStep(int(task.global_step))
      batch = task.GetInputBatch()
      if ig is not None:
        batch = ig.ToDevice(batch, self._device)
      metrics = task.TrainStep(batch, grad_sync_finalize=finalize)
      step = task.global_step
      examples = py_utils.ToScalar(
          metrics.get('num_samples_in_batch', (batch_size_of(batch), 1))[0])
      self._tracker.Update(step, examples)
      if self._watchdog is not None:
        self._watchdog.Pet()
      if step % 10 == 0 or step <= 1:
        loss = py_utils.ToScalar(metrics[task.learners[0].p.loss_name][0])
        self._SetStatusMessage(
            f'step {step} loss {loss:.6f} '
            f'{self._tracker.steps_per_sec:.2f} steps/s '
            f'{self._tracker.examples_per_sec:.1f} ex/s')
        with open(self._metrics_log, 'a') as f:
          f.write(json.dumps({
              'step': step, 'loss': loss,
              'steps_per_sec': round(self._tracker.steps_per_sec, 4)}) + '\n')
        if self._tb is None:
          from lingvo_amd.core.summary_utils import TbEventWriter
          self._tb = TbEventWriter(os.path.join(self._logdir, 'train'))
        self._tb.scalars(
            {'loss': loss,
             'steps_per_sec': self._tracker.steps_per_sec,
             'examples_per_sec': self._tracker.examples_per_sec},
            step)
        self._ExportMetrics(
            int(step), loss=loss,
            steps_per_sec=self._tracker.steps_per_sec,
            examples_per_sec=self._tracker.examples_per_sec)
      ckpt.MaybeSave()
    ckpt.Save()
    ckpt.Sync()


def batch_size_of(batch: NestedMap) -> int:
  for v in batch.Flatten():
    if isinstance(v, torch.Tensor) and v.dim() > 0:
      return v.shape[0]
  return 1


class Controller(BaseRunner):
  """Writes params.txt / model_analysis.txt artifacts
  (reference runners.py:70-187)."""

  def __init__(self, model_params, logdir: str, **kwargs):
    super().__init__(model_params, logdir, 'controller', **kwargs)
    self._control_dir = os.path.join(logdir, 'control')
    os.makedirs(self._control_dir, exist_ok=True)

  def Start(self) -> None:
    with open(os.path.join(self._control_dir, 'params.txt'), 'w') as f:
      f.write(self._params.ToText())
    model = self.model
    lines = []
    total = 0
    for name, prm in model.named_parameters():
      n = prm.numel()
      total += n
      lines.append(f'{name} {tuple(prm.shape)} {n}')
    lines.append(f'total #params: {total}')
    with open(os.path.join(self._control_dir, 'model_analysis.txt'),
              'w') as f:
      f.write('\n'.join(lines) + '\n')


class _CheckpointPoller(BaseRunner):
  """Shared poll-new-checkpoints loop for Evaler/Decoder
  (reference base_runner.py:224 _FindNewCheckpoint)."""

  def __init__(self, model_params, logdir: str, job_name: str,
               dataset: str = 'Dev', run_once: bool = False,
               max_eval_batches: int = 10, **kwargs):
    super().__init__(model_params, logdir, job_name, **kwargs)
    self._dataset = dataset
    self._run_once = run_once
    self._max_eval_batches = max_eval_batches
    self._processed_path = os.path.join(
        logdir, f'{job_name}_{dataset.lower()}', 'processed_ckpts.txt')
    os.makedirs(os.path.dirname(self._processed_path), exist_ok=True)

  def _Processed(self) -> set:
    if not os.path.exists(self._processed_path):
      return set()
    with open(self._processed_path) as f:
      return set(l.strip() for l in f if l.strip())

  def _MarkProcessed(self, path: str) -> None:
    with open(self._processed_path, 'a') as f:
      f.write(path + '\n')

  def Start(self) -> None:
    self._RunLoop(self._Loop)

  def _Loop(self) -> None:
    while True:
      path = LatestCheckpoint(self._train_dir)
      if path and path not in self._Processed():
        self._RunOnCheckpoint(path)
        self._MarkProcessed(path)
        if self._run_once:
          return
      elif self._run_once:
        if path:
          return
        time.sleep(0.5)
      else:
        time.sleep(5.0)

  def _RunOnCheckpoint(self, path: str) -> None:
    raise NotImplementedError


class Evaler(_CheckpointPoller):
  """Polls checkpoints, runs eval batches (reference runners.py:860)."""

  def __init__(self, model_params, logdir: str, **kwargs):
    super().__init__(model_params, logdir, 'evaler', **kwargs)

  def _RunOnCheckpoint(self, path: str) -> None:
    model = self.model
    ckpt = Checkpointer(Checkpointer.Params(), self._train_dir, model)
    step = ckpt.Restore(path)
    task = model.GetTask()
    task.eval()
    agg: List[NestedMap] = []
    import contextlib as _ctx
    ema_scope = (task.ema.AsWeights(task) if task.ema is not None
                 else _ctx.nullcontext())  # eval under EMA shadows
    with ema_scope:
      for _ in range(self._max_eval_batches):
        batch = task.GetInputBatch()
        batch = task.input_generator.ToDevice(batch, self._device)
        agg.append(task.EvalStep(batch))
    avg = py_utils.WeightedAvgOfMetrics(agg)
    out_dir = os.path.join(self._logdir, f'eval_{self._dataset.lower()}')
    os.makedirs(out_dir, exist_ok=True)
    vals = {k: py_utils.ToScalar(v[0]) for k, v in avg.items()}
    with open(os.path.join(out_dir, 'metrics.jsonl'), 'a') as f:
      f.write(json.dumps({'step': step, **vals}) + '\n')
    if not hasattr(self, '_tb') or self._tb is None:
      from lingvo_amd.core.summary_utils import TbEventWriter
      self._tb = TbEventWriter(out_dir)
    self._tb.scalars(vals, step)
    # Early-stop history + trial report (reference early_stop.py:126
    # BestStep wiring + base_trial hooks in runners).
    es_p = task.p.train.early_stop
    if es_p is not None and es_p.metric_name in vals:
      from lingvo_amd.core.early_stop import MetricHistory
      MetricHistory(self._logdir, 'eval', es_p.metric_name,
                    minimize=es_p.minimize).ConditionalAppend(
                        step, vals[es_p.metric_name])
    if self._trial is not None:
      self._trial.ShouldStopAndMaybeReport(step, vals)
    task.train()


def _DecodeOutToRecords(out) -> list:
  """Flattens a Decode() NestedMap into per-example JSON records."""
  import torch
  b = None
  for _, v in out.FlattenItems():
    if isinstance(v, torch.Tensor) and v.dim() >= 1:
      b = v.shape[0]
      break
  if b is None:
    return []
  recs = []
  for i in range(b):
    rec = {}
    for key, v in out.FlattenItems():
      if isinstance(v, torch.Tensor) and v.dim() >= 1 and \
          v.shape[0] == b:
        rec[key] = v[i].tolist()
    recs.append(rec)
  return recs


class Decoder(_CheckpointPoller):
  """Polls checkpoints, runs Decode + decoder metrics
  (reference runners.py:1105)."""

  def __init__(self, model_params, logdir: str, **kwargs):
    super().__init__(model_params, logdir, 'decoder', **kwargs)

  def _RunOnCheckpoint(self, path: str) -> None:
    model = self.model
    ckpt = Checkpointer(Checkpointer.Params(), self._train_dir, model)
    step = ckpt.Restore(path)
    task = model.GetTask()
    task.eval()
    dec_metrics = task.CreateDecoderMetrics()
    out_dir = os.path.join(self._logdir, f'decoder_{self._dataset.lower()}')
    os.makedirs(out_dir, exist_ok=True)
    # decode outputs serialized per example (reference decoder_lib.py
    # decode-output records; jsonl instead of tfrecord)
    out_path = os.path.join(out_dir, f'decode_out-{step:08d}.jsonl')
    with open(out_path, 'w') as outf:
      for _ in range(self._max_eval_batches):
        batch = task.GetInputBatch()
        batch = task.input_generator.ToDevice(batch, self._device)
        out = task.Decode(batch)
        task.PostProcessDecodeOut(out, dec_metrics)
        for rec in _DecodeOutToRecords(out):
          outf.write(json.dumps(rec) + '\n')
    with open(os.path.join(out_dir, 'metrics.jsonl'), 'a') as f:
      f.write(json.dumps(
          {'step': step,
           **{k: v.value for k, v in dec_metrics.items()}}) + '\n')
    task.train()
