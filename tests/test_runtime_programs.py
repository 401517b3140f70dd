"""Executor/programs, early stop, inference export/predictor tests."""

import re
import json
import os

import pytest
import torch

from lingvo_amd.core import registry


def test_executor_multi_program(tmp_path):
  from lingvo_amd.runtime import program as program_lib
  model_p = registry.GetParams('image.mnist.LeNet5', 'Train')
  model_p.task.random_seed = 3
  sched_p = program_lib.SimpleProgramSchedule.Params()
  sched_p.train_program.steps_per_loop = 3
  ep = program_lib.EvalProgram.Params().Set(name='eval_dev',
                                            steps_per_loop=2)
  ep.Define('cls', program_lib.EvalProgram, 'class')
  sched_p.eval_programs = [ep]
  ex = program_lib.Executor(model_p, str(tmp_path), sched_p,
                            device='cpu', max_steps=6)
  ex.Start()
  assert ex.task.global_step == 6
  with open(tmp_path / 'eval_dev' / 'metrics.jsonl') as f:
    recs = [json.loads(l) for l in f]
  assert len(recs) == 2 and 'loss' in recs[0]
  # checkpoint written
  assert os.path.exists(tmp_path / 'train' / 'checkpoint')


def test_metric_history_and_early_stop(tmp_path):
  from lingvo_amd.core.early_stop import EarlyStop, MetricHistory
  mh = MetricHistory(str(tmp_path), 'eval', 'loss', minimize=True)
  for step, val in [(10, 5.0), (20, 3.0), (30, 3.5), (40, 3.4)]:
    mh.ConditionalAppend(step, val)
  assert mh.BestStep() == 20
  es = EarlyStop(EarlyStop.Params().Set(
      metric_history=mh, window=15, min_steps=0))
  assert not es.Stop(30)
  assert es.Stop(40)   # 40 - 20 > 15
  assert es.Stop(41)   # latched


def test_inference_export_and_predictor(tmp_path):
  from lingvo_amd.runtime.inference import InferenceGraphExporter, Predictor
  model_p = registry.GetParams(
      'asr.librispeech.Librispeech960WpmConformerL', 'Train')
  model_p.task.fprop_dtype = torch.float32
  model_p.task.encoder.Set(num_layers=1, model_dim=64, num_heads=1,
                           kernel_size=4)
  model_p.task.decoder.Set(rnn_cell_dim=32, source_dim=64, emb_dim=16,
                           vocab_size=32)
  model_p.input.Set(batch_size=2, frame_len=32, target_len=6,
                    vocab_size=32)
  model_p.task.random_seed = 4
  path = str(tmp_path / 'inference.pt')
  InferenceGraphExporter.Export(model_p, path)
  pred = Predictor(path, device='cpu')
  assert 'default' in pred.subgraphs and 'encode' in pred.subgraphs
  src = torch.randn(2, 32, 80)
  pad = torch.zeros(2, 32)
  out = pred.Run('default', src_inputs=src, paddings=pad)
  assert out.hyps.shape[0] == 2
  enc = pred.Run('encode', src_inputs=src, paddings=pad)
  assert enc.encoded.shape[-1] == 64


def test_runner_retries_transient_errors(tmp_path):
  """_RunLoop retry policy (reference base_runner.py:399-527)."""
  from lingvo_amd.runtime.runners import BaseRunner
  from lingvo_amd.core import registry

  model_p = registry.GetParams('image.mnist.LeNet5', 'Train')
  r = BaseRunner(model_p, str(tmp_path), 'test', device='cpu',
                 max_retries=3)
  calls = []

  def flaky():
    calls.append(1)
    if len(calls) < 3:
      raise ConnectionError('transient')

  import time as _time
  orig_sleep = _time.sleep
  _time.sleep = lambda s: None  # no backoff wait in tests
  try:
    r._RunLoop(flaky)
  finally:
    _time.sleep = orig_sleep
  assert len(calls) == 3

  # fatal errors are not retried
  def fatal():
    raise FloatingPointError('nan loss')

  with pytest.raises(FloatingPointError):
    r._RunLoop(fatal)

  # retry budget exhausts
  def always():
    raise TimeoutError('down')

  _time.sleep = lambda s: None
  try:
    with pytest.raises(TimeoutError):
      r._RunLoop(always)
  finally:
    _time.sleep = orig_sleep


def test_mlperf_logging_emitted(tmp_path, capsys):
  """ml_perf_log=True emits :::MLLOG lines from the executor loop."""
  from lingvo_amd.runtime import program as program_lib
  mp = registry.GetParams('image.mnist.LeNet5', 'Train')
  mp.task.random_seed = 3
  sched = program_lib.SimpleProgramSchedule.Params()
  sched.train_program.steps_per_loop = 2
  sched.train_program.ml_perf_log = True
  ex = program_lib.Executor(mp, str(tmp_path), sched, device='cpu',
                            max_steps=2)
  ex.Start()
  out = capsys.readouterr().out
  lines = [l for l in out.splitlines() if l.startswith(':::MLLOG')]
  keys = [json.loads(l.split(' ', 1)[1])['key'] for l in lines]
  assert 'run_start' in keys and 'run_stop' in keys
  assert 'block_stop' in keys


def test_inference_server_endpoints(tmp_path):
  """ASGI in-process test of the serving front-end (no sockets)."""
  from lingvo_amd.runtime.inference import InferenceGraphExporter, Predictor
  from lingvo_amd.runtime.server import MakeApp
  from fastapi.testclient import TestClient

  model_p = registry.GetParams('image.mnist.LeNet5', 'Train')
  model_p.task.random_seed = 9
  path = str(tmp_path / 'inference.pt')
  InferenceGraphExporter.Export(model_p, path)
  app = MakeApp(Predictor(path, device='cpu'))
  client = TestClient(app)

  r = client.get('/health')
  assert r.status_code == 200 and r.json()['status'] == 'ok'
  subgraphs = r.json()['subgraphs']
  assert 'default' in subgraphs

  imgs = torch.zeros(2, 28, 28, 1).tolist()
  r = client.post('/predict/default', json={'images': imgs})
  assert r.status_code == 200, r.text
  out = r.json()
  key = 'logits' if 'logits' in out else sorted(out)[0]
  assert len(out[key]) == 2

  r = client.post('/predict/nope', json={})
  assert r.status_code == 404


def test_trial_early_stop_in_executor(tmp_path):
  from lingvo_amd.runtime import program as program_lib
  from lingvo_amd.utils import helpers

  class StopAfterOne(helpers.Trial):
    def __init__(self):
      self.reports = []
    def ReportEvalMeasure(self, step, metrics, ckpt):
      self.reports.append((step, metrics))
      return False
    def ShouldStop(self):
      return len(self.reports) >= 1

  mp = registry.GetParams('image.mnist.LeNet5', 'Train')
  mp.task.random_seed = 3
  sched = program_lib.SimpleProgramSchedule.Params()
  sched.train_program.steps_per_loop = 2
  trial = StopAfterOne()
  ex = program_lib.Executor(mp, str(tmp_path), sched, device='cpu',
                            max_steps=100, trial=trial)
  ex.Start()
  # stopped by the trial after one loop, far before max_steps
  assert ex.task.global_step == 2
  assert trial.reports and 'loss' in trial.reports[0][1]


def test_server_micro_batching(tmp_path):
  """Concurrent single-example requests coalesce into stacked calls."""
  import threading
  from lingvo_amd.runtime.inference import InferenceGraphExporter, Predictor
  from lingvo_amd.runtime.server import MakeApp
  from fastapi.testclient import TestClient

  model_p = registry.GetParams('image.mnist.LeNet5', 'Train')
  model_p.task.random_seed = 9
  path = str(tmp_path / 'inference.pt')
  InferenceGraphExporter.Export(model_p, path)
  app = MakeApp(Predictor(path, device='cpu'), micro_batch=True,
                max_wait_ms=30.0)
  client = TestClient(app)

  img = torch.zeros(1, 28, 28, 1).tolist()
  ref = client.post('/predict/default', json={'images': img}).json()

  results = [None] * 6
  def call(i):
    results[i] = client.post('/predict/default',
                             json={'images': img}).json()
  threads = [threading.Thread(target=call, args=(i,)) for i in range(6)]
  for t in threads:
    t.start()
  for t in threads:
    t.join(30)
  for r in results:
    assert r is not None and r['label'] == ref['label']
  b = app.state.batcher
  assert b.examples_run >= 7
  assert b.batches_run < b.examples_run  # at least one coalesced batch


def test_speculative_decoding_exact_and_fewer_target_calls():
  from lingvo_amd.models import lm as lm_lib
  from lingvo_amd.runtime import speculative
  V = 24
  def mk(seed, layers):
    p = lm_lib.TransformerLm.Params().Set(
        name='lm', vocab_size=V, model_dim=16, num_layers=layers,
        num_heads=1, hidden_dim=32, dropout_prob=0.0, random_seed=seed)
    m = p.Instantiate()
    m.eval()
    return m
  target = mk(3, 2)
  draft = mk(9, 1)
  g = torch.Generator().manual_seed(1)
  prefix = torch.randint(3, V, (2, 4), generator=g)
  spec = speculative.SpeculativeDecoder(target, target.theta,
                                        draft, draft.theta, lookahead=4)
  out = spec.Generate(prefix, max_new=16)
  ref = speculative.GreedyReference(target, target.theta, prefix, 16)
  n = min(out.ids.shape[1], ref.shape[1])
  assert torch.equal(out.ids[:, :n], ref[:, :n])
  # batched verification: fewer target calls than generated tokens
  assert out.stats['target_calls'] < out.new_tokens
  # a draft equal to the target accepts (nearly) everything
  spec2 = speculative.SpeculativeDecoder(target, target.theta,
                                         target, target.theta,
                                         lookahead=4)
  out2 = spec2.Generate(prefix, max_new=12)
  assert out2.stats['accepted'] == out2.stats['proposed']


def test_runner_retry_policy(tmp_path, monkeypatch):
  """Transient errors retry with backoff; fatal errors re-raise."""
  import time as time_mod
  from lingvo_amd.runtime import runners
  monkeypatch.setattr(time_mod, 'sleep', lambda s: None)
  mp2 = registry.GetParams('image.mnist.LeNet5', 'Train')
  r = runners.BaseRunner(mp2, str(tmp_path), 'test', max_retries=3)

  calls = {'n': 0}
  def flaky():
    calls['n'] += 1
    if calls['n'] < 3:
      raise ConnectionError('transient')
  r._RunLoop(flaky)
  assert calls['n'] == 3

  def always_broken():
    raise ConnectionError('never recovers')
  with pytest.raises(ConnectionError):
    r._RunLoop(always_broken)

  def fatal():
    raise FloatingPointError('nan loss')
  with pytest.raises(FloatingPointError):
    r._RunLoop(fatal)
  # status message recorded the retries
  with open(tmp_path / 'test_status.txt') as f:
    assert 'transient error' in f.read()


def test_checkpoint_gc_keeps_latest_n(tmp_path):
  from lingvo_amd.core import checkpointer as ckpt_lib
  import glob as globlib
  saver = ckpt_lib.Saver(str(tmp_path), keep_latest_n=3)
  for step in range(1, 8):
    saver.Save({'model': {'w': torch.ones(2)}, 'step': step}, step)
  kept = sorted(globlib.glob(str(tmp_path / 'ckpt-*.pt')))
  assert len(kept) == 3
  assert kept[-1].endswith('ckpt-00000007.pt')
  # the state file lists only surviving checkpoints
  with open(tmp_path / 'checkpoint') as f:
    txt = f.read()
  assert 'ckpt-00000007.pt' in txt and 'ckpt-00000001.pt' not in txt
  # non-finite payloads refuse to save
  with pytest.raises(FloatingPointError):
    saver.Save({'model': {'w': torch.tensor([float('nan')])},
                'step': 9}, 9)


def test_async_save_completes(tmp_path):
  from lingvo_amd.core import checkpointer as ckpt_lib
  saver = ckpt_lib.Saver(str(tmp_path), async_save=True)
  saver.Save({'model': {'w': torch.arange(4.0)}, 'step': 1}, 1)
  saver.Sync()
  payload = torch.load(tmp_path / 'ckpt-00000001.pt',
                       weights_only=False)
  assert torch.equal(payload['model']['w'], torch.arange(4.0))


def test_evaler_decoder_poll_loop(tmp_path):
  """Trainer saves -> Evaler/Decoder pick up the checkpoint once and
  record it in the processed ledger (idempotent)."""
  from lingvo_amd.core.checkpointer import Checkpointer
  from lingvo_amd.runtime import runners

  model_p = registry.GetParams('image.mnist.LeNet5', 'Train')
  model_p.task.random_seed = 5
  model = model_p.Instantiate()
  task = model.GetTask()
  task.TrainStep(task.GetInputBatch())
  ck = Checkpointer(Checkpointer.Params(),
                    str(tmp_path / 'train'), model,
                    [l.EnsureOptimizer(task) for l in task.learners])
  ck.Save()

  ev = runners.Evaler(model_p, str(tmp_path), dataset='Dev',
                      run_once=True, max_eval_batches=2, device='cpu')
  ev.Start()
  recs = [json.loads(l)
          for l in open(tmp_path / 'eval_dev' / 'metrics.jsonl')]
  assert len(recs) == 1 and recs[0]['step'] == 1

  dec = runners.Decoder(model_p, str(tmp_path), dataset='Dev',
                        run_once=True, max_eval_batches=1, device='cpu')
  dec.Start()
  drecs = [json.loads(l)
           for l in open(tmp_path / 'decoder_dev' / 'metrics.jsonl')]
  assert len(drecs) == 1

  # re-running against the same checkpoint is a no-op (ledger)
  ev2 = runners.Evaler(model_p, str(tmp_path), dataset='Dev',
                       run_once=True, max_eval_batches=2, device='cpu')
  ev2.Start()
  recs2 = [json.loads(l)
           for l in open(tmp_path / 'eval_dev' / 'metrics.jsonl')]
  assert len(recs2) == 1


def test_decode_program_in_schedule(tmp_path):
  from lingvo_amd.runtime import program as program_lib
  model_p = registry.GetParams('image.mnist.LeNet5', 'Train')
  model_p.task.random_seed = 3
  sched = program_lib.SimpleProgramSchedule.Params()
  sched.train_program.steps_per_loop = 1
  dp = program_lib.DecodeProgram.Params().Set(name='decode_dev',
                                              steps_per_loop=2)
  dp.Define('cls', program_lib.DecodeProgram, 'class')
  sched.eval_programs = [dp]
  ex = program_lib.Executor(model_p, str(tmp_path), sched, device='cpu',
                            max_steps=1)
  ex.Start()
  recs = [json.loads(l)
          for l in open(tmp_path / 'decode_dev' / 'metrics.jsonl')]
  assert recs and 'accuracy' in recs[0] or recs[0].keys()
  assert recs[0]['step'] == 1


def test_export_uses_ema_shadows(tmp_path):
  from lingvo_amd.core.checkpointer import Checkpointer
  from lingvo_amd.runtime.inference import InferenceGraphExporter, Predictor
  model_p = registry.GetParams('image.mnist.LeNet5', 'Train')
  model_p.task.random_seed = 6
  model_p.task.train.ema_decay = 0.5
  model = model_p.Instantiate()
  task = model.GetTask()
  for _ in range(3):
    task.TrainStep(task.GetInputBatch())
  ck = Checkpointer(Checkpointer.Params(), str(tmp_path / 'train'),
                    model, [l.EnsureOptimizer(task) for l in task.learners])
  path = ck.Save()
  bundle = str(tmp_path / 'inference.pt')
  InferenceGraphExporter.Export(model_p, bundle, checkpoint_path=path,
                                use_ema=True)
  pred = Predictor(bundle, device='cpu')
  # exported weights equal the EMA shadows, not the live weights
  shadows = task.ema.StateDict()
  name, shadow = next(iter(shadows.items()))
  live = dict(task.named_parameters())[name].detach()
  exported = dict(pred._model.named_parameters())
  exp_t = next(v for k, v in exported.items() if k.endswith(name))
  assert torch.allclose(exp_t.detach(), shadow)
  assert not torch.allclose(exp_t.detach(), live)


def test_decoder_writes_decode_outputs(tmp_path):
  from lingvo_amd.core.checkpointer import Checkpointer
  from lingvo_amd.runtime import runners
  model_p = registry.GetParams('image.mnist.LeNet5', 'Train')
  model_p.task.random_seed = 5
  model = model_p.Instantiate()
  task = model.GetTask()
  task.TrainStep(task.GetInputBatch())
  ck = Checkpointer(Checkpointer.Params(), str(tmp_path / 'train'),
                    model, [l.EnsureOptimizer(task) for l in task.learners])
  ck.Save()
  dec = runners.Decoder(model_p, str(tmp_path), dataset='Dev',
                        run_once=True, max_eval_batches=1, device='cpu')
  dec.Start()
  import glob as globlib
  outs = globlib.glob(str(tmp_path / 'decoder_dev' / 'decode_out-*.jsonl'))
  assert len(outs) == 1
  recs = [json.loads(l) for l in open(outs[0])]
  assert recs and 'correct_top1' in recs[0]


def test_summarize_metrics_tool(tmp_path):
  import sys
  sys.path.insert(0, 'tools')
  from summarize_metrics import Summarize
  from lingvo_amd.runtime import program as program_lib
  mp2 = registry.GetParams('image.mnist.LeNet5', 'Train')
  mp2.task.random_seed = 3
  sched = program_lib.SimpleProgramSchedule.Params()
  sched.train_program.steps_per_loop = 1
  ex = program_lib.Executor(mp2, str(tmp_path), sched, device='cpu',
                            max_steps=1)
  ex.Start()
  out = Summarize(str(tmp_path))
  assert 'metrics.jsonl' in out and 'loss=' in out


def test_early_stop_and_trial_wiring(tmp_path):
  """Trainer stops when (a) the eval metric history shows no
  improvement for `window` steps, (b) a FileTrial requests a stop."""
  import torch
  from lingvo_amd.core.early_stop import EarlyStop, MetricHistory
  from lingvo_amd.runtime.trial import FileTrial, NoOpTrial
  from lingvo_amd.runtime import runners
  from lingvo_amd.core import registry

  logdir = str(tmp_path / 'log')
  # Simulate an evaler-written history: best at step 1, flat after.
  mh = MetricHistory(logdir, 'eval', 'loss', minimize=True)
  for step, v in [(1, 1.0), (5, 1.5), (9, 2.0)]:
    mh.ConditionalAppend(step, v)
  assert mh.BestStep() == 1

  p = registry.GetParams('image.mnist.LeNet5', 'Train')
  p.task.train.early_stop = EarlyStop.Params().Set(
      metric_name='loss', window=3, min_steps=2)
  tr = runners.Trainer(p, logdir, max_steps=100)
  task = tr.model.GetTask()
  task.global_step_var += 6  # step 6: best=1, 6-1 > window=3 -> stop
  assert tr._ShouldStop(task)

  # Trial stop file
  trial = FileTrial(str(tmp_path / 'trial'))
  p2 = registry.GetParams('image.mnist.LeNet5', 'Train')
  tr2 = runners.Trainer(p2, str(tmp_path / 'log2'), max_steps=100,
                        trial=trial)
  task2 = tr2.model.GetTask()
  assert not tr2._ShouldStop(task2)
  trial.ReportEvalMeasure(3, {'loss': 0.5})
  open(trial._stopfile, 'w').close()
  assert tr2._ShouldStop(task2)
  assert not NoOpTrial().ShouldStop()


def test_error_classification_taxonomy():
  from lingvo_amd.runtime.runners import ClassifyError
  assert ClassifyError(StopIteration()) == 'oor'
  assert ClassifyError(ConnectionError('x')) == 'transient'
  assert ClassifyError(RuntimeError('NCCL communicator was aborted')) \
      == 'transient'
  assert ClassifyError(RuntimeError('HIP out of memory')) == 'fatal'
  assert ClassifyError(RuntimeError('shape mismatch')) == 'fatal'
  assert ClassifyError(ValueError('bad')) == 'fatal'
  assert ClassifyError(FloatingPointError('nan')) == 'fatal'


def test_watchdog_fires_and_pets():
  import time as _t
  from lingvo_amd.runtime.runners import Watchdog
  fired = []
  wd = Watchdog(0.4, 'test', exit_fn=lambda code: fired.append(code))
  for _ in range(4):      # petted: must not fire
    _t.sleep(0.15)
    wd.Pet()
  assert not fired
  _t.sleep(1.2)           # starved: fires with exit code 42
  assert fired == [42]
  wd.Stop()


def test_runloop_oor_is_clean_finish(tmp_path):
  from lingvo_amd.runtime import runners
  from lingvo_amd.core import registry
  p = registry.GetParams('image.mnist.LeNet5', 'Train')
  tr = runners.Trainer(p, str(tmp_path), max_steps=5)
  calls = []

  def loop():
    calls.append(1)
    raise StopIteration  # end of data

  tr._RunLoop(loop)      # returns cleanly, no retry storm
  assert len(calls) == 1


def test_export_metrics_hook(tmp_path):
  from lingvo_amd.runtime import runners
  from lingvo_amd.models import mnist as mnist_model
  p = mnist_model.ModelV1.Params().Set(
      name='m', hidden_dim=8, filter_shapes=[(3, 3, 1, 2)])
  p.softmax.num_classes = 10
  p.input = mnist_model.FakeMnistData.Params().Set(batch_size=4)
  from lingvo_amd.core.base_model import SingleTaskModel
  mp_ = SingleTaskModel.Params(p)
  tr = runners.Trainer(mp_, str(tmp_path), max_steps=2)
  exported = []
  tr.SetExportMetricsFn(lambda **kw: exported.append(kw))
  tr.Start()
  assert exported and 'loss' in exported[0] and 'step' in exported[0]


def test_warm_start_rules_remap_and_restore(tmp_path):
  """init_from_checkpoint_rules: regex-remapped partial restore
  (reference checkpointer init rules)."""
  import torch
  from lingvo_amd.core.checkpointer import Checkpointer
  from lingvo_amd.models import mnist as mnist_model
  from lingvo_amd.core.base_model import SingleTaskModel

  def build(name, seed):
    p = mnist_model.ModelV1.Params().Set(
        name=name, hidden_dim=8, filter_shapes=[(3, 3, 1, 2)],
        random_seed=seed)
    p.softmax.num_classes = 10
    p.input = mnist_model.FakeMnistData.Params().Set(batch_size=4)
    return SingleTaskModel.Params(p).Instantiate()

  src = build('a', 1)
  ckpt_path = tmp_path / 'src.pt'
  torch.save({'model': src.state_dict(), 'step': 7}, ckpt_path)

  dst = build('b', 2)
  # Pick one parameter name; remap identity.
  names = [n for n, _ in dst.named_parameters()]
  target = names[0]
  before = dict(dst.named_parameters())[target].detach().clone()
  cp = Checkpointer.Params()
  cp.init_from_checkpoint_rules = {
      str(ckpt_path): [(re.escape(target), target)]}
  ck = Checkpointer(cp, str(tmp_path / 'train'), dst)
  # Warm-start applies on Restore() when no checkpoint exists yet.
  assert ck.Restore() is None
  after = dict(dst.named_parameters())[target].detach()
  want = dict(src.named_parameters())[target].detach()
  assert torch.allclose(after, want)
  assert not torch.allclose(after, before)
