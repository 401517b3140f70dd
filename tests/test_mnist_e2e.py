"""End-to-end: image.mnist.LeNet5 trains on CPU via the trainer CLI
(BASELINE config 1; reference trainer_test.py capability)."""

import json
import os

import pytest
import torch

from lingvo_amd.core import registry
from lingvo_amd.runtime import trainer as trainer_cli


def test_registry_lookup():
  cls = registry.GetClass('image.mnist.LeNet5')
  assert cls is not None
  model_p = registry.GetParams('image.mnist.LeNet5', 'Train')
  assert model_p.input is not None
  model = model_p.Instantiate()
  total = sum(p.numel() for p in model.parameters())
  assert total > 10_000


def test_mnist_loss_decreases(tmp_path):
  logdir = str(tmp_path / 'log')
  trainer_cli.main([
      '--model=image.mnist.LeNet5', f'--logdir={logdir}',
      '--job=trainer_client', '--max_steps=30', '--device=cpu'])
  # checkpoint layout contract
  train_dir = os.path.join(logdir, 'train')
  assert os.path.exists(os.path.join(train_dir, 'checkpoint'))
  cks = [f for f in os.listdir(train_dir) if f.startswith('ckpt-')]
  assert any(f == 'ckpt-00000030.pt' for f in cks), cks
  # control artifacts
  assert os.path.exists(os.path.join(logdir, 'control', 'params.txt'))
  assert os.path.exists(os.path.join(logdir, 'control',
                                     'model_analysis.txt'))
  # loss decreased
  with open(os.path.join(train_dir, 'metrics.jsonl')) as f:
    recs = [json.loads(l) for l in f]
  assert recs[-1]['loss'] < recs[0]['loss']


def test_resume_from_checkpoint(tmp_path):
  logdir = str(tmp_path / 'log')
  trainer_cli.main([
      '--model=image.mnist.LeNet5', f'--logdir={logdir}',
      '--job=trainer', '--max_steps=5', '--device=cpu'])
  trainer_cli.main([
      '--model=image.mnist.LeNet5', f'--logdir={logdir}',
      '--job=trainer', '--max_steps=8', '--device=cpu'])
  train_dir = os.path.join(logdir, 'train')
  cks = sorted(f for f in os.listdir(train_dir) if f.endswith('.pt'))
  assert cks[-1] == 'ckpt-00000008.pt'
  payload = torch.load(os.path.join(train_dir, cks[-1]),
                       map_location='cpu', weights_only=False)
  assert payload['step'] == 8


def test_evaler_and_decoder_run_once(tmp_path):
  logdir = str(tmp_path / 'log')
  trainer_cli.main([
      '--model=image.mnist.LeNet5', f'--logdir={logdir}',
      '--job=trainer', '--max_steps=3', '--device=cpu'])
  trainer_cli.main([
      '--model=image.mnist.LeNet5', f'--logdir={logdir}',
      '--job=evaler_test', '--run_once', '--device=cpu'])
  trainer_cli.main([
      '--model=image.mnist.LeNet5', f'--logdir={logdir}',
      '--job=decoder_test', '--run_once', '--device=cpu'])
  with open(os.path.join(logdir, 'eval_test', 'metrics.jsonl')) as f:
    rec = json.loads(f.readline())
  assert 'loss' in rec and rec['step'] == 3
  with open(os.path.join(logdir, 'decoder_test', 'metrics.jsonl')) as f:
    rec = json.loads(f.readline())
  assert rec['step'] == 3


def test_inspect_modes(tmp_path, capsys):
  trainer_cli.main(['--model=image.mnist.LeNet5',
                    f'--logdir={tmp_path}', '--mode=inspect_params'])
  out = capsys.readouterr().out
  assert 'task.hidden_dim : 300' in out


def test_bf16_master_weight_resume(tmp_path):
  """bf16-weight training resumes exactly (fp32 masters round-trip
  through the checkpoint)."""
  import torch
  from lingvo_amd.core import registry
  from lingvo_amd.core.checkpointer import Checkpointer

  def build():
    p = registry.GetParams('image.mnist.LeNet5', 'Train')
    p.task.random_seed = 9
    p.task.fprop_dtype = torch.bfloat16
    p.task.train.bf16_weights = True
    p.input.batch_size = 4
    return p.Instantiate()

  m1 = build()
  t1 = m1.GetTask()
  for _ in range(3):
    t1.TrainStep(t1.GetInputBatch())
  ck = Checkpointer(Checkpointer.Params(), str(tmp_path), m1,
                    [l.EnsureOptimizer(t1) for l in t1.learners])
  ck.Save()

  m2 = build()
  t2 = m2.GetTask()
  t2.MaybeConvertBf16Weights()
  opts = [l.EnsureOptimizer(t2) for l in t2.learners]
  ck2 = Checkpointer(Checkpointer.Params(), str(tmp_path), m2, opts)
  step = ck2.Restore()
  assert step == 3
  # one more step on both: trajectories identical
  b = t1.GetInputBatch()
  t1.input_generator._batch_count = 3
  t2.input_generator._batch_count = 3
  t1.TrainStep(t1.GetInputBatch())
  t2.TrainStep(t2.GetInputBatch())
  w1 = torch.cat([p.detach().float().reshape(-1)
                  for p in t1.parameters()])
  w2 = torch.cat([p.detach().float().reshape(-1)
                  for p in t2.parameters()])
  assert torch.equal(w1, w2)


def test_saver_gc_and_async(tmp_path):
  """keep_latest_n GC + async save (reference saver.py:139-152)."""
  import torch
  from lingvo_amd.core.checkpointer import (LatestCheckpoint, Saver,
                                            StepFromPath)
  saver = Saver(str(tmp_path), keep_latest_n=2, async_save=True)
  payload = {'model': {'w': torch.ones(3)}, 'step': 0}
  for step in (1, 2, 3, 4):
    payload['step'] = step
    saver.Save(dict(payload), step)
  saver.Sync()
  saver._GC(); saver._WriteStateFile()
  import glob
  cks = sorted(glob.glob(str(tmp_path / 'ckpt-*.pt')))
  assert len(cks) == 2
  latest = LatestCheckpoint(str(tmp_path))
  assert StepFromPath(latest) == 4


def test_saver_rejects_nonfinite(tmp_path):
  import torch
  from lingvo_amd.core.checkpointer import Saver
  saver = Saver(str(tmp_path))
  bad = {'model': {'w': torch.tensor([1.0, float('nan')])}, 'step': 1}
  with pytest.raises(FloatingPointError):
    saver.Save(bad, 1)


def test_pruning_hook_in_train_loop():
  """train.pruner_hparams wires MagnitudePruner into TrainStep."""
  from lingvo_amd.core import registry
  model_p = registry.GetParams('image.mnist.LeNet5', 'Train')
  model_p.task.random_seed = 6
  model_p.task.train.pruner_hparams = dict(
      weight_regex='fc', final_sparsity=0.6, begin_step=0, end_step=4,
      frequency=1, min_numel=64)
  model = model_p.Instantiate()
  task = model.GetTask()
  for _ in range(5):
    task.TrainStep(task.GetInputBatch())
  sp = task._pruner.MeasuredSparsity()
  assert abs(sp - 0.6) < 0.05, sp
