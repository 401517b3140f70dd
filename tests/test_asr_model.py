"""ASR model (Conformer) CPU tests + bench harness smoke."""

import json
import subprocess
import sys
import os

import pytest
import torch

from lingvo_amd.core import registry


def _tiny_asr_params():
  model_p = registry.GetParams(
      'asr.librispeech.Librispeech960WpmConformerL', 'Train')
  model_p.task.fprop_dtype = torch.float32
  model_p.task.encoder.Set(num_layers=2, model_dim=64, num_heads=1,
                           kernel_size=8)
  model_p.task.decoder.Set(rnn_cell_dim=32, source_dim=64, emb_dim=16,
                           vocab_size=64)
  model_p.input.Set(batch_size=2, frame_len=64, target_len=8,
                    vocab_size=64)
  model_p.task.random_seed = 7
  return model_p


def test_conformer_asr_train_step_cpu():
  model = _tiny_asr_params().Instantiate()
  task = model.GetTask()
  losses = []
  for _ in range(3):
    batch = task.GetInputBatch()
    metrics = task.TrainStep(batch)
    losses.append(float(metrics['loss'][0]))
  assert all(l == l for l in losses), 'NaN loss'
  assert task.global_step == 3


def test_conformer_asr_decode_cpu():
  model = _tiny_asr_params().Instantiate()
  task = model.GetTask()
  task.eval()
  batch = task.GetInputBatch()
  out = task.Decode(batch)
  assert out.topk_decoded.dim() == 2
  dm = task.CreateDecoderMetrics()
  task.PostProcessDecodeOut(out, dm)
  assert dm.num_samples_in_batch.value == 2


def test_librispeech_registry_params():
  model_p = registry.GetParams(
      'asr.librispeech.Librispeech960WpmConformerL', 'Train')
  assert model_p.task.encoder.num_layers == 17
  assert model_p.task.encoder.model_dim == 512
  assert model_p.task.encoder.kernel_size == 32
  assert model_p.task.fprop_dtype == torch.bfloat16


def test_bench_harness_cpu():
  """bench.py runs end-to-end on CPU and prints the JSON contract line."""
  env = dict(os.environ)
  root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
  res = subprocess.run(
      [sys.executable, 'bench.py', '--steps', '2', '--warmup', '1',
       '--batch', '2'],
      cwd=root, env=env, capture_output=True, text=True, timeout=600)
  assert res.returncode == 0, res.stderr[-2000:]
  line = res.stdout.strip().splitlines()[-1]
  rec = json.loads(line)
  for key in ('metric', 'value', 'unit', 'n_gpus', 'steps', 'warmup',
              'ms_per_step', 'higher_is_better', 'scaling', 'vs_baseline',
              'dtype', 'data', 'config'):
    assert key in rec, key
  assert rec['n_gpus'] == 1
  assert rec['value'] > 0
