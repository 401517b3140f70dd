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


def test_conformer_layer_streaming_matches_fprop():
  """Causal ConformerLayer chunked StreamStep == full FProp."""
  import torch
  from lingvo_amd.core.nested_map import NestedMap
  from lingvo_amd.layers import conformer as conformer_lib
  p = conformer_lib.ConformerLayer.Params().Set(
      name='c', input_dim=16, atten_num_heads=2, kernel_size=4,
      is_causal=True, conv_norm='layer', atten_left_context=8,
      random_seed=9)
  layer = p.Instantiate()
  layer.eval()
  g = torch.Generator().manual_seed(3)
  x = torch.randn(2, 12, 16, generator=g)
  pad = torch.zeros(2, 12)
  full = layer.FProp(layer.theta, x, pad)
  st = layer.InitStreamState(layer.theta, 2, 12, 'cpu', torch.float32)
  outs = []
  for c0 in range(0, 12, 3):
    o, st = layer.StreamStep(layer.theta, x[:, c0:c0 + 3],
                             pad[:, c0:c0 + 3], st)
    outs.append(o)
  stream = torch.cat(outs, dim=1)
  assert (full - stream).abs().max() < 1e-3, \
      (full - stream).abs().max()


def test_conformer_encoder_stack_streaming():
  """2-block causal encoder stack streams == block-wise full FProp."""
  import torch
  from lingvo_amd.models import asr as asr_lib
  p = asr_lib.ConformerEncoder.Params().Set(
      name='enc', input_dim=8, model_dim=16, num_layers=2, num_heads=2,
      kernel_size=4, dropout_prob=0.0, specaug_tpl=None, random_seed=11)
  p.conformer_tpl.is_causal = True
  p.conformer_tpl.conv_norm = 'layer'
  p.conformer_tpl.atten_left_context = 16
  enc = p.Instantiate()
  enc.eval()
  g = torch.Generator().manual_seed(5)
  feats = torch.randn(2, 8, 16, generator=g)  # post-frontend features
  pad = torch.zeros(2, 8)
  # full pass over the blocks only
  x = feats
  for i, b in enumerate(enc.blocks):
    x = b.FProp(enc.theta.blocks[i], x, pad)
  st = enc.InitStreamState(enc.theta, 2, 8, 'cpu', torch.float32)
  outs = []
  for c0 in range(0, 8, 2):
    o, st = enc.StreamStep(enc.theta, feats[:, c0:c0 + 2],
                           pad[:, c0:c0 + 2], st)
    outs.append(o)
  stream = torch.cat(outs, dim=1)
  assert (x - stream).abs().max() < 1e-3


def test_mel_frontend_streaming_matches_full():
  import torch
  from lingvo_amd.layers import asr_frontend
  fe = asr_frontend.MelAsrFrontend.Params().Set(name='fe').Instantiate()
  g = torch.Generator().manual_seed(8)
  wav = torch.randn(2, 4000, generator=g)
  full, _ = fe.FProp(fe.theta, wav, torch.zeros(2, 4000))
  st = fe.InitStreamState(2)
  outs = []
  for c0 in range(0, 4000, 700):  # uneven chunks vs hop=160
    o, st = fe.StreamStep(fe.theta, wav[:, c0:c0 + 700], st)
    if o.shape[1]:
      outs.append(o)
  stream = torch.cat(outs, dim=1)
  assert stream.shape[1] == full.shape[1], (stream.shape, full.shape)
  assert (stream - full).abs().max() < 1e-4


def test_streaming_recognizer_matches_offline():
  """Audio-chunk streaming == offline encoder + greedy decode."""
  import torch
  from lingvo_amd.core.nested_map import NestedMap
  from lingvo_amd.layers import asr_frontend
  from lingvo_amd.models import asr as asr_lib
  ep = asr_lib.ConformerEncoder.Params().Set(
      name='enc', input_dim=80, model_dim=32, num_layers=2, num_heads=2,
      kernel_size=4, dropout_prob=0.0, specaug_tpl=None, random_seed=13)
  ep.conformer_tpl.is_causal = True
  ep.conformer_tpl.conv_norm = 'layer'
  ep.conformer_tpl.atten_left_context = 64
  mp = asr_lib.AsrModel.Params().Set(name='m', encoder=ep,
                                     random_seed=13)
  mp.decoder.Set(vocab_size=16, emb_dim=8, rnn_cell_dim=16,
                 source_dim=32, dropout_prob=0.0)
  model = mp.Instantiate()
  model.eval()
  fe = asr_frontend.MelAsrFrontend.Params().Set(name='fe').Instantiate()
  g = torch.Generator().manual_seed(3)
  wav = torch.randn(2, 16000, generator=g)

  # offline: frontend -> encoder -> greedy
  with torch.no_grad():
    mel, mel_pad = fe.FProp(fe.theta, wav, torch.zeros(2, 16000))
    enc, enc_pad = model.encoder.FProp(model.theta.encoder, mel, mel_pad)
    hyps_off = model.decoder.GreedyDecode(model.theta.decoder, enc,
                                          enc_pad)

  rec = asr_lib.StreamingRecognizer(model, fe, batch=2)
  for c0 in range(0, 16000, 3000):
    rec.Push(wav[:, c0:c0 + 3000])
  out = rec.Finish()
  assert out.encoded.shape == enc.shape, (out.encoded.shape, enc.shape)
  assert (out.encoded - enc).abs().max() < 1e-3
  assert torch.equal(out.hyps, hyps_off)


def test_shallow_fusion_biases_decode():
  """An LM that strongly prefers one token steers greedy decode."""
  import torch
  from lingvo_amd.models import asr as asr_lib
  from lingvo_amd.models import lm as lm_lib
  V = 16
  dec = asr_lib.AsrDecoder.Params().Set(
      name='d', vocab_size=V, emb_dim=8, rnn_cell_dim=16, source_dim=16,
      num_lstm_layers=1, dropout_prob=0.0, random_seed=3).Instantiate()
  dec.eval()
  g = torch.Generator().manual_seed(2)
  enc = torch.randn(2, 6, 16, generator=g)
  pad = torch.zeros(2, 6)
  base = dec.GreedyDecode(dec.theta, enc, pad, max_len=5)

  lm = lm_lib.TransformerLm.Params().Set(
      name='lm', vocab_size=V, model_dim=16, num_layers=1, num_heads=1,
      hidden_dim=32, dropout_prob=0.0, random_seed=5).Instantiate()
  lm.eval()
  fusion = asr_lib.ShallowFusion(lm, lm.theta, weight=100.0)
  fused = dec.GreedyDecode(dec.theta, enc, pad, max_len=5,
                           fusion=fusion)
  assert fused.shape[0] == 2
  # overwhelming LM weight changes the hypotheses
  assert not torch.equal(base[:, :fused.shape[1]], fused[:, :base.shape[1]])
  # zero weight reproduces the acoustic-only decode
  fusion0 = asr_lib.ShallowFusion(lm, lm.theta, weight=0.0)
  same = dec.GreedyDecode(dec.theta, enc, pad, max_len=5, fusion=fusion0)
  assert torch.equal(base, same)


def test_specaugment_time_warp():
  import torch
  from lingvo_amd.core import py_utils
  from lingvo_amd.layers import spectrum_augmenter as sa
  aug = sa.SpectrumAugmenter.Params().Set(
      name='sa', freq_mask_count=0, time_mask_count=0,
      time_warp_max_frames=8).Instantiate()
  aug.train()
  g = torch.Generator().manual_seed(4)
  x = torch.randn(2, 40, 8, generator=g)
  pad = torch.zeros(2, 40)
  pad[1, 30:] = 1.0
  with py_utils.StepSeedScope(7, 1):
    out = aug.FProp(aug.theta, x, pad)
  assert out.shape == x.shape
  # warped (not identical) but value range preserved (interpolation)
  assert (out - x).abs().max() > 1e-4
  assert out.min() >= x.min() - 1e-4 and out.max() <= x.max() + 1e-4
  # padded frames zeroed by the layer's ApplyPadding contract
  assert out[1, 30:].abs().max() < 1e-6
  # deterministic per (seed, step)
  with py_utils.StepSeedScope(7, 1):
    out2 = aug.FProp(aug.theta, x, pad)
  assert torch.equal(out, out2)


def test_decoder_fast_path_matches_loop_path():
  """_FastPredictions (fused GEMM loop) == generic per-cell loop."""
  import torch
  from lingvo_amd.models import asr as asr_lib
  from lingvo_amd.core.nested_map import NestedMap
  dec = asr_lib.AsrDecoder.Params().Set(
      name='d', vocab_size=24, emb_dim=12, rnn_cell_dim=16,
      source_dim=20, num_lstm_layers=2, dropout_prob=0.0,
      random_seed=5).Instantiate()
  dec.eval()
  g = torch.Generator().manual_seed(1)
  enc = torch.randn(2, 7, 20, generator=g)
  pad = torch.zeros(2, 7)
  pad[1, 5:] = 1.0
  tgt = NestedMap(ids=torch.randint(3, 24, (2, 6), generator=g),
                  paddings=torch.zeros(2, 6))
  fast = dec._FastPredictions(dec.theta, enc, pad, tgt)
  slow = dec._LoopPredictions(dec.theta, enc, pad, tgt)
  assert (fast.atten_vecs - slow.atten_vecs).abs().max() < 1e-4


def test_asr_tfrecord_input_end_to_end(tmp_path):
  """Real-data shaped pipeline: tfrecord shard -> C++ yielder -> codec
  -> bucketing batcher -> AsrModel train step."""
  import torch
  from lingvo_amd.core import tf_example
  from lingvo_amd.models import asr as asr_lib
  recs = []
  g = torch.Generator().manual_seed(5)
  for i in range(24):
    t = 12 + int(torch.randint(0, 8, (1,), generator=g))
    frames = torch.randn(t, 8, generator=g).reshape(-1).tolist()
    toks = torch.randint(3, 30, (5,), generator=g).tolist()
    recs.append(tf_example.EncodeExample(
        {'frames': frames, 'tokens': toks}))
  shard = tmp_path / 's.tfrecord'
  tf_example.WriteTfRecord(str(shard), recs)

  ip = asr_lib.AsrTfRecordInput.Params().Set(
      name='in', files=[str(shard)], feature_dim=8, target_len=8,
      batch_size=4, bucket_upper_bound=[64]).Instantiate()
  batch = ip.GetPreprocessedInputBatch()
  assert batch.src.src_inputs.shape[0] == 4
  assert batch.src.src_inputs.shape[2] == 8
  assert batch.tgt.ids[:, 0].eq(1).all()
  # feeds a real train step
  mp2 = asr_lib.AsrModel.Params().Set(name='m', random_seed=3)
  mp2.encoder.Set(input_dim=8, model_dim=32, num_layers=1, num_heads=2,
                  kernel_size=4, dropout_prob=0.0, specaug_tpl=None)
  mp2.decoder.Set(vocab_size=32, emb_dim=8, rnn_cell_dim=16,
                  source_dim=32, dropout_prob=0.0)
  task = mp2.Instantiate()
  metrics, _ = task.FPropDefaultTheta(batch) if hasattr(
      task, 'FPropDefaultTheta') else task.FProp(task.theta, batch)
  assert torch.isfinite(metrics['loss'][0])
  ip.Stop()


def test_asr_beam_search_decode():
  """Beam-1 matches greedy; beam-4 returns ranked hypotheses."""
  import torch
  from lingvo_amd.models import asr as asr_lib
  dec = asr_lib.AsrDecoder.Params().Set(
      name='d', vocab_size=20, emb_dim=8, rnn_cell_dim=16,
      source_dim=16, num_lstm_layers=2, dropout_prob=0.0,
      random_seed=7).Instantiate()
  dec.eval()
  g = torch.Generator().manual_seed(3)
  enc = torch.randn(2, 6, 16, generator=g)
  pad = torch.zeros(2, 6)
  greedy = dec.GreedyDecode(dec.theta, enc, pad, max_len=6)
  out1 = dec.BeamSearchDecode(dec.theta, enc, pad, num_hyps=1,
                              max_steps=6)
  for b in range(2):
    n = int(out1.topk_lens[b, 0])
    ids = out1.topk_ids[b, 0, :n].tolist()
    ids = [t for t in ids if t != 2]
    gt = [t for t in greedy[b].tolist() if t != 2][:len(ids)]
    assert ids == gt[:len(ids)], (b, ids, gt)
  out4 = dec.BeamSearchDecode(dec.theta, enc, pad, num_hyps=4,
                              max_steps=6)
  s = out4.topk_scores
  assert bool((s[:, :-1] >= s[:, 1:]).all())  # ranked


def test_las_models_train_in_bf16():
  """Regression: fp32 paddings used to promote the biLSTM recurrent
  state to fp32 on step 2 (dtype-mismatch crash in bf16 training)."""
  import torch
  from lingvo_amd.core import registry
  registry.ImportAllParams()
  p = registry.GetParams('asr.librispeech.Librispeech960Base', 'Train')
  p.input.batch_size = 2
  p.input.frame_len = 40
  task = p.Instantiate().GetTask()
  task.to(torch.bfloat16)
  m = task.TrainStep(task.input_generator.GetPreprocessedInputBatch())
  loss = float(m[task.learners[0].p.loss_name][0].detach())
  assert loss == loss  # not NaN
