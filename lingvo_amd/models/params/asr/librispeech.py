"""Registered Librispeech ASR params (reference
lingvo/tasks/asr/params/librispeech.py:28-310).

Librispeech960WpmConformerL is the BASELINE.json north-star config:
Conformer-L per the Conformer paper (17 blocks, d=512, h=8, conv kernel
32) with an attention LSTM decoder, bf16 compute, synthetic 80-dim
features (no network for the real corpus)."""

from __future__ import annotations

import torch

from lingvo_amd.core import learner as learner_lib
from lingvo_amd.core import optimizer as optimizer_lib
from lingvo_amd.core import registry
from lingvo_amd.core import schedule as schedule_lib
from lingvo_amd.core.base_model_params import SingleTaskModelParams
from lingvo_amd.models import asr as asr_model


@registry.RegisterSingleTaskModel
class Librispeech960WpmConformerL(SingleTaskModelParams):
  """Conformer-L on (synthetic) Librispeech 960h, WPM targets."""

  BATCH_SIZE = 16
  FRAME_LEN = 1200
  TARGET_LEN = 64
  VOCAB = 1024

  def Train(self):
    return asr_model.SyntheticAsrInput.Params().Set(
        name='train', batch_size=self.BATCH_SIZE,
        frame_len=self.FRAME_LEN, target_len=self.TARGET_LEN,
        vocab_size=self.VOCAB)

  def Dev(self):
    return self.Train().Set(name='dev')

  def Test(self):
    return self.Train().Set(name='test')

  def Task(self):
    p = asr_model.AsrModel.Params().Set(name='librispeech_conformer_l')
    p.fprop_dtype = torch.bfloat16
    p.train.bf16_weights = True
    p.encoder.Set(input_dim=80, model_dim=512, num_layers=17, num_heads=8,
                  kernel_size=32, dropout_prob=0.1)
    p.decoder.Set(vocab_size=self.VOCAB, emb_dim=128, rnn_cell_dim=640,
                  num_lstm_layers=2, source_dim=512, dropout_prob=0.1)
    p.train.learner = learner_lib.Learner.Params().Set(
        learning_rate=1.0,
        optimizer=optimizer_lib.Adam.Params().Set(
            beta1=0.9, beta2=0.98, epsilon=1e-9),
        lr_schedule=schedule_lib.TransformerSchedule.Params().Set(
            warmup_steps=10_000, model_dim=512),
        clip_gradient_norm_to_value=5.0)
    return p


@registry.RegisterSingleTaskModel
class Librispeech960Grapheme(Librispeech960WpmConformerL):
  """Grapheme variant (reference librispeech.py:156): smaller vocab."""

  VOCAB = 76


@registry.RegisterSingleTaskModel
class Librispeech960ConformerS(Librispeech960WpmConformerL):
  """Conformer-S: 16 blocks, d=144, h=4 — quick tests."""

  def Task(self):
    p = super().Task()
    # d=144 is not H-kernel friendly; use d=256/h=4 -> H=64.
    p.encoder.Set(model_dim=256, num_layers=4, num_heads=4)
    p.decoder.Set(rnn_cell_dim=320, source_dim=256)
    return p


@registry.RegisterSingleTaskModel
class Librispeech960Base(Librispeech960WpmConformerL):
  """LAS baseline (reference librispeech.py:28 Librispeech960Base):
  conv frontend + 4x biLSTM-1024 encoder, attention LSTM decoder."""

  def Task(self):
    p = asr_model.AsrModel.Params().Set(name='librispeech_las')
    p.encoder = asr_model.LasEncoder.Params().Set(
        input_dim=80, model_dim=1024, num_lstm_layers=4,
        dropout_prob=0.2)
    p.fprop_dtype = torch.bfloat16
    p.train.bf16_weights = True
    p.decoder.Set(vocab_size=self.VOCAB, emb_dim=128, rnn_cell_dim=1024,
                  num_lstm_layers=2, source_dim=1024, dropout_prob=0.2)
    p.train.learner = learner_lib.Learner.Params().Set(
        learning_rate=2.5e-4,
        optimizer=optimizer_lib.Adam.Params().Set(
            beta1=0.9, beta2=0.999, epsilon=1e-6),
        lr_schedule=schedule_lib.ContinuousSchedule.Params().Set(
            start_step=50_000, half_life_steps=100_000),
        clip_gradient_norm_to_value=1.0)
    return p


@registry.RegisterSingleTaskModel
class Librispeech960BaseGrapheme(Librispeech960Base):
  """Grapheme-target LAS (reference librispeech.py:156)."""

  VOCAB = 76


@registry.RegisterSingleTaskModel
class Librispeech960Wpm(Librispeech960Base):
  """WPM-target LAS (reference librispeech.py:239
  Librispeech960Wpm: 16k word pieces, 96-dim embeddings, target len
  140). Synthetic inputs mirror the WPM shapes; plug a vocab file into
  tokenizers.WpmTokenizer for real data."""

  VOCAB = 16328
  TARGET_LEN = 140

  def Task(self):
    p = super().Task()
    p.name = 'librispeech_las_wpm'
    p.decoder.Set(vocab_size=self.VOCAB, emb_dim=96)
    return p
