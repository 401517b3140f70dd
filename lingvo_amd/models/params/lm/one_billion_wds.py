"""Registered 1-Billion-Word LM params (reference
lingvo/tasks/lm/params/one_billion_wds.py:138,181)."""

from __future__ import annotations

import torch

from lingvo_amd.core import learner as learner_lib
from lingvo_amd.core import optimizer as optimizer_lib
from lingvo_amd.core import registry
from lingvo_amd.core import schedule as schedule_lib
from lingvo_amd.core.base_model_params import SingleTaskModelParams
from lingvo_amd.models import lm as lm_model


@registry.RegisterSingleTaskModel
class OneBWdsTransformerLm(SingleTaskModelParams):
  """Transformer LM, bf16, sized for 1x MI355X (BASELINE config 2).

  Matches the reference GPipe config's transformer geometry
  (one_billion_wds.py:181-198: 32 layers, d=2048, ff=8192, 16 heads,
  WPM 32k vocab) but runs on a single MI355X — 288 GB HBM makes the
  4-way pipeline split unnecessary at this size.
  """

  BATCH = 8
  SEQ = 1024
  VOCAB = 32000
  LAYERS = 32
  DIM = 2048
  HEADS = 16

  def Train(self):
    return lm_model.SyntheticLmInput.Params().Set(
        name='train', batch_size=self.BATCH, seq_len=self.SEQ,
        vocab_size=self.VOCAB)

  def Dev(self):
    return self.Train().Set(name='dev')

  def Test(self):
    return self.Train().Set(name='test')

  def Task(self):
    p = lm_model.LanguageModel.Params().Set(name='1bwds_transformer_lm')
    p.fprop_dtype = torch.bfloat16
    p.train.bf16_weights = True
    p.lm = lm_model.TransformerLm.Params().Set(
        vocab_size=self.VOCAB, model_dim=self.DIM, num_layers=self.LAYERS,
        num_heads=self.HEADS, hidden_dim=4 * self.DIM, dropout_prob=0.1)
    p.train.learner = learner_lib.Learner.Params().Set(
        learning_rate=1.0,
        optimizer=optimizer_lib.Adam.ParamsB(),
        lr_schedule=schedule_lib.TransformerSchedule.Params().Set(
            warmup_steps=4000, model_dim=self.DIM),
        clip_gradient_norm_to_value=1.0)
    return p


@registry.RegisterSingleTaskModel
class WordLevelOneBwdsRnnLm(SingleTaskModelParams):
  """RNN LM baseline (reference one_billion_wds.py:138
  WordLevelOneBwdsSimpleSampledSoftmax: 2x2048 LSTM, 1024 proj)."""

  def Train(self):
    return lm_model.SyntheticLmInput.Params().Set(
        name='train', batch_size=16, seq_len=128, vocab_size=32000)

  def Dev(self):
    return self.Train().Set(name='dev')

  def Test(self):
    return self.Train().Set(name='test')

  def Task(self):
    p = lm_model.LanguageModel.Params().Set(name='1bwds_rnn_lm')
    p.fprop_dtype = torch.bfloat16
    p.train.bf16_weights = True
    p.lm = lm_model.RnnLm.Params().Set(
        vocab_size=32000, emb_dim=1024, rnn_dims=[2048, 2048],
        rnn_proj=1024, dropout_prob=0.1,
        num_sampled=4096)  # reference: SimpleSampledSoftmax baseline
    p.train.learner = learner_lib.Learner.Params().Set(
        learning_rate=0.1,
        optimizer=optimizer_lib.Adagrad.Params(),
        clip_gradient_norm_to_value=1.0)
    return p


@registry.RegisterSingleTaskModel
class OneBWdsGPipeTransformerWPM(OneBWdsTransformerLm):
  """GPipe pipelined variant (reference one_billion_wds.py:181: 32
  layers, 4 splits, 32 microbatches on 4x V100-16GB). On MI355X the
  same model fits on one GPU (see OneBWdsTransformerLm); this config
  carries the pipeline topology for multi-GPU PP runs via
  lingvo_amd.parallel.gpipe_lm (one process per stage over RCCL P2P).
  """

  NUM_STAGES = 4
  NUM_MICRO_BATCHES = 32

  def Task(self):
    p = super().Task().Set(name='1bwds_gpipe_transformer_lm')
    # Pipeline topology consumed by the gpipe_lm launcher.
    p.Define('pipeline_num_stages', self.NUM_STAGES, 'Pipeline stages.')
    p.Define('pipeline_num_micro_batches', self.NUM_MICRO_BATCHES,
             'Microbatches per step (>= 4x stages per reference '
             'guidance, one_billion_wds.py:197).')
    return p
