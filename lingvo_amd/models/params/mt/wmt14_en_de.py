"""Registered WMT'14 En->De params (reference
lingvo/tasks/mt/params/wmt14_en_de.py:27,100 WmtEnDeTransformerBase/Big)."""

from __future__ import annotations

import torch

from lingvo_amd.core import learner as learner_lib
from lingvo_amd.core import optimizer as optimizer_lib
from lingvo_amd.core import registry
from lingvo_amd.core import schedule as schedule_lib
from lingvo_amd.core.base_model_params import SingleTaskModelParams
from lingvo_amd.models import mt as mt_model


@registry.RegisterSingleTaskModel
class WmtEnDeTransformerBase(SingleTaskModelParams):
  """Transformer base: 6+6 layers, d=512, ff=2048, 8 heads, 32k WPM."""

  DIM = 512
  FF = 2048
  HEADS = 8
  LAYERS = 6
  VOCAB = 32000
  BATCH = 32

  def Train(self):
    return mt_model.SyntheticNmtInput.Params().Set(
        name='train', batch_size=self.BATCH, src_len=64, tgt_len=64,
        vocab_size=self.VOCAB)

  def Dev(self):
    return self.Train().Set(name='dev')

  def Test(self):
    return self.Train().Set(name='test')

  def Task(self):
    p = mt_model.TransformerModel.Params().Set(name='wmt14_en_de')
    p.fprop_dtype = torch.bfloat16
    p.train.bf16_weights = True
    p.encoder.Set(vocab_size=self.VOCAB, model_dim=self.DIM,
                  num_layers=self.LAYERS, num_heads=self.HEADS,
                  hidden_dim=self.FF, dropout_prob=0.1)
    p.decoder.Set(vocab_size=self.VOCAB, model_dim=self.DIM,
                  num_layers=self.LAYERS, num_heads=self.HEADS,
                  hidden_dim=self.FF, dropout_prob=0.1,
                  label_smoothing=0.1)
    p.train.learner = learner_lib.Learner.Params().Set(
        learning_rate=1.0,
        optimizer=optimizer_lib.Adam.ParamsB(),
        lr_schedule=schedule_lib.TransformerSchedule.Params().Set(
            warmup_steps=4000, model_dim=self.DIM),
        clip_gradient_norm_to_value=0.0)
    return p


@registry.RegisterSingleTaskModel
class WmtEnDeTransformerBig(WmtEnDeTransformerBase):
  """Transformer big (reference wmt14_en_de.py:100): d=1024, ff=4096,
  16 heads — BASELINE config 4 model (GPipe across 8 GPUs)."""

  DIM = 1024
  FF = 4096
  HEADS = 16


@registry.RegisterSingleTaskModel
class WmtEnDeTransformerSmall(WmtEnDeTransformerBase):
  """Transformer small (reference wmt14_en_de.py:100): d=256, ff=1024,
  4 heads, 2+2 layers — debugging config."""

  DIM = 256
  FF = 1024
  HEADS = 4
  LAYERS = 2


@registry.RegisterSingleTaskModel
class WmtEnDeRNMT(WmtEnDeTransformerBase):
  """RNMT+ (reference wmt14_en_de.py:141 WmtEnDeRNMT): biLSTM encoder
  + attention LSTM decoder."""

  def Task(self):
    p = mt_model.RnmtModel.Params().Set(name='wmt14_en_de_rnmt')
    p.fprop_dtype = torch.bfloat16
    p.train.bf16_weights = True
    p.encoder.Set(vocab_size=self.VOCAB, model_dim=1024,
                  num_lstm_layers=4, dropout_prob=0.2)
    p.decoder.Set(vocab_size=self.VOCAB, emb_dim=1024,
                  rnn_cell_dim=1024, num_lstm_layers=2,
                  source_dim=1024, dropout_prob=0.2)
    p.train.learner = learner_lib.Learner.Params().Set(
        learning_rate=1e-4,
        optimizer=optimizer_lib.Adam.Params().Set(beta2=0.999),
        clip_gradient_norm_to_value=5.0)
    return p
