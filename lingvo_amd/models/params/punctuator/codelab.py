"""Punctuator task (reference lingvo/tasks/punctuator: a tutorial
seq2seq task that restores punctuation/capitalization; params/codelab.py).
Reuses the MT transformer with a character tokenizer over synthetic
brown-corpus-shaped data."""

from __future__ import annotations

import torch

from lingvo_amd.core import learner as learner_lib
from lingvo_amd.core import optimizer as optimizer_lib
from lingvo_amd.core import registry
from lingvo_amd.core import schedule as schedule_lib
from lingvo_amd.core.base_model_params import SingleTaskModelParams
from lingvo_amd.models import mt as mt_model


@registry.RegisterSingleTaskModel
class RNMTModel(SingleTaskModelParams):
  """Character-level punctuation restoration (codelab model)."""

  VOCAB = 96  # ascii chars

  def Train(self):
    return mt_model.SyntheticNmtInput.Params().Set(
        name='train', batch_size=16, src_len=96, tgt_len=96,
        vocab_size=self.VOCAB)

  def Dev(self):
    return self.Train().Set(name='dev')

  def Test(self):
    return self.Train().Set(name='test')

  def Task(self):
    p = mt_model.TransformerModel.Params().Set(name='punctuator')
    p.fprop_dtype = torch.bfloat16
    p.train.bf16_weights = True
    p.encoder.Set(vocab_size=self.VOCAB, model_dim=256, num_layers=2,
                  num_heads=4, hidden_dim=1024, dropout_prob=0.1)
    p.decoder.Set(vocab_size=self.VOCAB, model_dim=256, num_layers=2,
                  num_heads=4, hidden_dim=1024, dropout_prob=0.1,
                  label_smoothing=0.1)
    p.train.learner = learner_lib.Learner.Params().Set(
        learning_rate=1.0,
        optimizer=optimizer_lib.Adam.ParamsB(),
        lr_schedule=schedule_lib.TransformerSchedule.Params().Set(
            warmup_steps=1000, model_dim=256))
    return p
