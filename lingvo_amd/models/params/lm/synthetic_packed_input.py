"""Registered synthetic dense/MoE LM params (reference
lingvo/tasks/lm/params/synthetic_packed_input.py:53-481 DenseLm family).

DenseLm8B-scale geometry is reachable on one MI355X (288 GB HBM); the
MoE config is BASELINE config 5 (64-expert expert-parallel all-to-all).
"""

from __future__ import annotations

import torch

from lingvo_amd.core import learner as learner_lib
from lingvo_amd.core import optimizer as optimizer_lib
from lingvo_amd.core import registry
from lingvo_amd.core import schedule as schedule_lib
from lingvo_amd.core.base_model_params import SingleTaskModelParams
from lingvo_amd.layers import transformer as transformer_lib
from lingvo_amd.models import lm as lm_model


@registry.RegisterSingleTaskModel
class DenseLm1B(SingleTaskModelParams):
  """~1.3B-param dense LM (synthetic data)."""

  BATCH = 4
  SEQ = 1024
  VOCAB = 32000
  LAYERS = 24
  DIM = 2048

  def Train(self):
    return lm_model.SyntheticLmInput.Params().Set(
        name='train', batch_size=self.BATCH, seq_len=self.SEQ,
        vocab_size=self.VOCAB)

  def Dev(self):
    return self.Train().Set(name='dev')

  def Test(self):
    return self.Train().Set(name='test')

  def Task(self):
    p = lm_model.LanguageModel.Params().Set(name='dense_lm')
    p.fprop_dtype = torch.bfloat16
    p.train.bf16_weights = True
    p.lm = lm_model.TransformerLm.Params().Set(
        vocab_size=self.VOCAB, model_dim=self.DIM, num_layers=self.LAYERS,
        num_heads=self.DIM // 128, hidden_dim=4 * self.DIM,
        dropout_prob=0.0, remat=False)
    p.train.learner = learner_lib.Learner.Params().Set(
        learning_rate=1.0,
        optimizer=optimizer_lib.Adafactor.Params(),
        lr_schedule=schedule_lib.TransformerSchedule.Params().Set(
            warmup_steps=1000, model_dim=self.DIM),
        clip_gradient_norm_to_value=1.0)
    return p


@registry.RegisterSingleTaskModel
class MoELm64E(DenseLm1B):
  """64-expert top-2 MoE Transformer-LM, expert-parallel all-to-all over
  RCCL/xGMI (BASELINE config 5; reference gshard MoE layers)."""

  LAYERS = 12
  DIM = 1024
  NUM_EXPERTS = 64

  def Task(self):
    p = super().Task().Set(name='moe_lm_64e')
    # Every other layer gets a 64-expert MoE FFN (GShard pattern).
    p.lm.Set(model_dim=self.DIM, num_layers=self.LAYERS,
             num_heads=self.DIM // 128, hidden_dim=4 * self.DIM,
             moe_every_n=2, num_experts=self.NUM_EXPERTS,
             expert_capacity_factor=2.0)
    return p


@registry.RegisterSingleTaskModel
class DenseLm8B(DenseLm1B):
  """~8B-param dense LM (reference synthetic_packed_input.py:53
  DenseLm8B shape: 32 layers, d=4096). Fits one MI355X (288 GB HBM);
  for TP runs the stack carries sharding annotations lowered by
  LowerShardingAnnotations."""

  LAYERS = 32
  DIM = 4096

  def Task(self):
    p = super().Task()
    p.lm.name = 'dense_lm_8b'
    return p


@registry.RegisterSingleTaskModel
class DenseLm128B8x8(DenseLm8B):
  """~128B dense LM for 8-way TP x 8-way DP (the scaled GShard family,
  reference synthetic_packed_input.py:330 DenseLm1T16x16 pattern —
  sized for one 8-GPU MI355X node instead of a TPU pod: 2.3 TB HBM
  holds the fp32 masters + bf16 weights sharded 8 ways)."""

  LAYERS = 64
  DIM = 12288
  BATCH = 1

  def Task(self):
    p = super().Task()
    p.lm.name = 'dense_lm_128b'
    # GShard-style annotation; propagated to the transformer stack and
    # lowered to explicit column/row-parallel layers + RCCL collectives
    # by parallel.tensor_parallel.LowerShardingAnnotations.
    p.lm.weight_split_dims_mapping = [-1, 0]
    return p


@registry.RegisterSingleTaskModel
class DenseLm1T16x16(DenseLm128B8x8):
  """~1T-param dense LM config (reference synthetic_packed_input.py:330
  DenseLm1T16x16). Registered for config parity: training it needs
  multi-node TP x DP (16x16 in the reference's TPU terms); on MI355X
  the layout is 8-way TP inside a node x DP across nodes with the same
  sharding annotations."""

  LAYERS = 128
  DIM = 24576

  def Task(self):
    p = super().Task()
    p.lm.name = 'dense_lm_1t'
    return p
