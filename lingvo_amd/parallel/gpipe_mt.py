"""Pipelined MT Transformer stages for GPipe (BASELINE config 4:
WMT14 Transformer-Big pipeline-parallel across 8 GPUs).

Reference: lingvo/core/layers_with_gpipe.py GPipeTransformerStack — the
pipeline stream carries (src activations, src paddings, tgt activations,
tgt paddings); ALL encoder layers are placed before any decoder layer,
so every decoder layer sees the final encoder output flowing along the
stream. Stage 0 owns both embeddings; the last stage owns the softmax.
"""

from __future__ import annotations

from typing import List, Optional

import torch

from lingvo_amd.core import py_utils
from lingvo_amd.core.base_layer import BaseLayer
from lingvo_amd.core.nested_map import NestedMap
from lingvo_amd.layers import layers as lingvo_layers
from lingvo_amd.layers import transformer as transformer_lib
from lingvo_amd.parallel.pipeline import (GPipeRunner,
                                          PartitionSequentialLayers)


class TransformerMtStage(BaseLayer):
  """One pipeline stage holding a contiguous slice of the combined
  [enc_0..enc_E-1, dec_0..dec_D-1] layer list."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('vocab_size', 32000, 'Shared vocab.')
    p.Define('model_dim', 1024, 'Model dim.')
    p.Define('num_encoder_layers', 6, 'Encoder layers total.')
    p.Define('num_decoder_layers', 6, 'Decoder layers total.')
    p.Define('num_heads', 16, 'Heads.')
    p.Define('hidden_dim', 4096, 'FFN hidden.')
    p.Define('dropout_prob', 0.0, 'Dropout.')
    p.Define('stage_idx', 0, 'This stage.')
    p.Define('num_stages', 1, 'Total stages.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    total = p.num_encoder_layers + p.num_decoder_layers
    parts = PartitionSequentialLayers(list(range(total)), p.num_stages)
    self._my_layers = parts[p.stage_idx]
    self.is_first = p.stage_idx == 0
    self.is_last = p.stage_idx == p.num_stages - 1
    if self.is_first:
      self.CreateChild('emb', lingvo_layers.EmbeddingLayer.Params().Set(
          vocab_size=p.vocab_size, embedding_dim=p.model_dim,
          scale_sqrt_depth=True))
      self.CreateChild('pos_emb',
                       lingvo_layers.PositionalEmbeddingLayer.Params().Set(
                           embedding_dim=p.model_dim))
    layer_ps = []
    self._kinds = []
    for li in self._my_layers:
      is_dec = li >= p.num_encoder_layers
      lp = transformer_lib.TransformerLayer.Params().Set(
          name=f'{"dec" if is_dec else "enc"}_{li}',
          input_dim=p.model_dim, num_heads=p.num_heads,
          mask_self_atten=is_dec, has_aux_atten=is_dec)
      lp.tr_fflayer_tpl.hidden_dim = p.hidden_dim
      lp.tr_atten_tpl.residual_dropout_prob = p.dropout_prob
      lp.tr_fflayer_tpl.residual_dropout_prob = p.dropout_prob
      layer_ps.append(lp)
      self._kinds.append('dec' if is_dec else 'enc')
    self.CreateChildren('layers', layer_ps)
    if self.is_last:
      self.CreateChild('softmax',
                       lingvo_layers.SimpleFullSoftmax.Params().Set(
                           input_dim=p.model_dim,
                           num_classes=p.vocab_size))

  def FProp(self, theta: NestedMap, nmap: NestedMap) -> NestedMap:
    p = self.p
    if self.is_first:
      src = self.emb.EmbLookup(theta.emb, nmap.src_ids.long()).to(
          self.fprop_dtype)
      pos_s = self.pos_emb.FProp(theta.pos_emb, src.shape[1],
                                 device=src.device)
      src = src + pos_s.unsqueeze(0).to(src.dtype)
      tgt = self.emb.EmbLookup(theta.emb, nmap.tgt_ids.long()).to(
          self.fprop_dtype)
      pos_t = self.pos_emb.FProp(theta.pos_emb, tgt.shape[1],
                                 device=tgt.device)
      tgt = tgt + pos_t.unsqueeze(0).to(tgt.dtype)
    else:
      src, tgt = nmap.src, nmap.tgt
    src_pad = nmap.src_paddings
    tgt_pad = nmap.tgt_paddings
    for i, layer in enumerate(self.layers):
      if self._kinds[i] == 'enc':
        src = layer.FProp(theta.layers[i], src, src_pad)
      else:
        tgt = layer.FProp(theta.layers[i], tgt, tgt_pad,
                          aux_vecs=src, aux_paddings=src_pad)
    return NestedMap(src=src, src_paddings=src_pad, tgt=tgt,
                     tgt_paddings=tgt_pad)

  def XentLoss(self, theta, act, labels, weights):
    return self.softmax.XentLoss(theta.softmax, act,
                                 class_weights=weights,
                                 class_ids=labels)


def RunGPipeMtStep(stage: TransformerMtStage, runner: GPipeRunner,
                   batches: List[NestedMap],
                   schedule: str = 'fill_drain'
                   ) -> Optional[torch.Tensor]:
  """One pipelined MT train step over microbatches.

  batches[m]: NestedMap(src=..ids/paddings.., tgt=..ids/paddings/
  labels/weights..) per microbatch. Gradients accumulate in the stage's
  parameters; the caller applies the per-stage optimizer.
  """
  theta = stage.theta

  def fprop(nmap):
    return stage.FProp(theta, nmap)

  def input_fn(m):
    b = batches[m]
    return NestedMap(src_ids=b.src.ids,
                     src_paddings=b.src.paddings.float(),
                     tgt_ids=b.tgt.ids,
                     tgt_paddings=b.tgt.paddings.float())

  def loss_fn(nmap, m):
    b = batches[m]
    xent = stage.XentLoss(theta, nmap.tgt, b.tgt.labels, b.tgt.weights)
    return xent.avg_xent

  return runner.RunStep(fprop, input_fn=input_fn, loss_fn=loss_fn,
                        schedule=schedule)
