"""Machine-translation task: batch-major Transformer encoder/decoder with
beam-search decoding (reference lingvo/tasks/mt/model.py:176
TransformerModel, encoder.py:836, decoder.py:2361 batch-major variants).
"""

from __future__ import annotations


import torch

from lingvo_amd.core import beam_search_helper, py_utils
from lingvo_amd.core import metrics as metrics_lib
from lingvo_amd.core.base_input_generator import BaseSequenceInputGenerator
from lingvo_amd.core.base_layer import BaseLayer
from lingvo_amd.core.base_model import BaseTask
from lingvo_amd.core.nested_map import NestedMap
from lingvo_amd.layers import layers as lingvo_layers
from lingvo_amd.layers import transformer as transformer_lib


class SyntheticNmtInput(BaseSequenceInputGenerator):
  """Synthetic WMT-shaped paired batches (reference
  tasks/mt/input_generator.py NmtInput)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.batch_size = 16
    p.Define('src_len', 64, 'Source length.')
    p.Define('tgt_len', 64, 'Target length.')
    p.Define('vocab_size', 32000, 'Shared WPM vocab.')
    return p

  def _InputBatch(self) -> NestedMap:
    p = self.p
    g = torch.Generator().manual_seed(5000 + self._batch_count)
    b = p.batch_size
    src_ids = torch.randint(3, p.vocab_size, (b, p.src_len), generator=g)
    src_lens = torch.randint(int(0.7 * p.src_len), p.src_len + 1, (b,),
                             generator=g)
    src_pad = py_utils.PaddingsFromLengths(src_lens, p.src_len)
    tgt_ids = torch.randint(3, p.vocab_size, (b, p.tgt_len), generator=g)
    tgt_lens = torch.randint(int(0.7 * p.tgt_len), p.tgt_len + 1, (b,),
                             generator=g)
    tgt_pad = py_utils.PaddingsFromLengths(tgt_lens, p.tgt_len)
    labels = tgt_ids.roll(-1, dims=1)
    return NestedMap(
        src=NestedMap(ids=(src_ids * (1 - src_pad).long()),
                      paddings=src_pad),
        tgt=NestedMap(ids=(tgt_ids * (1 - tgt_pad).long()),
                      paddings=tgt_pad,
                      labels=(labels * (1 - tgt_pad).long()),
                      weights=1.0 - tgt_pad))


class TransformerEncoder(BaseLayer):
  """(reference tasks/mt/encoder.py:836 TransformerBatchMajorEncoder)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('vocab_size', 32000, 'Vocab.')
    p.Define('model_dim', 512, 'Model dim.')
    p.Define('num_layers', 6, 'Layers.')
    p.Define('num_heads', 8, 'Heads.')
    p.Define('hidden_dim', 2048, 'FFN hidden.')
    p.Define('dropout_prob', 0.1, 'Dropout.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    self.CreateChild('emb', lingvo_layers.EmbeddingLayer.Params().Set(
        vocab_size=p.vocab_size, embedding_dim=p.model_dim,
        scale_sqrt_depth=True))
    self.CreateChild('pos_emb',
                     lingvo_layers.PositionalEmbeddingLayer.Params().Set(
                         embedding_dim=p.model_dim))
    sp = transformer_lib.StackedTransformerLayers.Params().Set(
        model_dim=p.model_dim, num_layers=p.num_layers,
        num_heads=p.num_heads, hidden_dim=p.hidden_dim)
    sp.transformer_tpl.tr_atten_tpl.residual_dropout_prob = p.dropout_prob
    sp.transformer_tpl.tr_fflayer_tpl.residual_dropout_prob = p.dropout_prob
    self.CreateChild('stack', sp)

  def FProp(self, theta, ids, paddings):
    x = self.emb.EmbLookup(theta.emb, ids.long()).to(self.fprop_dtype)
    pos = self.pos_emb.FProp(theta.pos_emb, ids.shape[1], device=ids.device)
    x = x + pos.unsqueeze(0).to(x.dtype)
    if self.p.dropout_prob and not self.do_eval:
      x = py_utils.DeterministicDropout(x, 1.0 - self.p.dropout_prob)
    return self.stack.FProp(theta.stack, x, paddings)


class TransformerDecoder(BaseLayer):
  """(reference tasks/mt/decoder.py:2361 TransformerBatchMajorDecoder)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('vocab_size', 32000, 'Vocab.')
    p.Define('model_dim', 512, 'Model dim.')
    p.Define('num_layers', 6, 'Layers.')
    p.Define('num_heads', 8, 'Heads.')
    p.Define('hidden_dim', 2048, 'FFN hidden.')
    p.Define('dropout_prob', 0.1, 'Dropout.')
    p.Define('label_smoothing', 0.1, 'Label smoothing uncertainty.')
    p.Define('use_flat_beam_search', False,
             'Use the fully-tensorized flat beam search (no host-side '
             'hypothesis loops; hipGraph-capturable decode).')
    p.Define('beam_search', beam_search_helper.BeamSearchHelper.Params(),
             'Beam search params.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    self.CreateChild('emb', lingvo_layers.EmbeddingLayer.Params().Set(
        vocab_size=p.vocab_size, embedding_dim=p.model_dim,
        scale_sqrt_depth=True))
    self.CreateChild('pos_emb',
                     lingvo_layers.PositionalEmbeddingLayer.Params().Set(
                         embedding_dim=p.model_dim))
    sp = transformer_lib.StackedTransformerLayers.Params().Set(
        model_dim=p.model_dim, num_layers=p.num_layers,
        num_heads=p.num_heads, hidden_dim=p.hidden_dim,
        mask_self_atten=True, has_aux_atten=True)
    sp.transformer_tpl.tr_atten_tpl.residual_dropout_prob = p.dropout_prob
    sp.transformer_tpl.tr_fflayer_tpl.residual_dropout_prob = p.dropout_prob
    self.CreateChild('stack', sp)
    self.CreateChild('softmax', lingvo_layers.SimpleFullSoftmax.Params().Set(
        input_dim=p.model_dim, num_classes=p.vocab_size))
    if p.label_smoothing:
      self.CreateChild('smoother',
                       lingvo_layers.UniformLabelSmoother.Params().Set(
                           num_classes=p.vocab_size,
                           uncertainty=p.label_smoothing))

  def FProp(self, theta, enc, enc_paddings, targets):
    x = self.emb.EmbLookup(theta.emb, targets.ids.long()).to(
        self.fprop_dtype)
    pos = self.pos_emb.FProp(theta.pos_emb, x.shape[1], device=x.device)
    x = x + pos.unsqueeze(0).to(x.dtype)
    if self.p.dropout_prob and not self.do_eval:
      x = py_utils.DeterministicDropout(x, 1.0 - self.p.dropout_prob)
    return self.stack.FProp(theta.stack, x, targets.paddings,
                            aux_vecs=enc, aux_paddings=enc_paddings)

  def ComputeXent(self, theta, act, targets):
    p = self.p
    if p.label_smoothing and not self.do_eval:
      probs = self.smoother.FProp(theta.smoother, targets.labels.long())
      return self.softmax.XentLoss(theta.softmax, act,
                                   class_weights=targets.weights,
                                   class_probabilities=probs)
    return self.softmax.XentLoss(theta.softmax, act,
                                 class_weights=targets.weights,
                                 class_ids=targets.labels)

  # ---- beam search -------------------------------------------------------
  def BeamSearchDecode(self, theta, enc, enc_paddings) -> NestedMap:
    p = self.p
    if p.use_flat_beam_search:
      from lingvo_amd.core import flat_beam_search_helper as fbsh
      fp = fbsh.FlatBeamSearchHelper.Params().Set(
          num_hyps_per_beam=p.beam_search.num_hyps_per_beam,
          max_steps=p.beam_search.max_steps,
          target_sos_id=p.beam_search.target_sos_id,
          target_eos_id=p.beam_search.target_eos_id,
          length_norm_alpha=p.beam_search.length_normalization)
      helper = fbsh.FlatBeamSearchHelper(fp)
    else:
      helper = beam_search_helper.BeamSearchHelper(p.beam_search)
    k = p.beam_search.num_hyps_per_beam
    batch = enc.shape[0]
    max_steps = p.beam_search.max_steps

    enc_tiled = enc.repeat_interleave(k, dim=0)
    pad_tiled = enc_paddings.repeat_interleave(k, dim=0)

    def init_fn(b, num_hyps):
      states = self.stack.InitStates(theta.stack, b * num_hyps, max_steps,
                                     enc.device, self.fprop_dtype)
      return NestedMap(stack=states, t=[0])

    def step_fn(state, prev_ids):
      t = state.t[0]
      x = self.emb.EmbLookup(theta.emb, prev_ids.long()).to(
          self.fprop_dtype).unsqueeze(1)
      pos = self.pos_emb.FProp(theta.pos_emb, t + 1, device=enc.device)
      x = x + pos[t].reshape(1, 1, -1).to(x.dtype)
      out, st = self.stack.ExtendStep(theta.stack, x, state.stack,
                                      aux_vecs=enc_tiled,
                                      aux_paddings=pad_tiled)
      state.stack = st
      state.t[0] = t + 1
      logits = self.softmax.Logits(theta.softmax, out.squeeze(1))
      return torch.log_softmax(logits.float(), dim=-1), state

    def reorder_fn(state, gather_idx):
      def reorder(v):
        if isinstance(v, torch.Tensor) and v.dim() >= 1 and \
            v.shape[0] == gather_idx.shape[0]:
          return v[gather_idx]
        return v
      state.stack = state.stack.Transform(reorder)
      return state

    return helper.BeamSearchDecode(batch, init_fn, step_fn, reorder_fn)


class TransformerModel(BaseTask):
  """MT task (reference tasks/mt/model.py:176)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('encoder', TransformerEncoder.Params(), 'Encoder.')
    p.Define('decoder', TransformerDecoder.Params(), 'Decoder.')
    return p

  def __init__(self, params):
    super().__init__(params)
    self.CreateChild('encoder', self.p.encoder)
    self.CreateChild('decoder', self.p.decoder)

  def ComputePredictions(self, theta, input_batch):
    enc = self.encoder.FProp(theta.encoder, input_batch.src.ids,
                             input_batch.src.paddings)
    act = self.decoder.FProp(theta.decoder, enc, input_batch.src.paddings,
                             input_batch.tgt)
    return NestedMap(activations=act, encoder_outputs=enc)

  def ComputeLoss(self, theta, predictions, input_batch):
    xent = self.decoder.ComputeXent(theta.decoder, predictions.activations,
                                    input_batch.tgt)
    b = input_batch.src.ids.shape[0]
    metrics = NestedMap(
        loss=(xent.avg_xent, xent.total_weight),
        log_pplx=(xent.avg_xent.detach(), xent.total_weight),
        num_samples_in_batch=(torch.tensor(float(b)), torch.ones(())))
    return metrics, NestedMap(per_example_xent=xent.per_example_xent)

  def Decode(self, input_batch) -> NestedMap:
    with torch.no_grad():
      enc = self.encoder.FProp(self.theta.encoder, input_batch.src.ids,
                               input_batch.src.paddings)
      out = self.decoder.BeamSearchDecode(self.theta.decoder, enc,
                                          input_batch.src.paddings)
    out.target_ids = input_batch.tgt.ids
    return out

  def Inference(self) -> NestedMap:
    """Beam-search translate subgraph (reference base_model.py:943)."""

    def default(src_ids, src_paddings):
      enc = self.encoder.FProp(self.theta.encoder, src_ids, src_paddings)
      out = self.decoder.BeamSearchDecode(self.theta.decoder, enc,
                                          src_paddings)
      return NestedMap(topk_ids=out.topk_ids, topk_lens=out.topk_lens,
                       topk_scores=out.topk_scores)

    return NestedMap(default=default)

  def CreateDecoderMetrics(self) -> NestedMap:
    return NestedMap(corpus_bleu=metrics_lib.CorpusBleuMetric(),
                     num_samples_in_batch=metrics_lib.AverageMetric())

  def PostProcessDecodeOut(self, decode_out, decode_metrics) -> None:
    hyps = decode_out.topk_ids[:, 0]  # best hyp
    refs = decode_out.target_ids
    for i in range(hyps.shape[0]):
      hyp = ' '.join(str(int(x)) for x in hyps[i] if int(x) > 2)
      ref = ' '.join(str(int(x)) for x in refs[i] if int(x) > 2)
      decode_metrics.corpus_bleu.Update(ref, hyp)
    decode_metrics.num_samples_in_batch.Update(float(hyps.shape[0]))


class RnmtEncoder(BaseLayer):
  """RNMT+ text encoder: embedding + stacked biLSTM layers (reference
  tasks/mt/encoder.py MTEncoderBiRNN / the WmtEnDeRNMT config,
  wmt14_en_de.py:141)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('vocab_size', 32000, 'Vocab.')
    p.Define('model_dim', 1024, 'Output dim (= 2 * per-dir LSTM).')
    p.Define('num_lstm_layers', 4, 'biLSTM layers.')
    p.Define('dropout_prob', 0.0, 'Inter-layer dropout.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    from lingvo_amd.layers import lstm_frnn_layer
    self.CreateChild('emb', lingvo_layers.EmbeddingLayer.Params().Set(
        vocab_size=p.vocab_size, embedding_dim=p.model_dim,
        scale_sqrt_depth=True))
    half = p.model_dim // 2
    layer_ps = []
    for i in range(p.num_lstm_layers):
      # Hoisted-projection biLSTM (see layers/lstm_frnn_layer.py).
      cell = lstm_frnn_layer.LSTMCellSimpleExt.Params().Set(
          num_input_nodes=p.model_dim, num_output_nodes=half)
      layer_ps.append(
          lstm_frnn_layer.BidirectionalLstmFRNN.Params().Set(
              name=f'blstm_{i}', fwd=cell.Copy(), bak=cell.Copy()))
    self.CreateChildren('rnn', layer_ps)

  def FProp(self, theta: NestedMap, ids: torch.Tensor,
            paddings: torch.Tensor) -> torch.Tensor:
    x = self.emb.EmbLookup(theta.emb, ids.long()).to(self.fprop_dtype)
    for i, layer in enumerate(self.rnn):
      y = layer.FProp(theta.rnn[i], x, paddings)
      if self.p.dropout_prob and not self.do_eval:
        y = py_utils.DeterministicDropout(y, 1.0 - self.p.dropout_prob)
      x = y + x if y.shape == x.shape else y  # residual from layer 2 on
    return py_utils.ApplyPadding(paddings, x)


class RnmtModel(BaseTask):
  """RNN-based MT (reference WmtEnDeRNMT, wmt14_en_de.py:141): biLSTM
  encoder + the attention LSTM decoder family. The decoder reuses the
  LAS-style attention LSTM (tasks/asr/decoder.py:48 and the RNMT
  decoder share that shape)."""

  @classmethod
  def Params(cls):
    from lingvo_amd.models import asr as asr_model
    p = super().Params()
    p.Define('encoder', RnmtEncoder.Params(), 'Encoder.')
    p.Define('decoder', asr_model.AsrDecoder.Params(), 'Decoder.')
    return p

  def __init__(self, params):
    super().__init__(params)
    self.CreateChild('encoder', self.p.encoder)
    self.CreateChild('decoder', self.p.decoder)

  def ComputePredictions(self, theta, input_batch):
    enc = self.encoder.FProp(theta.encoder, input_batch.src.ids,
                             input_batch.src.paddings)
    preds = self.decoder.ComputePredictions(
        theta.decoder, enc, input_batch.src.paddings, input_batch.tgt)
    return preds

  def ComputeLoss(self, theta, predictions, input_batch):
    metrics, per_example = self.decoder.ComputeLoss(
        theta.decoder, predictions, input_batch.tgt)
    b = input_batch.src.ids.shape[0]
    metrics.num_samples_in_batch = (torch.tensor(float(b)),
                                    torch.ones(()))
    return metrics, per_example

  def Decode(self, input_batch) -> NestedMap:
    with torch.no_grad():
      enc = self.encoder.FProp(self.theta.encoder, input_batch.src.ids,
                               input_batch.src.paddings)
      hyps = self.decoder.GreedyDecode(self.theta.decoder, enc,
                                       input_batch.src.paddings)
    return NestedMap(topk_decoded=hyps, target_ids=input_batch.tgt.ids)


class NmtTfRecordInput(BaseSequenceInputGenerator):
  """Real-data MT input: tfrecord Examples with 'src_ids'/'tgt_ids'
  int64 features (the NmtInput export shape, reference
  tasks/mt/input_generator.py). C++ yielder + TF-free codec + length
  bucketing by max(src, tgt) tokens."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.batch_size = 16
    p.Define('files', [], 'TFRecord shards.')
    p.Define('max_len', 128, 'Crop length.')
    p.Define('input_seed', 301, 'Shuffle seed.')
    p.bucket_upper_bound = [128]
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    from lingvo_amd.core import tf_example
    from lingvo_amd.core.generic_input import RecordBatcher
    from lingvo_amd.ops import _loader
    ext = _loader.get_ext(required=True)
    self._yielder = ext.RecordYielder(list(p.files), 'tfrecord',
                                      p.input_seed, 1000, 2, True)
    limits = list(p.bucket_batch_limit) or \
        [p.batch_size] * len(p.bucket_upper_bound)

    def proc(rec):
      ex = tf_example.ParseExample(rec)
      src = torch.tensor(ex['src_ids'][:p.max_len], dtype=torch.long)
      tgt = torch.tensor(ex['tgt_ids'][:p.max_len - 1], dtype=torch.long)
      return NestedMap(src=src, tgt=tgt,
                       src_len=torch.tensor([src.numel()]),
                       tgt_len=torch.tensor([tgt.numel()])), \
          max(src.numel(), tgt.numel() + 1)

    self._batcher = RecordBatcher(self._yielder, proc,
                                  p.bucket_upper_bound, limits,
                                  num_threads=2)

  def _InputBatch(self) -> NestedMap:
    batch = self._batcher.GetNext()
    assert batch is not None, 'input exhausted'
    b = batch.src.shape[0]
    src_pad = py_utils.PaddingsFromLengths(batch.src_len.reshape(-1),
                                           batch.src.shape[1])
    lmax = int(batch.tgt_len.max()) + 1
    ids = torch.full((b, lmax), 2, dtype=torch.long)
    tgt_pad = torch.ones(b, lmax)
    for i in range(b):
      n = int(batch.tgt_len[i])
      ids[i, 0] = 1
      ids[i, 1:n + 1] = batch.tgt[i, :n]
      tgt_pad[i, :n + 1] = 0.0
    labels = ids.roll(-1, dims=1)
    labels[:, -1] = 2
    return NestedMap(
        src=NestedMap(ids=batch.src, paddings=src_pad),
        tgt=NestedMap(ids=ids, paddings=tgt_pad,
                      labels=labels * (1 - tgt_pad).long(),
                      weights=1.0 - tgt_pad))

  def Stop(self):
    self._batcher.Stop()
    self._yielder.stop()
