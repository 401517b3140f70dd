"""ASR task: Conformer encoder + attention LSTM decoder (LAS-style).

Reference: lingvo/tasks/asr/model.py:30 AsrModel, encoder.py:32,
decoder.py:48; Conformer blocks from lingvo/core/conformer_layer.py:471.
The Librispeech Conformer-L config (tasks/asr/params/librispeech.py
composes the LAS baseline; the Conformer-L encoder follows the Conformer
paper: 17 blocks, d=512, h=8, conv kernel 32) is the BASELINE.json
north-star model.
"""

from __future__ import annotations

import math

import torch

from lingvo_amd.core import py_utils
from lingvo_amd.core.base_input_generator import BaseSequenceInputGenerator
from lingvo_amd.core.base_layer import BaseLayer
from lingvo_amd.core.base_model import BaseTask
from lingvo_amd.core.nested_map import NestedMap
from lingvo_amd.layers import conformer as conformer_lib
from lingvo_amd.layers import layers as lingvo_layers
from lingvo_amd.layers import rnn_cell
from lingvo_amd.core import metrics as metrics_lib


class SyntheticAsrInput(BaseSequenceInputGenerator):
  """Synthetic Librispeech-shaped batches: [B, T, 80] log-mel + token ids
  (no network for real data; shapes match librispeech.py:42 80-dim
  features)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.batch_size = 16
    p.Define('frame_len', 1200, 'Input frames T (10ms frames).')
    p.Define('feature_dim', 80, 'Mel bins.')
    p.Define('target_len', 64, 'Target tokens L.')
    p.Define('vocab_size', 1024, 'WPM vocab size.')
    return p

  def _InputBatch(self) -> NestedMap:
    p = self.p
    g = torch.Generator().manual_seed(1000 + self._batch_count)
    b, t, f, l = p.batch_size, p.frame_len, p.feature_dim, p.target_len
    src = torch.randn(b, t, f, generator=g)
    src_lens = torch.randint(int(0.8 * t), t + 1, (b,), generator=g)
    src_paddings = py_utils.PaddingsFromLengths(src_lens, t)
    ids = torch.randint(2, p.vocab_size, (b, l), generator=g)
    tgt_lens = torch.randint(int(0.75 * l), l + 1, (b,), generator=g)
    tgt_paddings = py_utils.PaddingsFromLengths(tgt_lens, l)
    ids = (ids * (1 - tgt_paddings).long())
    return NestedMap(
        src=NestedMap(src_inputs=src, paddings=src_paddings),
        tgt=NestedMap(ids=ids, paddings=tgt_paddings,
                      labels=ids.roll(-1, dims=1) * (1 - tgt_paddings).long(),
                      weights=1.0 - tgt_paddings))


class ConformerEncoder(BaseLayer):
  """Subsampling frontend + N Conformer blocks
  (reference tasks/asr/encoder.py shape; core/conformer_layer.py blocks)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('input_dim', 80, 'Mel bins.')
    p.Define('model_dim', 512, 'Encoder dim.')
    p.Define('num_layers', 17, 'Conformer blocks.')
    p.Define('num_heads', 8, 'MHSA heads.')
    p.Define('kernel_size', 32, 'LConv kernel.')
    p.Define('dropout_prob', 0.1, 'Dropout.')
    p.Define('remat', False, 'Checkpoint each block.')
    p.Define('subsample_channels', 0, 'Frontend conv channels '
             '(0 = model_dim).')
    p.Define('conformer_tpl', conformer_lib.ConformerLayer.Params(),
             'Block template.')
    from lingvo_amd.layers import spectrum_augmenter
    p.Define('specaug_tpl', spectrum_augmenter.SpectrumAugmenter.Params(),
             'SpecAugment params; None disables.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    if p.specaug_tpl is not None:
      self.CreateChild('specaug', p.specaug_tpl)
    self.CreateChild('sub', conformer_lib.ConvSubsampling.Params().Set(
        input_freq_dim=p.input_dim, output_dim=p.model_dim,
        channels=p.subsample_channels or min(p.model_dim, 256)))
    blocks = []
    for i in range(p.num_layers):
      bp = p.conformer_tpl.Copy().Set(
          name=f'conformer_{i}', input_dim=p.model_dim,
          atten_num_heads=p.num_heads, kernel_size=p.kernel_size,
          dropout_prob=p.dropout_prob, remat=p.remat)
      blocks.append(bp)
    self.CreateChildren('blocks', blocks)

  def FProp(self, theta: NestedMap, src_inputs: torch.Tensor,
            paddings: torch.Tensor):
    x = src_inputs.to(self.fprop_dtype)
    if self.p.specaug_tpl is not None and not self.do_eval:
      x = self.specaug.FProp(theta.specaug, x, paddings)
    x, out_pad = self.sub.FProp(theta.sub, x, paddings)
    for i, block in enumerate(self.blocks):
      x = block.FProp(theta.blocks[i], x, out_pad)
    return x, out_pad

  # ---- streaming over the conformer stack (post-frontend chunks;
  # reference conformer StreamStep composition). The frontend runs
  # per-chunk upstream (ConvSubsampling is local: receptive field 7
  # input frames), streaming frontend fusion is a round-2 item.
  def InitStreamState(self, theta: NestedMap, batch: int, max_len: int,
                      device, dtype=torch.float32) -> NestedMap:
    return NestedMap(blocks=[
        b.InitStreamState(theta.blocks[i], batch, max_len, device, dtype)
        for i, b in enumerate(self.blocks)])

  def StreamStep(self, theta: NestedMap, feat_chunk: torch.Tensor,
                 paddings_chunk: torch.Tensor, state: NestedMap):
    """feat_chunk [B, C, model_dim]: already-subsampled features."""
    x = feat_chunk
    for i, block in enumerate(self.blocks):
      x, state.blocks[i] = block.StreamStep(
          theta.blocks[i], x, paddings_chunk, state.blocks[i])
    return x, state


class LasEncoder(BaseLayer):
  """LAS-style ASR encoder: conv subsampling frontend + stacked
  bidirectional LSTMs (reference tasks/asr/encoder.py:32 AsrEncoder and
  the Librispeech960Base config, librispeech.py:106-117: conv + 4x
  biLSTM-1024 with per-layer projection back to model_dim)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('input_dim', 80, 'Mel bins.')
    p.Define('model_dim', 1024, 'Per-direction LSTM dim * 2.')
    p.Define('num_lstm_layers', 4, 'biLSTM layers.')
    p.Define('dropout_prob', 0.0, 'Inter-layer dropout.')
    p.Define('subsample_channels', 32, 'Frontend conv channels.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    from lingvo_amd.layers import lstm_frnn_layer
    self.CreateChild('sub', conformer_lib.ConvSubsampling.Params().Set(
        input_freq_dim=p.input_dim, output_dim=p.model_dim,
        channels=p.subsample_channels))
    half = p.model_dim // 2
    layer_ps = []
    for i in range(p.num_lstm_layers):
      # Hoisted-input-projection biLSTM (reference lstm_frnn_layer.py):
      # one whole-sequence GEMM + a light scan with the fused K11 gate
      # kernel on GPU.
      cell = lstm_frnn_layer.LSTMCellSimpleExt.Params().Set(
          num_input_nodes=p.model_dim, num_output_nodes=half)
      layer_ps.append(lstm_frnn_layer.BidirectionalLstmFRNN.Params().Set(
          name=f'blstm_{i}', fwd=cell.Copy(), bak=cell.Copy()))
    self.CreateChildren('rnn', layer_ps)

  def FProp(self, theta: NestedMap, src_inputs: torch.Tensor,
            paddings: torch.Tensor):
    x = src_inputs.to(self.fprop_dtype)
    x, out_pad = self.sub.FProp(theta.sub, x, paddings)
    for i, layer in enumerate(self.rnn):
      y = layer.FProp(theta.rnn[i], x, out_pad)
      if self.p.dropout_prob and not self.do_eval:
        y = py_utils.DeterministicDropout(y, 1.0 - self.p.dropout_prob)
      x = y
    return x, out_pad


class AsrDecoder(BaseLayer):
  """Teacher-forced attention LSTM decoder (LAS-style,
  reference tasks/asr/decoder.py:48). Dot-product attention over
  projected encoder outputs, 2 LSTM layers, shared softmax."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('vocab_size', 1024, 'Output vocab.')
    p.Define('emb_dim', 128, 'Token embedding dim.')
    p.Define('rnn_cell_dim', 640, 'LSTM dim.')
    p.Define('num_lstm_layers', 2, 'LSTM layers.')
    p.Define('source_dim', 512, 'Encoder output dim.')
    p.Define('dropout_prob', 0.1, 'Dropout.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    self.CreateChild('emb', lingvo_layers.EmbeddingLayer.Params().Set(
        vocab_size=p.vocab_size, embedding_dim=p.emb_dim))
    self.CreateVariable('atten_query_w', py_utils.WeightParams(
        [p.rnn_cell_dim, p.source_dim], p.params_init, p.dtype))
    cells = []
    for i in range(p.num_lstm_layers):
      in_dim = (p.emb_dim + p.source_dim) if i == 0 else p.rnn_cell_dim
      cells.append(rnn_cell.LSTMCellSimple.Params().Set(
          name=f'lstm_{i}', num_input_nodes=in_dim,
          num_output_nodes=p.rnn_cell_dim))
    self.CreateChildren('rnns', cells)
    self.CreateChild('softmax', lingvo_layers.SimpleFullSoftmax.Params().Set(
        input_dim=p.rnn_cell_dim + p.source_dim, num_classes=p.vocab_size))

  def _Attend(self, theta: NestedMap, query_m: torch.Tensor,
              enc: torch.Tensor, enc_paddings: torch.Tensor):
    """Dot-product attention: query [B,H] x enc [B,S,D] -> ctx [B,D]."""
    q = torch.matmul(query_m, theta.atten_query_w)  # [B, D]
    logits = torch.einsum('bd,bsd->bs', q.float(), enc.float())
    logits = logits / math.sqrt(enc.shape[-1])
    logits = logits.masked_fill(enc_paddings > 0.5, -1e30)
    probs = torch.softmax(logits, dim=-1)
    return torch.einsum('bs,bsd->bd', probs, enc.float()).to(enc.dtype)

  def ComputePredictions(self, theta: NestedMap, enc: torch.Tensor,
                         enc_paddings: torch.Tensor,
                         targets: NestedMap) -> NestedMap:
    if self.p.num_lstm_layers == 2:
      return self._FastPredictions(theta, enc, enc_paddings, targets)
    return self._LoopPredictions(theta, enc, enc_paddings, targets)

  def _LoopPredictions(self, theta: NestedMap, enc: torch.Tensor,
                       enc_paddings: torch.Tensor,
                       targets: NestedMap) -> NestedMap:
    """Generic per-cell teacher-forced loop (any layer count); the
    numerics oracle for _FastPredictions."""
    p = self.p
    b, l = targets.ids.shape
    emb_all = self.emb.EmbLookup(theta.emb, targets.ids.long()).to(
        self.fprop_dtype)
    states = [c.InitState(b, enc.device, self.fprop_dtype)
              for c in self.rnns]
    ctx = torch.zeros(b, p.source_dim, device=enc.device,
                      dtype=self.fprop_dtype)
    outs = []
    for t in range(l):
      x = torch.cat([emb_all[:, t], ctx], dim=-1)
      for i, cell in enumerate(self.rnns):
        states[i] = cell.FProp(theta.rnns[i], states[i], NestedMap(act=x))
        x = states[i].m
      ctx = self._Attend(theta, x, enc, enc_paddings)
      outs.append(torch.cat([x, ctx], dim=-1))
    return NestedMap(atten_vecs=torch.stack(outs, dim=1))  # [B, L, H+D]

  def _FastPredictions(self, theta: NestedMap, enc: torch.Tensor,
                       enc_paddings: torch.Tensor,
                       targets: NestedMap) -> NestedMap:
    """Launch-lean teacher-forced loop (identical math to the cell path):
    embedding gate contributions precomputed in one GEMM; one GEMM per
    LSTM layer per step; bf16 bmm attention."""
    p = self.p
    dt = self.fprop_dtype
    b, l = targets.ids.shape
    h = p.rnn_cell_dim
    ids = targets.ids.long()
    emb_all = self.emb.EmbLookup(theta.emb, ids).to(dt)
    wm0, b0 = theta.rnns[0].wm, theta.rnns[0].b
    wm1, b1 = theta.rnns[1].wm, theta.rnns[1].b
    cap = self.rnns[0].p.cell_value_cap
    e_dim = p.emb_dim
    emb_gates = torch.matmul(emb_all, wm0[:e_dim]) + b0  # [B, L, 4H]
    w_cm0 = wm0[e_dim:]  # [(src + H), 4H]
    wq = theta.atten_query_w
    inv_sqrt_d = 1.0 / math.sqrt(p.source_dim)
    neg_mask = (enc_paddings.float() * -1e30).unsqueeze(1)  # [B,1,S]

    fgb = self.rnns[0].p.forget_gate_bias
    use_fused = enc.is_cuda and dt == torch.bfloat16

    if use_fused:
      # Whole-recurrence fused path (ops/las_decoder.py): per-step
      # small-M MFMA GEMMs + fused attention kernel, wgrads batched
      # over the L steps. Identical math to the loop below.
      from lingvo_amd.ops import las_decoder
      out = las_decoder.decoder_recurrence(
          emb_gates, w_cm0, b1, wm1, wq, enc,
          enc_paddings, fgb, cap if cap is not None else 0.0,
          p.source_dim)
      return NestedMap(atten_vecs=out)

    def lstm_pointwise(gates, c_prev):
      if use_fused:
        from lingvo_amd.ops import lstm_gates as lstm_ops
        return lstm_ops.lstm_gates(gates, c_prev, fgb,
                                   cap if cap is not None else 0.0)
      i_i, i_g, f_g, o_g = gates.split([h, h, h, h], dim=-1)
      if fgb:
        f_g = f_g + fgb
      c = torch.sigmoid(f_g) * c_prev + torch.sigmoid(i_g) * torch.tanh(i_i)
      if cap is not None:
        c = torch.clamp(c, -cap, cap)
      return c, torch.sigmoid(o_g) * torch.tanh(c)

    zeros = lambda d: torch.zeros(b, d, device=enc.device, dtype=dt)
    c0, m0, c1, m1 = zeros(h), zeros(h), zeros(h), zeros(h)
    ctx = zeros(p.source_dim)
    outs = []
    for t in range(l):
      gates0 = torch.addmm(emb_gates[:, t], torch.cat([ctx, m0], dim=-1),
                           w_cm0)
      c0, m0 = lstm_pointwise(gates0, c0)
      gates1 = torch.addmm(b1.unsqueeze(0), torch.cat([m0, m1], dim=-1),
                           wm1)
      c1, m1 = lstm_pointwise(gates1, c1)
      q = torch.matmul(m1, wq)  # [B, D]
      logits = torch.bmm(enc, q.unsqueeze(-1)).transpose(1, 2).float()
      logits = logits * inv_sqrt_d + neg_mask  # [B,1,S]
      probs = torch.softmax(logits, dim=-1).to(dt)
      ctx = torch.bmm(probs, enc).squeeze(1)  # [B, D]
      outs.append(torch.cat([m1, ctx], dim=-1))
    return NestedMap(atten_vecs=torch.stack(outs, dim=1))

  def ComputeLoss(self, theta: NestedMap, predictions: NestedMap,
                  targets: NestedMap):
    act = predictions.atten_vecs
    xent = self.softmax.XentLoss(
        theta.softmax, act, class_weights=targets.weights,
        class_ids=targets.labels)
    metrics = NestedMap(
        loss=(xent.avg_xent, xent.total_weight),
        log_pplx=(xent.avg_xent.detach(), xent.total_weight))
    return metrics, NestedMap(per_example_xent=xent.per_example_xent)

  def BeamSearchDecode(self, theta: NestedMap, enc: torch.Tensor,
                       enc_paddings: torch.Tensor,
                       num_hyps: int = 8, max_steps: int = 100,
                       length_norm: float = 0.0) -> NestedMap:
    """Beam search over the attention-LSTM decoder (reference LAS
    decoding, tasks/asr/decoder.py beam path): the generic helper
    drives per-step LSTM+attention state with tiled encoder outputs."""
    from lingvo_amd.core import beam_search_helper as bsh
    p = self.p
    helper = bsh.BeamSearchHelper(bsh.BeamSearchHelper.Params().Set(
        num_hyps_per_beam=num_hyps, max_steps=max_steps,
        length_normalization=length_norm))
    enc_t = enc.repeat_interleave(num_hyps, dim=0)
    pad_t = enc_paddings.repeat_interleave(num_hyps, dim=0)
    dt = self.fprop_dtype

    def init_fn(b, k):
      bk = b * k
      return NestedMap(
          cells=[c.InitState(bk, enc.device, dt) for c in self.rnns],
          ctx=torch.zeros(bk, p.source_dim, device=enc.device, dtype=dt))

    def step_fn(state, prev_ids):
      e = self.emb.EmbLookup(theta.emb, prev_ids.long()).to(dt)
      x = torch.cat([e, state.ctx], dim=-1)
      for i, cell in enumerate(self.rnns):
        state.cells[i] = cell.FProp(theta.rnns[i], state.cells[i],
                                    NestedMap(act=x))
        x = state.cells[i].m
      state.ctx = self._Attend(theta, x, enc_t, pad_t)
      logits = self.softmax.Logits(theta.softmax,
                                   torch.cat([x, state.ctx], dim=-1))
      return torch.log_softmax(logits.float(), dim=-1), state

    def reorder_fn(state, gather):
      for st in state.cells:
        for key, val in st.FlattenItems():
          if isinstance(val, torch.Tensor) and \
              val.shape[0] == gather.shape[0]:
            st.Set(key, val[gather])
      state.ctx = state.ctx[gather]
      return state

    return helper.BeamSearchDecode(enc.shape[0], init_fn, step_fn,
                                   reorder_fn)

  def GreedyDecode(self, theta: NestedMap, enc: torch.Tensor,
                   enc_paddings: torch.Tensor, max_len: int = 100,
                   sos_id: int = 1, eos_id: int = 2,
                   fusion=None) -> torch.Tensor:
    """Greedy decode; optional `fusion` (ShallowFusion) mixes an
    external LM's log-probs into each step (reference
    tasks/asr/fusion.py shallow fusion)."""
    p = self.p
    b = enc.shape[0]
    tok = torch.full((b,), sos_id, dtype=torch.long, device=enc.device)
    states = [c.InitState(b, enc.device, self.fprop_dtype)
              for c in self.rnns]
    ctx = torch.zeros(b, p.source_dim, device=enc.device,
                      dtype=self.fprop_dtype)
    done = torch.zeros(b, dtype=torch.bool, device=enc.device)
    fusion_state = fusion.InitState(b, enc.device) if fusion else None
    out = []
    for _ in range(max_len):
      e = self.emb.EmbLookup(theta.emb, tok).to(self.fprop_dtype)
      x = torch.cat([e, ctx], dim=-1)
      for i, cell in enumerate(self.rnns):
        states[i] = cell.FProp(theta.rnns[i], states[i], NestedMap(act=x))
        x = states[i].m
      ctx = self._Attend(theta, x, enc, enc_paddings)
      logits = self.softmax.Logits(theta.softmax,
                                   torch.cat([x, ctx], dim=-1))
      scores = torch.log_softmax(logits.float(), dim=-1)
      if fusion is not None:
        lm_scores, fusion_state = fusion.Score(tok, fusion_state)
        scores = scores + fusion.weight * lm_scores
      tok = scores.argmax(-1)
      tok = torch.where(done, torch.full_like(tok, eos_id), tok)
      done = done | (tok == eos_id)
      out.append(tok)
      if bool(done.all()):
        break
    return torch.stack(out, dim=1)


class ShallowFusion:
  """LM shallow fusion for decoding (reference tasks/asr/fusion.py):
  combined score = log p_am + weight * log p_lm, with the LM advanced
  token by token via its ExtendStep-style callback."""

  def __init__(self, lm_layer, lm_theta, weight: float = 0.3,
               max_len: int = 512):
    self.lm = lm_layer
    self.theta = lm_theta
    self.weight = weight
    self.max_len = max_len

  def InitState(self, batch: int, device) -> NestedMap:
    return NestedMap(ids=torch.zeros(batch, 0, dtype=torch.long,
                                     device=device))

  def Score(self, prev_tok: torch.Tensor, state: NestedMap):
    """Returns (log-probs [B, V], new state). Re-scores the growing
    prefix each step (CPU-oracle form; incremental KV-cache scoring is
    the GPU path once the LM exposes ExtendStep here)."""
    ids = torch.cat([state.ids, prev_tok.unsqueeze(1)], dim=1)
    act = self.lm.FProp(self.theta, ids,
                        torch.zeros(ids.shape, device=ids.device,
                                    dtype=torch.float32))
    logits = self.lm.softmax.Logits(
        getattr(self.theta, 'softmax'), act[:, -1])
    return torch.log_softmax(logits.float(), dim=-1), NestedMap(ids=ids)


class AsrModel(BaseTask):
  """Encoder/decoder ASR task (reference tasks/asr/model.py:30)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('encoder', ConformerEncoder.Params(), 'Encoder params.')
    p.Define('decoder', AsrDecoder.Params(), 'Decoder params.')
    p.Define('decode_num_hyps', 1,
             'Beam width for Decode(); 1 = greedy.')
    return p

  def __init__(self, params):
    super().__init__(params)
    self.CreateChild('encoder', self.p.encoder)
    self.CreateChild('decoder', self.p.decoder)

  def ComputePredictions(self, theta: NestedMap,
                         input_batch: NestedMap) -> NestedMap:
    enc, enc_pad = self.encoder.FProp(
        theta.encoder, input_batch.src.src_inputs,
        input_batch.src.paddings)
    preds = self.decoder.ComputePredictions(theta.decoder, enc, enc_pad,
                                            input_batch.tgt)
    preds.encoder_outputs = enc
    preds.encoder_paddings = enc_pad
    return preds

  def ComputeLoss(self, theta: NestedMap, predictions: NestedMap,
                  input_batch: NestedMap):
    metrics, per_example = self.decoder.ComputeLoss(
        theta.decoder, predictions, input_batch.tgt)
    b = input_batch.src.src_inputs.shape[0]
    metrics.num_samples_in_batch = (
        torch.tensor(float(b)), torch.ones(()))
    return metrics, per_example

  def Decode(self, input_batch: NestedMap) -> NestedMap:
    with torch.no_grad():
      enc, enc_pad = self.encoder.FProp(
          self.theta.encoder, input_batch.src.src_inputs,
          input_batch.src.paddings)
      if self.p.decode_num_hyps > 1:
        beam = self.decoder.BeamSearchDecode(
            self.theta.decoder, enc, enc_pad,
            num_hyps=self.p.decode_num_hyps)
        hyps = beam.topk_ids[:, 0]
      else:
        hyps = self.decoder.GreedyDecode(self.theta.decoder, enc,
                                         enc_pad)
    return NestedMap(topk_decoded=hyps,
                     transcripts=input_batch.tgt.ids)

  def Inference(self) -> NestedMap:
    """Named inference subgraphs (reference base_model.py:943)."""

    def default(src_inputs, paddings):
      enc, enc_pad = self.encoder.FProp(self.theta.encoder,
                                        src_inputs, paddings)
      hyps = self.decoder.GreedyDecode(self.theta.decoder, enc, enc_pad)
      return NestedMap(hyps=hyps)

    def encode(src_inputs, paddings):
      enc, enc_pad = self.encoder.FProp(self.theta.encoder,
                                        src_inputs, paddings)
      return NestedMap(encoded=enc, padding=enc_pad)

    return NestedMap(default=default, encode=encode)

  def CreateDecoderMetrics(self) -> NestedMap:
    return NestedMap(wer=metrics_lib.WerMetric(),
                     num_samples_in_batch=metrics_lib.AverageMetric())

  def PostProcessDecodeOut(self, decode_out: NestedMap,
                           decode_metrics: NestedMap) -> None:
    hyps = decode_out.topk_decoded
    refs = decode_out.transcripts
    for i in range(hyps.shape[0]):
      hyp = ' '.join(str(int(x)) for x in hyps[i] if int(x) > 2)
      ref = ' '.join(str(int(x)) for x in refs[i] if int(x) > 2)
      decode_metrics.wer.Update(ref, hyp)
    decode_metrics.num_samples_in_batch.Update(float(hyps.shape[0]))


class StreamingRecognizer:
  """End-to-end streaming ASR driver (reference: the streaming chunk
  mode of frontend.py:413 + conformer StreamStep composition).

  Audio chunks stream through the exact chunked Mel frontend; the
  accumulated mel features run through the (local) conv subsampling
  with the last output frame held back — 'same' right-edge padding only
  affects the final frame, so every released frame is final — and new
  subsampled frames stream through the conformer stack's StreamStep.
  Finish() flushes the held-back frame and greedy-decodes.

  Requires a causal encoder config (is_causal, conv_norm='layer').
  """

  def __init__(self, model: AsrModel, frontend, batch: int,
               max_enc_frames: int = 4096, device='cpu'):
    self.model = model
    self.frontend = frontend
    self.batch = batch
    self.device = device
    self.fe_state = frontend.InitStreamState(batch, device)
    self.enc_state = model.encoder.InitStreamState(
        model.theta.encoder, batch, max_enc_frames, device,
        torch.float32)
    self.mel = torch.zeros(batch, 0, frontend.p.num_bins, device=device)
    self.emitted = 0          # subsampled frames already streamed
    self.enc_frames = []

  def _SubAll(self, final: bool):
    enc = self.model.encoder
    t = self.mel.shape[1]
    if t == 0:
      return
    feats, out_pad = enc.sub.FProp(self.model.theta.encoder.sub,
                                   self.mel, torch.zeros(
                                       self.batch, t, device=self.device))
    avail = feats.shape[1] if final else max(0, feats.shape[1] - 1)
    if avail > self.emitted:
      new = feats[:, self.emitted:avail]
      out, self.enc_state = enc.StreamStep(
          self.model.theta.encoder, new,
          torch.zeros(self.batch, new.shape[1], device=self.device),
          self.enc_state)
      self.enc_frames.append(out)
      self.emitted = avail

  @torch.no_grad()
  def Push(self, wav_chunk: torch.Tensor) -> None:
    frames, self.fe_state = self.frontend.StreamStep(
        self.frontend.theta, wav_chunk, self.fe_state)
    if frames.shape[1]:
      self.mel = torch.cat([self.mel, frames], dim=1)
      self._SubAll(final=False)

  @torch.no_grad()
  def Finish(self) -> NestedMap:
    self._SubAll(final=True)
    enc = torch.cat(self.enc_frames, dim=1) if self.enc_frames else \
        torch.zeros(self.batch, 0, self.model.encoder.p.model_dim)
    pad = torch.zeros(self.batch, enc.shape[1], device=self.device)
    hyps = self.model.decoder.GreedyDecode(self.model.theta.decoder,
                                           enc, pad)
    return NestedMap(encoded=enc, hyps=hyps)


class AsrTfRecordInput(BaseSequenceInputGenerator):
  """Real-data ASR input: TFRecord shards of tf.train.Examples with
  'frames' (float list, T*feature_dim log-mel) and 'tokens' (int64
  list) features — the Librispeech export shape (reference
  tasks/asr/input_generator.py:24 AsrInput). Decoding runs through the
  C++ RecordYielder + the TF-free Example codec; batching buckets by
  frame count."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.batch_size = 16
    p.Define('files', [], 'TFRecord shards.')
    p.Define('feature_dim', 80, 'Mel bins per frame.')
    p.Define('target_len', 64, 'Max target tokens.')
    p.Define('input_seed', 301, 'Shuffle seed.')
    p.bucket_upper_bound = [1200]
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    from lingvo_amd.core import tf_example
    from lingvo_amd.core.generic_input import RecordBatcher
    from lingvo_amd.ops import _loader
    ext = _loader.get_ext(required=True)
    self._codec = tf_example
    self._yielder = ext.RecordYielder(list(p.files), 'tfrecord',
                                      p.input_seed, 1000, 2, True)
    limits = list(p.bucket_batch_limit) or \
        [p.batch_size] * len(p.bucket_upper_bound)

    def proc(rec):
      ex = self._codec.ParseExample(rec)
      frames = torch.tensor(ex['frames'], dtype=torch.float32)
      t = frames.numel() // p.feature_dim
      frames = frames.reshape(t, p.feature_dim)
      toks = torch.tensor(ex['tokens'], dtype=torch.long)
      return NestedMap(frames=frames, tokens=toks,
                       frame_len=torch.tensor([t]),
                       tok_len=torch.tensor([toks.numel()])), t

    self._batcher = RecordBatcher(self._yielder, proc,
                                  p.bucket_upper_bound, limits,
                                  num_threads=2)

  def _InputBatch(self) -> NestedMap:
    p = self.p
    batch = self._batcher.GetNext()
    assert batch is not None, 'input exhausted'
    b, tmax = batch.frames.shape[0], batch.frames.shape[1]
    src_pad = py_utils.PaddingsFromLengths(
        batch.frame_len.reshape(-1), tmax)
    lmax = min(p.target_len, int(batch.tok_len.max()) + 1)
    ids = torch.full((b, lmax), 2, dtype=torch.long)   # eos fill
    tgt_pad = torch.ones(b, lmax)
    for i in range(b):
      n = min(int(batch.tok_len[i]), lmax - 1)
      ids[i, 0] = 1                                    # sos
      ids[i, 1:n + 1] = batch.tokens[i, :n]
      tgt_pad[i, :n + 1] = 0.0
    labels = ids.roll(-1, dims=1)
    labels[:, -1] = 2
    return NestedMap(
        src=NestedMap(src_inputs=batch.frames, paddings=src_pad),
        tgt=NestedMap(ids=ids, paddings=tgt_pad,
                      labels=labels * (1 - tgt_pad).long(),
                      weights=1.0 - tgt_pad))

  def Stop(self):
    self._batcher.Stop()
    self._yielder.stop()
