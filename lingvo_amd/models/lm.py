"""Language-model tasks (reference lingvo/tasks/lm/model.py:25
LanguageModel, layers.py:495 RnnLm / TransformerLm).
"""

from __future__ import annotations

from typing import Optional

import torch

from lingvo_amd.core import py_utils
from lingvo_amd.core.base_input_generator import BaseSequenceInputGenerator
from lingvo_amd.core.base_layer import BaseLayer
from lingvo_amd.core.base_model import BaseTask
from lingvo_amd.core.nested_map import NestedMap
from lingvo_amd.layers import layers as lingvo_layers
from lingvo_amd.layers import rnn_cell
from lingvo_amd.layers import rnn_layers
from lingvo_amd.layers import transformer as transformer_lib


class SyntheticLmInput(BaseSequenceInputGenerator):
  """Synthetic token batches (reference
  tasks/lm/params/synthetic_packed_input.py:29 SyntheticTrain)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.batch_size = 8
    p.Define('seq_len', 1024, 'Sequence length.')
    p.Define('vocab_size', 32000, 'Vocab size.')
    return p

  def _InputBatch(self) -> NestedMap:
    p = self.p
    g = torch.Generator().manual_seed(7000 + self._batch_count)
    ids = torch.randint(1, p.vocab_size, (p.batch_size, p.seq_len),
                        generator=g)
    labels = ids.roll(-1, dims=1)
    weights = torch.ones(p.batch_size, p.seq_len)
    weights[:, -1] = 0.0
    return NestedMap(ids=ids, labels=labels,
                     paddings=torch.zeros(p.batch_size, p.seq_len),
                     weights=weights)


class TextFileLmInput(BaseSequenceInputGenerator):
  """Real-corpus LM input over the native C++ pipeline: TextLmBatcher
  (yield -> WPM tokenize -> bucket -> pad) feeds token batches without
  touching the GIL (reference BaseInputGeneratorFromFiles + the
  record_batcher/tokenizer op chain)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.batch_size = 16  # used as the per-bucket limit default
    p.Define('files', [], 'Text files, one example per line.')
    p.Define('tokens', [], 'WPM vocab pieces (index = id).')
    p.Define('num_threads', 2, 'Tokenizer worker threads.')
    p.Define('input_seed', 301, 'Shuffle seed.')
    p.bucket_upper_bound = [64]  # inherited bucketing params
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    from lingvo_amd.ops import _loader
    ext = _loader.get_ext(required=True)
    limits = list(p.bucket_batch_limit) or \
        [p.batch_size] * len(p.bucket_upper_bound)
    self._batcher = ext.TextLmBatcher(
        list(p.files), list(p.tokens), 0, 1, 2,
        list(p.bucket_upper_bound), limits, p.input_seed,
        p.num_threads, 10000, True)

  def _InputBatch(self) -> NestedMap:
    ids, labels, paddings = self._batcher.get_batch()
    return NestedMap(ids=ids, labels=labels, paddings=paddings,
                     weights=1.0 - paddings)

  def __del__(self):
    if hasattr(self, '_batcher'):
      self._batcher.stop()


class TransformerLm(BaseLayer):
  """Causal transformer LM (reference tasks/lm/layers.py TransformerLm /
  GPipeTransformerLm at one_billion_wds.py:181)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('vocab_size', 32000, 'Vocab.')
    p.Define('model_dim', 1024, 'Model dim.')
    p.Define('num_layers', 12, 'Layers.')
    p.Define('num_heads', 16, 'Heads.')
    p.Define('hidden_dim', 0, 'FFN hidden (0 = 4x).')
    p.Define('dropout_prob', 0.1, 'Dropout.')
    p.Define('shared_emb', True, 'Tie softmax and embedding weights.')
    p.Define('remat', False, 'Checkpoint layers.')
    p.Define('use_rope', False,
             'Rotary position embedding in self-attention (replaces '
             'the additive sinusoidal positions).')
    p.Define('moe_every_n', 0, 'Every n-th layer uses MoE FFN.')
    p.Define('num_experts', 0, 'MoE experts (when moe_every_n > 0).')
    p.Define('expert_capacity_factor', 2.0, 'MoE capacity factor.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    if p.shared_emb:
      self.CreateChild('softmax',
                       lingvo_layers.SharedSoftmaxLayer.Params().Set(
                           input_dim=p.model_dim, num_classes=p.vocab_size))
    else:
      self.CreateChild('emb', lingvo_layers.EmbeddingLayer.Params().Set(
          vocab_size=p.vocab_size, embedding_dim=p.model_dim,
          scale_sqrt_depth=True))
      self.CreateChild('softmax',
                       lingvo_layers.SimpleFullSoftmax.Params().Set(
                           input_dim=p.model_dim, num_classes=p.vocab_size))
    self.CreateChild('pos_emb',
                     lingvo_layers.PositionalEmbeddingLayer.Params().Set(
                         embedding_dim=p.model_dim))
    stack_p = transformer_lib.StackedTransformerLayers.Params().Set(
        model_dim=p.model_dim, num_layers=p.num_layers,
        num_heads=p.num_heads, hidden_dim=p.hidden_dim,
        mask_self_atten=True, remat=p.remat)
    if p.use_rope:
      stack_p.transformer_tpl.tr_atten_tpl.atten_tpl.use_rope = True
    # propagate GShard-style sharding annotations to the stack so
    # LowerShardingAnnotations can rewrite it to explicit TP layers
    if p.weight_split_dims_mapping is not None:
      stack_p.weight_split_dims_mapping = p.weight_split_dims_mapping
      stack_p.device_mesh = p.device_mesh
    stack_p.transformer_tpl.tr_atten_tpl.residual_dropout_prob = \
        p.dropout_prob
    stack_p.transformer_tpl.tr_fflayer_tpl.residual_dropout_prob = \
        p.dropout_prob
    stack_p.transformer_tpl.tr_fflayer_tpl.relu_dropout_prob = \
        p.dropout_prob
    if p.moe_every_n:
      stack_p.moe_every_n = p.moe_every_n
      stack_p.moe_tpl.num_experts = p.num_experts
      stack_p.moe_tpl.expert_capacity_factor = p.expert_capacity_factor
      stack_p.moe_tpl.residual_dropout_prob = p.dropout_prob
    self.CreateChild('stack', stack_p)

  def _Emb(self, theta, ids):
    if self.p.shared_emb:
      return self.softmax.EmbLookup(theta.softmax, ids)
    return self.emb.EmbLookup(theta.emb, ids)

  def FProp(self, theta: NestedMap, ids: torch.Tensor,
            paddings: torch.Tensor,
            segment_ids: Optional[torch.Tensor] = None,
            segment_pos: Optional[torch.Tensor] = None) -> torch.Tensor:
    """Packed inputs: segment_ids [B,T] block cross-segment attention,
    segment_pos [B,T] restarts positions per segment (reference
    PackedBatchMajorLanguageModel, tasks/lm/model.py:408)."""
    x = self._Emb(theta, ids.long()).to(self.fprop_dtype)
    if not self.p.use_rope:
      pos = self.pos_emb.FProp(theta.pos_emb, ids.shape[1],
                               device=ids.device)
      if segment_pos is not None:
        x = x + pos[segment_pos.long()].to(x.dtype)
      else:
        x = x + pos.unsqueeze(0).to(x.dtype)
    if self.p.dropout_prob and not self.do_eval:
      x = py_utils.DeterministicDropout(x, 1.0 - self.p.dropout_prob)
    return self.stack.FProp(theta.stack, x, paddings,
                            segment_ids=segment_ids)

  def XentLoss(self, theta, act, labels, weights):
    return self.softmax.XentLoss(theta.softmax, act, class_weights=weights,
                                 class_ids=labels)

  @torch.no_grad()
  def Generate(self, theta: NestedMap, prefix: torch.Tensor,
               max_new: int, eos_id: int = 2) -> torch.Tensor:
    """Greedy incremental decode over the stack's KV cache
    (ExtendStep): each new token costs one single-position pass instead
    of re-running the whole prefix. Matches full-FProp greedy exactly.
    """
    b, t0 = prefix.shape
    total = t0 + max_new
    states = self.stack.InitStates(theta.stack, b, total,
                                   prefix.device, self.fprop_dtype)
    pos = self.pos_emb.FProp(theta.pos_emb, total, device=prefix.device)
    ids = prefix.clone()
    done = torch.zeros(b, dtype=torch.bool, device=prefix.device)
    tok = None
    for t in range(total - 1):
      cur = prefix[:, t] if t < t0 else tok
      x = self._Emb(theta, cur.long()).to(self.fprop_dtype)
      if not self.p.use_rope:
        x = x + pos[t].to(x.dtype)
      x = x.unsqueeze(1)                                 # [B, 1, D]
      act, states = self.stack.ExtendStep(theta.stack, x, states)
      if t < t0 - 1:
        continue  # prefill: just populate the cache
      logits = self.softmax.Logits(theta.softmax, act[:, 0])
      tok = logits.argmax(-1)
      tok = torch.where(done, torch.full_like(tok, eos_id), tok)
      done = done | (tok == eos_id)
      ids = torch.cat([ids, tok.unsqueeze(1)], dim=1)
      if bool(done.all()):
        break
    return ids


class RnnLm(BaseLayer):
  """LSTM LM (reference tasks/lm/layers.py:495 RnnLm; the
  WordLevelOneBwdsSimpleSampledSoftmax baseline model)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('vocab_size', 32000, 'Vocab.')
    p.Define('emb_dim', 1024, 'Embedding dim.')
    p.Define('rnn_dims', [2048, 2048], 'Per-layer LSTM dims.')
    p.Define('rnn_proj', 1024, 'LSTM projection dim (0 = none).')
    p.Define('dropout_prob', 0.1, 'Dropout.')
    p.Define('num_sampled', 0, 'Sampled-softmax negatives (training).')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    self.CreateChild('emb', lingvo_layers.EmbeddingLayer.Params().Set(
        vocab_size=p.vocab_size, embedding_dim=p.emb_dim))
    cells = []
    in_dim = p.emb_dim
    for i, d in enumerate(p.rnn_dims):
      out_dim = p.rnn_proj or d
      cells.append(rnn_cell.LSTMCellSimple.Params().Set(
          name=f'lstm_{i}', num_input_nodes=in_dim, num_output_nodes=out_dim,
          num_hidden_nodes=d if p.rnn_proj else 0))
      in_dim = out_dim
    self.CreateChild('rnns', rnn_layers.StackedFRNNLayerByLayer.Params().Set(
        cell_tpl=cells, skip_start=1))
    self.CreateChild('softmax', lingvo_layers.SimpleFullSoftmax.Params().Set(
        input_dim=in_dim, num_classes=p.vocab_size,
        num_sampled=p.num_sampled))

  def FProp(self, theta, ids, paddings):
    x = self.emb.EmbLookup(theta.emb, ids.long()).to(self.fprop_dtype)
    if self.p.dropout_prob and not self.do_eval:
      x = py_utils.DeterministicDropout(x, 1.0 - self.p.dropout_prob)
    return self.rnns.FProp(theta.rnns, x, paddings)

  def XentLoss(self, theta, act, labels, weights):
    return self.softmax.XentLoss(theta.softmax, act, class_weights=weights,
                                 class_ids=labels)


class LanguageModel(BaseTask):
  """LM task (reference tasks/lm/model.py:25)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('lm', TransformerLm.Params(), 'LM layer params.')
    return p

  def __init__(self, params):
    super().__init__(params)
    self.CreateChild('lm', self.p.lm)

  def ComputePredictions(self, theta, input_batch):
    kwargs = {}
    if input_batch.Get('segment_ids') is not None:
      kwargs = dict(segment_ids=input_batch.segment_ids,
                    segment_pos=input_batch.Get('segment_pos'))
    act = self.lm.FProp(theta.lm, input_batch.ids, input_batch.paddings,
                        **kwargs)
    return NestedMap(activations=act)

  def ComputeLoss(self, theta, predictions, input_batch):
    xent = self.lm.XentLoss(theta.lm, predictions.activations,
                            input_batch.labels, input_batch.weights)
    num_toks = xent.total_weight
    b = input_batch.ids.shape[0]
    loss = xent.avg_xent
    from lingvo_amd.parallel.moe import MoEFeedForwardLayer
    aux = [m.AuxLoss() for m in self.modules()
           if isinstance(m, MoEFeedForwardLayer) and
           m.AuxLoss() is not None]
    if aux:
      aux_total = torch.stack([a.float() for a in aux]).sum()
      loss = loss + aux_total
    metrics = NestedMap(
        loss=(loss, num_toks),
        log_pplx=(xent.avg_xent.detach(), num_toks),
        num_samples_in_batch=(torch.tensor(float(b)), torch.ones(())),
        tokens_per_batch=(num_toks.detach(), torch.ones(())))
    return metrics, NestedMap(per_example_xent=xent.per_example_xent)

  def Decode(self, input_batch):
    with torch.no_grad():
      act = self.lm.FProp(self.theta.lm, input_batch.ids,
                          input_batch.paddings)
      xent = self.lm.XentLoss(self.theta.lm, act, input_batch.labels,
                              input_batch.weights)
    return NestedMap(log_pplx=xent.avg_xent.reshape(1))

  def Inference(self) -> NestedMap:
    """Scoring subgraph (reference base_model.py:943)."""

    def default(ids, paddings):
      act = self.lm.FProp(self.theta.lm, ids, paddings)
      labels = ids.roll(-1, dims=1)
      weights = (1.0 - paddings)
      xent = self.lm.XentLoss(self.theta.lm, act, labels, weights)
      return NestedMap(per_example_xent=xent.per_example_xent,
                       avg_xent=xent.avg_xent)

    def generate(prefix, max_new):
      n = int(max_new.reshape(-1)[0]) if isinstance(
          max_new, torch.Tensor) else int(max_new)
      ids = self.lm.Generate(self.theta.lm, prefix.long(), n)
      return NestedMap(ids=ids)

    return NestedMap(default=default, generate=generate)


class InsertionLm(BaseTask):
  """Insertion-based LM (reference core/insertion.py consumers /
  KERMIT): sample a canvas from each sequence, encode it
  BIDIRECTIONALLY, and train per-slot content predictions for the
  missing symbols (slot j = insert before canvas position j; an
  appended learned END position covers slot C)."""

  @classmethod
  def Params(cls):
    from lingvo_amd.core import insertion
    p = super().Params()
    p.Define('vocab_size', 32000, 'Vocab.')
    p.Define('model_dim', 256, 'Model dim.')
    p.Define('num_layers', 4, 'Bidirectional layers.')
    p.Define('num_heads', 4, 'Heads.')
    p.Define('insertion_tpl', insertion.SymbolInsertionLayer.Params(),
             'Canvas sampler.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    self.CreateChild('insertion', p.insertion_tpl)
    self.CreateChild('emb', lingvo_layers.EmbeddingLayer.Params().Set(
        vocab_size=p.vocab_size, embedding_dim=p.model_dim,
        scale_sqrt_depth=True))
    self.CreateChild('pos_emb',
                     lingvo_layers.PositionalEmbeddingLayer.Params().Set(
                         embedding_dim=p.model_dim))
    self.CreateVariable('end_emb', py_utils.WeightParams(
        [p.model_dim], py_utils.WeightInit.Gaussian(0.02), p.dtype))
    self.CreateChild(
        'stack', transformer_lib.StackedTransformerLayers.Params().Set(
            model_dim=p.model_dim, num_layers=p.num_layers,
            num_heads=p.num_heads, mask_self_atten=False))
    self.CreateChild('softmax',
                     lingvo_layers.SimpleFullSoftmax.Params().Set(
                         input_dim=p.model_dim, num_classes=p.vocab_size))

  def ComputePredictions(self, theta, input_batch):
    p = self.p
    rollin = self.insertion.FProp(theta.insertion, input_batch.ids,
                                  input_batch.paddings)
    canvas, cpad = rollin.canvas, rollin.canvas_paddings
    b, c = canvas.shape
    x = self.emb.EmbLookup(theta.emb, canvas.long()).to(self.fprop_dtype)
    # append the learned END slot position
    end = theta.end_emb.reshape(1, 1, -1).expand(b, 1, -1).to(x.dtype)
    x = torch.cat([x, end], dim=1)
    pad = torch.cat([cpad, torch.zeros(b, 1, device=x.device)], dim=1)
    pos = self.pos_emb.FProp(theta.pos_emb, c + 1, device=x.device)
    x = x + pos.unsqueeze(0).to(x.dtype)
    act = self.stack.FProp(theta.stack, x, pad)
    return NestedMap(slot_acts=act, rollin=rollin)

  def ComputeLoss(self, theta, predictions, input_batch):
    tgt = predictions.rollin.target_indices   # [N, 3] (b, slot, symbol)
    act = predictions.slot_acts
    if tgt.shape[0] == 0:
      zero = act.sum() * 0.0
      w = torch.ones(())
      return NestedMap(loss=(zero, w),
                       num_samples_in_batch=(w, w)), NestedMap()
    gathered = act[tgt[:, 0], tgt[:, 1]]      # [N, D]
    logits = self.softmax.Logits(theta.softmax, gathered).float()
    loss = torch.nn.functional.cross_entropy(logits, tgt[:, 2])
    w = torch.tensor(float(tgt.shape[0]))
    metrics = NestedMap(
        loss=(loss, w),
        num_samples_in_batch=(
            torch.tensor(float(input_batch.ids.shape[0])),
            torch.ones(())))
    return metrics, NestedMap()
