"""Core layer library, tranche 1 (reference lingvo/core/layers.py).

ProjectionLayer (reference layers.py:845), FCLayer (:1586), FeedForwardNet
(:1597), Conv2DLayer (:182-771), PoolingLayer (:2285), EmbeddingLayer
(:2679), positional embeddings (:3380-3476), SimpleFullSoftmax (:3697),
DropoutLayer (:4842), LayerNorm (:4927), label smoothing (:5305).
GEMMs run on hipBLASLt via torch.matmul; LayerNorm / softmax-xent /
embedding dispatch to hand-written gfx950 HIP kernels on ROCm devices
(lingvo_amd/ops), plain torch on CPU.
"""

from __future__ import annotations

import math
from typing import List, Optional, Sequence

import torch
import torch.nn.functional as F

from lingvo_amd.core import py_utils
from lingvo_amd.core.base_layer import BaseLayer
from lingvo_amd.core.hyperparams import Params
from lingvo_amd.core.nested_map import NestedMap
from lingvo_amd.layers import activations


class ProjectionLayer(BaseLayer):
  """Linear projection + optional bias + activation. Optional
  quantization-aware training via qdomain_tpl (reference
  layers.py:845 ProjectionLayer QWeight/QAct wiring)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('input_dim', 0, 'Input dimension.')
    p.Define('output_dim', 0, 'Output dimension.')
    p.Define('has_bias', False, 'Add bias.')
    p.Define('activation', 'NONE', 'Activation name.')
    p.Define('qdomain_tpl', None,
             'QDomain params; None disables fake quantization.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    assert p.input_dim > 0 and p.output_dim > 0
    self.CreateVariable('w', py_utils.WeightParams(
        [p.input_dim, p.output_dim], p.params_init, p.dtype))
    if p.has_bias:
      self.CreateVariable('b', py_utils.WeightParams(
          [p.output_dim], py_utils.WeightInit.Constant(0.0), p.dtype))
    if p.qdomain_tpl is not None:
      self.CreateChild('qdomain', p.qdomain_tpl)

  def FProp(self, theta: NestedMap, inputs: torch.Tensor,
            paddings: Optional[torch.Tensor] = None) -> torch.Tensor:
    p = self.p
    w = theta.w
    if p.qdomain_tpl is not None:
      w = self.qdomain.QuantizeWeight(w)
    out = py_utils.MatmulBias(inputs, w,
                              theta.b if p.has_bias else None)
    out = activations.GetFn(p.activation)(out)
    if p.qdomain_tpl is not None:
      out = self.qdomain.QuantizeTensor(out, calibrate=True)
    if paddings is not None:
      out = py_utils.ApplyPadding(paddings, out)
    return out

  def PostTrainingStepUpdate(self, global_step: int) -> None:
    if self.p.qdomain_tpl is not None:
      self.qdomain.SetStep(global_step)
    super().PostTrainingStepUpdate(global_step)


class FCLayer(ProjectionLayer):
  """Fully-connected = projection with bias + RELU default
  (reference layers.py:1586)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.has_bias = True
    p.activation = 'RELU'
    return p


class FeedForwardNet(BaseLayer):
  """Stack of FC layers (reference layers.py:1597)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('input_dim', 0, 'Input dimension.')
    p.Define('hidden_layer_dims', [], 'Output dim of each layer.')
    p.Define('activation', 'RELU', 'One name, or list per layer.')
    p.Define('has_bias', True, 'Bias on each layer.')
    p.Define('dropout_prob', 0.0, 'Dropout after each hidden layer.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    dims = [p.input_dim] + list(p.hidden_layer_dims)
    acts = p.activation
    if isinstance(acts, str):
      acts = [acts] * (len(dims) - 1)
    layers_p = []
    for i in range(len(dims) - 1):
      layers_p.append(ProjectionLayer.Params().Set(
          input_dim=dims[i], output_dim=dims[i + 1],
          has_bias=p.has_bias, activation=acts[i]))
    self.CreateChildren('fc', layers_p)
    self._dropout = p.dropout_prob

  def FProp(self, theta: NestedMap, inputs: torch.Tensor) -> torch.Tensor:
    out = inputs
    for i, layer in enumerate(self.fc):
      out = layer.FProp(theta.fc[i], out)
      if self._dropout and not self.do_eval:
        out = py_utils.DeterministicDropout(out, 1.0 - self._dropout)
    return out


class Conv2DLayer(BaseLayer):
  """NHWC conv + optional BN-free bias + activation
  (reference layers.py:182-771, simplified: no weight norm)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('filter_shape', (3, 3, 1, 32),
             '(H, W, in_channels, out_channels).')
    p.Define('filter_stride', (1, 1), '(stride_h, stride_w).')
    p.Define('padding', 'SAME', 'SAME or VALID.')
    p.Define('has_bias', True, 'Bias add.')
    p.Define('activation', 'RELU', 'Activation.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    fh, fw, cin, cout = p.filter_shape
    self.CreateVariable('w', py_utils.WeightParams(
        [fh, fw, cin, cout], p.params_init, p.dtype))
    if p.has_bias:
      self.CreateVariable('b', py_utils.WeightParams(
          [cout], py_utils.WeightInit.Constant(0.0), p.dtype))

  def FProp(self, theta: NestedMap, inputs: torch.Tensor) -> torch.Tensor:
    """inputs: [B, H, W, C] (batch-major NHWC like the reference)."""
    p = self.p
    x = inputs.permute(0, 3, 1, 2)  # NCHW for torch/MIOpen
    w = theta.w.permute(3, 2, 0, 1)  # OIHW
    fh, fw = p.filter_shape[:2]
    sh, sw = p.filter_stride
    if p.padding == 'SAME':
      ih, iw = x.shape[2], x.shape[3]
      oh = (ih + sh - 1) // sh
      ow = (iw + sw - 1) // sw
      pad_h = max(0, (oh - 1) * sh + fh - ih)
      pad_w = max(0, (ow - 1) * sw + fw - iw)
      x = F.pad(x, (pad_w // 2, pad_w - pad_w // 2,
                    pad_h // 2, pad_h - pad_h // 2))
    out = F.conv2d(x, w, theta.b if p.has_bias else None, stride=(sh, sw))
    out = activations.GetFn(p.activation)(out)
    return out.permute(0, 2, 3, 1)


class PoolingLayer(BaseLayer):
  """Max/avg pooling, NHWC (reference layers.py:2285)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('window_shape', (2, 2), 'Pool window (H, W).')
    p.Define('window_stride', (2, 2), 'Stride (H, W).')
    p.Define('pooling_type', 'MAX', 'MAX or AVG.')
    return p

  def FProp(self, theta: NestedMap, inputs: torch.Tensor) -> torch.Tensor:
    p = self.p
    x = inputs.permute(0, 3, 1, 2)
    if p.pooling_type == 'MAX':
      out = F.max_pool2d(x, p.window_shape, p.window_stride)
    else:
      out = F.avg_pool2d(x, p.window_shape, p.window_stride)
    return out.permute(0, 2, 3, 1)


class EmbeddingLayer(BaseLayer):
  """Token embedding with scatter-add bwd HIP kernel on GPU
  (reference layers.py:2679 SimpleEmbeddingLayer)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('vocab_size', 0, 'Vocabulary size.')
    p.Define('embedding_dim', 0, 'Embedding dimension.')
    p.Define('scale_sqrt_depth', False,
             'Scale outputs by sqrt(dim) (transformer convention).')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    self.CreateVariable('wm', py_utils.WeightParams(
        [p.vocab_size, p.embedding_dim],
        py_utils.WeightInit.Gaussian(1.0 / math.sqrt(p.embedding_dim)),
        p.dtype))

  def EmbLookup(self, theta: NestedMap, ids: torch.Tensor) -> torch.Tensor:
    scale = (self.p.embedding_dim ** 0.5 if self.p.scale_sqrt_depth
             else 1.0)
    from lingvo_amd.ops import embedding as emb_ops
    return emb_ops.embedding_lookup(theta.wm, ids, scale)

  def FProp(self, theta: NestedMap, ids: torch.Tensor) -> torch.Tensor:
    return self.EmbLookup(theta, ids)


class PositionalEmbeddingLayer(BaseLayer):
  """Sinusoidal positions (reference layers.py:3380)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('embedding_dim', 0, 'Dimension.')
    p.Define('min_timescale', 1, 'Min timescale.')
    p.Define('max_timescale', 10_000, 'Max timescale.')
    return p

  def FProp(self, theta: NestedMap, seq_length: int,
            device=None) -> torch.Tensor:
    p = self.p
    pos = torch.arange(seq_length, dtype=torch.float32, device=device)
    num_ts = p.embedding_dim // 2
    log_inc = math.log(p.max_timescale / p.min_timescale) / max(1, num_ts - 1)
    inv_ts = p.min_timescale * torch.exp(
        torch.arange(num_ts, dtype=torch.float32, device=device) * -log_inc)
    scaled = pos[:, None] * inv_ts[None, :]
    emb = torch.cat([torch.sin(scaled), torch.cos(scaled)], dim=1)
    if p.embedding_dim % 2:
      emb = F.pad(emb, (0, 1))
    return emb.to(self.fprop_dtype)


class RotaryPositionalEmbeddingLayer(BaseLayer):
  """Rotary position embedding (reference layers.py:3476
  RotaryPositionalEmbeddingLayer; RoFormer arXiv:2104.09864).

  Rotates each head-dim pair (x[2i], x[2i+1]) by pos * theta_i where
  theta_i spans [1/min_timescale, 1/max_timescale] geometrically, so
  q·k after rotation depends only on relative position.
  """

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('embedding_dim', 0, 'Per-head dim H (must be even).')
    p.Define('min_timescale', 1, 'Min timescale.')
    p.Define('max_timescale', 10_000, 'Max timescale.')
    return p

  def _SinCos(self, seq_length, device, position=None):
    p = self.p
    half = p.embedding_dim // 2
    frac = torch.arange(half, dtype=torch.float32, device=device) / half
    timescale = p.min_timescale * (p.max_timescale / p.min_timescale) ** frac
    if position is None:
      position = torch.arange(seq_length, dtype=torch.float32,
                              device=device)[None, :]
    angles = position.float()[:, :, None] / timescale[None, None, :]
    return torch.sin(angles), torch.cos(angles)  # [B|1, T, H/2]

  def FProp(self, theta: NestedMap, inputs: torch.Tensor,
            position: Optional[torch.Tensor] = None) -> torch.Tensor:
    """inputs [B, T, N, H]; optional position [B, T] (decode offsets)."""
    h = inputs.shape[-1]
    assert h == self.p.embedding_dim and h % 2 == 0, h
    sin, cos = self._SinCos(inputs.shape[1], inputs.device, position)
    sin = sin[:, :, None, :]  # broadcast over heads
    cos = cos[:, :, None, :]
    x1, x2 = inputs.float().chunk(2, dim=-1)
    out = torch.cat([x1 * cos - x2 * sin, x2 * cos + x1 * sin], dim=-1)
    return out.to(inputs.dtype)


class LayerNorm(BaseLayer):
  """Layer normalization over the last dim (reference layers.py:4927).

  GPU: fused one-pass HIP kernel; CPU: torch reference.
  """

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('input_dim', 0, 'Normalized dimension.')
    p.Define('epsilon', 1e-6, 'Epsilon.')
    p.Define('use_fused_layernorm', True, 'Use the HIP kernel on GPU.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    self.CreateVariable('scale', py_utils.WeightParams(
        [p.input_dim], py_utils.WeightInit.Constant(0.0), p.dtype))
    self.CreateVariable('bias', py_utils.WeightParams(
        [p.input_dim], py_utils.WeightInit.Constant(0.0), p.dtype))

  def FProp(self, theta: NestedMap, inputs: torch.Tensor) -> torch.Tensor:
    p = self.p
    # Reference convention: scale is (1 + scale) so init 0 == identity.
    if inputs.is_cuda and p.use_fused_layernorm:
      from lingvo_amd.ops import layer_norm as ln_ops
      return ln_ops.layer_norm(inputs, theta.scale, theta.bias, p.epsilon)
    x = inputs.float()
    mean = x.mean(dim=-1, keepdim=True)
    var = x.var(dim=-1, unbiased=False, keepdim=True)
    out = (x - mean) * torch.rsqrt(var + p.epsilon)
    out = out * (1.0 + theta.scale.float()) + theta.bias.float()
    return out.to(inputs.dtype)


class RmsNorm(BaseLayer):
  """RMS normalization (no mean subtraction)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('input_dim', 0, 'Normalized dimension.')
    p.Define('epsilon', 1e-6, 'Epsilon.')
    return p

  def __init__(self, params):
    super().__init__(params)
    self.CreateVariable('scale', py_utils.WeightParams(
        [self.p.input_dim], py_utils.WeightInit.Constant(0.0), self.p.dtype))

  def FProp(self, theta: NestedMap, inputs: torch.Tensor) -> torch.Tensor:
    if inputs.is_cuda:
      from lingvo_amd.ops import layer_norm as ln_ops
      return ln_ops.rms_norm(inputs, theta.scale, self.p.epsilon)
    x = inputs.float()
    ms = x.pow(2).mean(dim=-1, keepdim=True)
    out = x * torch.rsqrt(ms + self.p.epsilon) * (1.0 + theta.scale.float())
    return out.to(inputs.dtype)


class DropoutLayer(BaseLayer):
  """Deterministic dropout under StepSeedScope (reference layers.py:4916)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('keep_prob', 1.0, 'Keep probability.')
    return p

  def FProp(self, theta: NestedMap, inputs: torch.Tensor) -> torch.Tensor:
    if self.do_eval or self.p.keep_prob >= 1.0:
      return inputs
    return py_utils.DeterministicDropout(inputs, self.p.keep_prob)


class _ChunkedLogitsXent(torch.autograd.Function):
  """Streaming vocab-chunked logits+xent: peak extra memory is
  [R, chunk] instead of [R, V] in both directions."""

  @staticmethod
  def forward(ctx, x, w, b, labels, chunk):
    xf = x.float()
    r = xf.shape[0]
    v = w.shape[1]
    run_max = torch.full((r,), -1e30, device=x.device)
    run_sum = torch.zeros(r, device=x.device)
    tgt = torch.zeros(r, device=x.device)
    for c0 in range(0, v, chunk):
      c1 = min(v, c0 + chunk)
      logits = torch.addmm(b[c0:c1].float(), xf, w[:, c0:c1].float())
      m = logits.max(dim=1).values
      new_max = torch.maximum(run_max, m)
      run_sum = run_sum * torch.exp(run_max - new_max) + \
          torch.exp(logits - new_max.unsqueeze(1)).sum(dim=1)
      run_max = new_max
      in_chunk = (labels >= c0) & (labels < c1)
      if bool(in_chunk.any()):
        idx = in_chunk.nonzero(as_tuple=True)[0]
        tgt[idx] = logits[idx, labels[idx] - c0]
    lse = run_max + torch.log(run_sum)
    ctx.save_for_backward(x, w, b, labels, lse)
    ctx.chunk = chunk
    return lse - tgt

  @staticmethod
  def backward(ctx, dy):
    x, w, b, labels, lse = ctx.saved_tensors
    chunk = ctx.chunk
    xf = x.float()
    v = w.shape[1]
    dyf = dy.float().unsqueeze(1)
    dx = torch.zeros_like(xf)
    dw = torch.zeros_like(w, dtype=torch.float32)
    db = torch.zeros(v, device=x.device)
    for c0 in range(0, v, chunk):
      c1 = min(v, c0 + chunk)
      logits = torch.addmm(b[c0:c1].float(), xf, w[:, c0:c1].float())
      p = torch.exp(logits - lse.unsqueeze(1))
      in_chunk = (labels >= c0) & (labels < c1)
      if bool(in_chunk.any()):
        idx = in_chunk.nonzero(as_tuple=True)[0]
        p[idx, labels[idx] - c0] -= 1.0
      p = p * dyf
      dx += p @ w[:, c0:c1].float().t()
      dw[:, c0:c1] = xf.t() @ p
      db[c0:c1] = p.sum(dim=0)
    return (dx.to(x.dtype), dw.to(w.dtype), db.to(b.dtype), None, None)


class SimpleFullSoftmax(BaseLayer):
  """Softmax + cross-entropy over a full vocab (reference layers.py:3697).

  XentLoss returns a NestedMap with total_xent/total_weight/per_example.
  On GPU the fused logits+log-softmax+gather HIP path avoids
  materializing [B*T, V] probabilities.
  """

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('input_dim', 0, 'Input dim.')
    p.Define('num_classes', 0, 'Number of classes.')
    p.Define('chunk_size', 0, 'If >0, compute xent in vocab chunks.')
    p.Define('num_sampled', 0,
             'If >0, sampled softmax with this many negatives during '
             'training (reference layers.py:3697 sampled support; the '
             'WordLevelOneBwdsSimpleSampledSoftmax baseline).')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    self.CreateVariable('linear_w', py_utils.WeightParams(
        [p.input_dim, p.num_classes], p.params_init, p.dtype))
    self.CreateVariable('bias', py_utils.WeightParams(
        [p.num_classes], py_utils.WeightInit.Constant(0.0), p.dtype))

  def Logits(self, theta: NestedMap, inputs: torch.Tensor) -> torch.Tensor:
    return py_utils.MatmulBias(inputs, theta.linear_w, theta.bias)

  def XentLossFromLogits(self, logits: torch.Tensor,
                         class_ids: Optional[torch.Tensor] = None,
                         class_probabilities: Optional[torch.Tensor] = None
                         ) -> NestedMap:
    log_probs = F.log_softmax(logits.float(), dim=-1)
    if class_probabilities is not None:
      per_example = -(class_probabilities.float() * log_probs).sum(-1)
    else:
      per_example = F.nll_loss(log_probs, class_ids.reshape(-1).long(),
                               reduction='none')
    return NestedMap(per_example_xent=per_example, log_probs=log_probs)

  def _SampledXent(self, theta, inputs2d, class_ids):
    """Uniform-negative sampled softmax (training only)."""
    p = self.p
    n = inputs2d.shape[0]
    labels = class_ids.reshape(-1).long()
    u = py_utils.GraphSafeUniform((p.num_sampled,), inputs2d.device)
    neg = (u * p.num_classes).long().clamp_max(p.num_classes - 1)
    cols = torch.cat([labels, neg])  # [N + S]
    w_cols = theta.linear_w[:, cols]  # [D, N+S]
    b_cols = theta.bias[cols]
    logits = torch.matmul(inputs2d, w_cols) + b_cols  # [N, N+S]
    # The true class of row i is column i.
    tgt = torch.arange(n, device=inputs2d.device)
    import torch.nn.functional as F2
    return F2.cross_entropy(logits.float(), tgt, reduction='none')

  def XentLoss(self, theta: NestedMap, inputs: torch.Tensor,
               class_weights: torch.Tensor,
               class_ids: Optional[torch.Tensor] = None,
               class_probabilities: Optional[torch.Tensor] = None
               ) -> NestedMap:
    p = self.p
    inputs2d = inputs.reshape(-1, p.input_dim)
    w = class_weights.reshape(-1).float()
    if p.num_sampled and self.training and class_ids is not None:
      per_example = self._SampledXent(theta, inputs2d, class_ids)
      total_weight = w.sum()
      total_xent = (per_example * w).sum()
      return NestedMap(
          total_xent=total_xent, total_weight=total_weight,
          avg_xent=total_xent / total_weight.clamp_min(1e-8),
          per_example_xent=per_example)
    if (inputs2d.is_cuda and class_probabilities is None and
        p.chunk_size == 0):
      from lingvo_amd.ops import softmax_xent
      per_example = softmax_xent.logits_xent(
          inputs2d, theta.linear_w, theta.bias, class_ids.reshape(-1))
    elif p.chunk_size > 0 and class_probabilities is None:
      # Vocab-chunked xent (reference SimpleFullSoftmax
      # softmax_max_alloc chunking, layers.py:3697): never materializes
      # [R, V]; fwd streams a running logsumexp over vocab chunks, bwd
      # recomputes each chunk's logits.
      per_example = _ChunkedLogitsXent.apply(
          inputs2d, theta.linear_w, theta.bias,
          class_ids.reshape(-1).long(), p.chunk_size)
    else:
      logits = self.Logits(theta, inputs2d)
      per_example = self.XentLossFromLogits(
          logits, class_ids,
          None if class_probabilities is None else
          class_probabilities.reshape(-1, p.num_classes)).per_example_xent
    total_weight = w.sum()
    total_xent = (per_example * w).sum()
    return NestedMap(
        total_xent=total_xent,
        total_weight=total_weight,
        avg_xent=total_xent / total_weight.clamp_min(1e-8),
        per_example_xent=per_example)


class SharedSoftmaxLayer(SimpleFullSoftmax):
  """Softmax sharing its weight as the embedding table
  (reference layers.py:4403)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('scale_sqrt_depth', True, 'Scale embedding by sqrt(dim).')
    return p

  def EmbLookup(self, theta: NestedMap, ids: torch.Tensor) -> torch.Tensor:
    emb = F.embedding(ids, theta.linear_w.t().contiguous())
    if self.p.scale_sqrt_depth:
      emb = emb * (self.p.input_dim ** 0.5)
    return emb


class UniformLabelSmoother(BaseLayer):
  """Uniform label smoothing (reference layers.py:5305)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('num_classes', 0, 'Number of classes.')
    p.Define('uncertainty', 0.1, 'Smoothing mass.')
    return p

  def FProp(self, theta: NestedMap, target_ids: torch.Tensor
            ) -> torch.Tensor:
    p = self.p
    off = p.uncertainty / (p.num_classes - 1)
    probs = torch.full(
        (*target_ids.shape, p.num_classes), off,
        dtype=torch.float32, device=target_ids.device)
    probs.scatter_(-1, target_ids.long().unsqueeze(-1), 1.0 - p.uncertainty)
    return probs


class HighwaySkipLayer(BaseLayer):
  """Highway connection y = g*x_trans + (1-g)*x (reference layers.py:5461)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('input_dim', 0, 'Dim.')
    p.Define('batch_norm', False, 'Unused (parity).')
    p.Define('carry_bias_init', -1.0, 'Carry gate bias init.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    self.CreateVariable('w_t', py_utils.WeightParams(
        [p.input_dim, p.input_dim], p.params_init, p.dtype))
    self.CreateVariable('b_t', py_utils.WeightParams(
        [p.input_dim], py_utils.WeightInit.Constant(p.carry_bias_init),
        p.dtype))
    self.CreateVariable('w_h', py_utils.WeightParams(
        [p.input_dim, p.input_dim], p.params_init, p.dtype))
    self.CreateVariable('b_h', py_utils.WeightParams(
        [p.input_dim], py_utils.WeightInit.Constant(0.0), p.dtype))

  def FProp(self, theta: NestedMap, x: torch.Tensor,
            transformed: Optional[torch.Tensor] = None) -> torch.Tensor:
    if transformed is None:
      transformed = torch.relu(py_utils.MatmulBias(x, theta.w_h, theta.b_h))
    gate = torch.sigmoid(py_utils.MatmulBias(x, theta.w_t, theta.b_t))
    return gate * transformed + (1.0 - gate) * x


class GluLayer(BaseLayer):
  """Gated linear unit block with residual (reference layers.py:6124)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('input_dim', 0, 'Dim.')
    p.Define('output_dim', 0, 'Output (0 = input_dim).')
    p.Define('dropout_prob', 0.0, 'Dropout.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    out = p.output_dim or p.input_dim
    self.CreateChild('ln', LayerNorm.Params().Set(input_dim=p.input_dim))
    self.CreateVariable('w', py_utils.WeightParams(
        [p.input_dim, 2 * out], p.params_init, p.dtype))
    self.CreateVariable('b', py_utils.WeightParams(
        [2 * out], py_utils.WeightInit.Constant(0.0), p.dtype))

  def FProp(self, theta: NestedMap, x: torch.Tensor,
            paddings: Optional[torch.Tensor] = None) -> torch.Tensor:
    p = self.p
    h = self.ln.FProp(theta.ln, x)
    h = py_utils.MatmulBias(h, theta.w, theta.b)
    a, g = h.chunk(2, dim=-1)
    out = a * torch.sigmoid(g)
    if p.dropout_prob and not self.do_eval:
      out = py_utils.DeterministicDropout(out, 1.0 - p.dropout_prob)
    if (p.output_dim or p.input_dim) == p.input_dim:
      out = out + x
    if paddings is not None:
      out = py_utils.ApplyPadding(paddings, out)
    return out


class GradNormTracker(BaseLayer):
  """Tracks a running log-grad-norm and flags outlier steps
  (reference layers.py:5590; the Learner consumes the verdict to skip)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('decay', 0.995, 'EMA decay of log grad norm stats.')
    p.Define('clip_threshold', 4.0, 'Allowed stds above the mean.')
    p.Define('grad_norm_clip_cap_min', 0.0, 'Floor for the cap.')
    return p

  def __init__(self, params):
    super().__init__(params)
    self.register_buffer('log_mean', torch.zeros(()))
    self.register_buffer('log_var', torch.ones(()))
    self.register_buffer('count', torch.zeros(()))

  def FProp(self, theta: NestedMap, grad_norm: torch.Tensor) -> bool:
    """Returns True if this step's grad norm is acceptable; updates
    running stats only on accepted steps."""
    p = self.p
    log_n = torch.log(grad_norm.detach().float().clamp_min(1e-10))
    if float(self.count) < 10:
      ok = True
    else:
      std = self.log_var.clamp_min(1e-6).sqrt()
      cap = self.log_mean + p.clip_threshold * std
      ok = bool(log_n <= torch.maximum(
          cap, torch.tensor(math.log(max(p.grad_norm_clip_cap_min,
                                         1e-10)))))
    if ok:
      d = p.decay
      delta = log_n - self.log_mean
      self.log_mean.mul_(d).add_((1 - d) * log_n)
      self.log_var.mul_(d).add_((1 - d) * delta * delta)
      self.count += 1
    return ok


class MultitaskAdapterLayer(BaseLayer):
  """Per-task residual bottleneck adapters (reference layers.py:6205
  MultitaskAdapterLayer; Houlsby et al. 2019). One [num_tasks, ...]
  weight stack; FProp gathers each example's task adapter:
      y = x + Wup_t * act(Wdown_t * LN(x) + b_t) + bup_t.
  """

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('num_tasks', 1, 'Task count.')
    p.Define('input_dim', 0, 'Model dim D.')
    p.Define('bottleneck_dim', 0, 'Adapter bottleneck.')
    p.Define('activation', 'RELU', 'Bottleneck activation.')
    p.Define('layer_norm', True, 'LN before the adapter.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    self.CreateVariable('down_w', py_utils.WeightParams(
        [p.num_tasks, p.input_dim, p.bottleneck_dim], p.params_init,
        p.dtype))
    self.CreateVariable('down_b', py_utils.WeightParams(
        [p.num_tasks, p.bottleneck_dim],
        py_utils.WeightInit.Constant(0.0), p.dtype))
    self.CreateVariable('up_w', py_utils.WeightParams(
        [p.num_tasks, p.bottleneck_dim, p.input_dim], p.params_init,
        p.dtype))
    self.CreateVariable('up_b', py_utils.WeightParams(
        [p.num_tasks, p.input_dim],
        py_utils.WeightInit.Constant(0.0), p.dtype))
    if p.layer_norm:
      self.CreateChild('ln', LayerNorm.Params().Set(
          input_dim=p.input_dim))

  def FProp(self, theta: NestedMap, inputs: torch.Tensor,
            tasks: torch.Tensor) -> torch.Tensor:
    """inputs [B, T, D]; tasks [B] int task ids."""
    p = self.p
    x = self.ln.FProp(theta.ln, inputs) if p.layer_norm else inputs
    t = tasks.long()
    dw = theta.down_w[t]       # [B, D, bottleneck]
    db = theta.down_b[t]       # [B, bottleneck]
    uw = theta.up_w[t]
    ub = theta.up_b[t]
    act = activations.GetFn(p.activation)
    h = act(torch.baddbmm(db.unsqueeze(1), x, dw))
    return inputs + torch.baddbmm(ub.unsqueeze(1), h, uw)


class ShardedEmbeddingLayer(BaseLayer):
  """Vocab-sharded embedding table over the data/model-parallel group
  (the MI355X counterpart of the reference's TPU-embedding subsystem,
  core/tpu_embedding_layers*.py, 2329 LoC: giant tables sharded across
  accelerators with gradients routed back to the owning shard).

  Rank r stores rows [r*V/W, (r+1)*V/W). Lookup: each rank gathers the
  ids it owns (others contribute zeros) and one all-reduce sums the
  partial embeddings — 288 GB HBM per GPU means even 100B-row tables
  shard across a single node without a separate parameter-server
  engine. Gradients flow only to the owning rank's shard (marked
  _ep_sharded so DP GradSync skips them)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('vocab_size', 0, 'GLOBAL vocabulary size.')
    p.Define('embedding_dim', 0, 'Embedding dimension.')
    p.Define('scale_sqrt_depth', False, 'Scale outputs by sqrt(dim).')
    p.Define('shard_group', None, 'torch.distributed group (None = '
             'default).')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    import torch.distributed as dist
    if dist.is_available() and dist.is_initialized():
      self._world = dist.get_world_size(p.shard_group)
      self._rank = dist.get_rank(p.shard_group)
    else:
      self._world, self._rank = 1, 0
    assert p.vocab_size % self._world == 0, (
        'vocab must divide the shard world')
    self._vshard = p.vocab_size // self._world
    self.CreateVariable('wm', py_utils.WeightParams(
        [self._vshard, p.embedding_dim],
        py_utils.WeightInit.Gaussian(1.0 / math.sqrt(p.embedding_dim)),
        p.dtype))
    if self._world > 1:
      # Deterministic slice of the full-table init (parity with the
      # unsharded layer under the same seed).
      g = self._InitGenerator('wm')
      full = py_utils.InitWeight(
          [p.vocab_size, p.embedding_dim],
          py_utils.WeightInit.Gaussian(1.0 / math.sqrt(p.embedding_dim)),
          g, p.dtype)
      with torch.no_grad():
        self.wm.copy_(full[self._rank * self._vshard:
                           (self._rank + 1) * self._vshard])
      self.wm._ep_sharded = True

  def EmbLookup(self, theta: NestedMap, ids: torch.Tensor) -> torch.Tensor:
    p = self.p
    scale = p.embedding_dim ** 0.5 if p.scale_sqrt_depth else 1.0
    ids = ids.long()
    if self._world == 1:
      from lingvo_amd.ops import embedding as emb_ops
      return emb_ops.embedding_lookup(theta.wm, ids, scale)
    lo = self._rank * self._vshard
    mine = (ids >= lo) & (ids < lo + self._vshard)
    local_ids = torch.where(mine, ids - lo, torch.zeros_like(ids))
    out = F.embedding(local_ids, theta.wm) * scale
    out = out * mine.unsqueeze(-1).to(out.dtype)
    from lingvo_amd.parallel.tensor_parallel import _ReduceFromTp
    return _ReduceFromTp.apply(out, p.shard_group)

  def FProp(self, theta: NestedMap, ids: torch.Tensor) -> torch.Tensor:
    return self.EmbLookup(theta, ids)
