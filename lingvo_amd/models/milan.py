"""Milan: dual-encoder image/text retrieval task
(reference lingvo/tasks/milan: dual_encoder.py, score_functions.py;
EfficientNetB4BertAdapter in params/cxc.py).

MI355X-native composition: an image tower (conv stack) and a text tower
(transformer) projected into a shared space, trained with a symmetric
in-batch softmax contrastive loss (the reference's dual-encoder loss).
"""

from __future__ import annotations

import torch
import torch.nn.functional as F

from lingvo_amd.core import py_utils
from lingvo_amd.core.base_input_generator import BaseInputGenerator
from lingvo_amd.core.base_layer import BaseLayer
from lingvo_amd.core.base_model import BaseTask
from lingvo_amd.core.nested_map import NestedMap
from lingvo_amd.layers import layers as lingvo_layers
from lingvo_amd.layers import transformer as transformer_lib


class SyntheticImageTextInput(BaseInputGenerator):

  @classmethod
  def Params(cls):
    p = super().Params()
    p.batch_size = 32
    p.Define('image_size', 64, 'Image side.')
    p.Define('text_len', 16, 'Caption length.')
    p.Define('vocab_size', 1000, 'Caption vocab.')
    return p

  def _InputBatch(self) -> NestedMap:
    p = self.p
    g = torch.Generator().manual_seed(2200 + self._batch_count)
    return NestedMap(
        image=torch.randn(p.batch_size, p.image_size, p.image_size, 3,
                          generator=g),
        text=torch.randint(1, p.vocab_size, (p.batch_size, p.text_len),
                           generator=g),
        text_paddings=torch.zeros(p.batch_size, p.text_len))


class DualEncoder(BaseTask):
  """Image tower + text tower + symmetric contrastive loss."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('joint_dim', 256, 'Shared embedding dim.')
    p.Define('image_channels', [32, 64, 128], 'Conv tower channels.')
    p.Define('text_dim', 256, 'Text tower dim.')
    p.Define('text_layers', 2, 'Text transformer layers.')
    p.Define('vocab_size', 1000, 'Caption vocab.')
    p.Define('temperature', 0.07, 'Softmax temperature.')
    p.Define('score_function', 'dot',
             "'dot' | 'bilinear' (learned W between towers; reference "
             "milan/score_functions.py).")
    p.Define('label_smoothing', 0.0,
             'Label smoothing on the in-batch softmax loss.')
    p.Define('id_feature', '',
             'Optional batch field of example ids: pairs sharing an id '
             'with the diagonal are masked out of the loss (reference '
             'dual_encoder.py:52 id-based duplicate masking).')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    convs = []
    cin = 3
    for i, ch in enumerate(p.image_channels):
      convs.append(lingvo_layers.Conv2DLayer.Params().Set(
          name=f'conv{i}', filter_shape=(3, 3, cin, ch),
          filter_stride=(2, 2), activation='RELU'))
      cin = ch
    self.CreateChildren('image_convs', convs)
    self.CreateChild('image_proj', lingvo_layers.ProjectionLayer.Params()
                     .Set(input_dim=cin, output_dim=p.joint_dim,
                          has_bias=True))
    self.CreateChild('text_emb', lingvo_layers.EmbeddingLayer.Params().Set(
        vocab_size=p.vocab_size, embedding_dim=p.text_dim))
    self.CreateChild('text_stack',
                     transformer_lib.StackedTransformerLayers.Params().Set(
                         model_dim=p.text_dim, num_layers=p.text_layers,
                         num_heads=max(1, p.text_dim // 64)))
    self.CreateChild('text_proj', lingvo_layers.ProjectionLayer.Params()
                     .Set(input_dim=p.text_dim, output_dim=p.joint_dim,
                          has_bias=True))
    if p.score_function == 'bilinear':
      self.CreateVariable('score_w', py_utils.WeightParams(
          [p.joint_dim, p.joint_dim], p.params_init, p.dtype))

  def EncodeImage(self, theta, images):
    x = images.to(self.fprop_dtype)
    for i, conv in enumerate(self.image_convs):
      x = conv.FProp(theta.image_convs[i], x)
    x = x.mean(dim=(1, 2))  # global average pool
    x = self.image_proj.FProp(theta.image_proj, x)
    return F.normalize(x.float(), dim=-1)

  def EncodeText(self, theta, text, paddings):
    x = self.text_emb.EmbLookup(theta.text_emb, text.long()).to(
        self.fprop_dtype)
    x = self.text_stack.FProp(theta.text_stack, x, paddings)
    mask = (1.0 - paddings).unsqueeze(-1).to(x.dtype)
    x = (x * mask).sum(1) / mask.sum(1).clamp_min(1.0)
    x = self.text_proj.FProp(theta.text_proj, x)
    return F.normalize(x.float(), dim=-1)

  def ComputePredictions(self, theta, input_batch):
    img = self.EncodeImage(theta, input_batch.image)
    txt = self.EncodeText(theta, input_batch.text,
                          input_batch.text_paddings)
    return NestedMap(image_emb=img, text_emb=txt)

  def Score(self, theta, image_emb, text_emb):
    """Tower-pair similarity (reference score_functions.py: dot product
    or learned bilinear form)."""
    if self.p.score_function == 'bilinear':
      return image_emb @ theta.score_w.float() @ text_emb.t()
    return image_emb @ text_emb.t()

  def ComputeLoss(self, theta, predictions, input_batch):
    p = self.p
    sims = self.Score(theta, predictions.image_emb,
                      predictions.text_emb) / p.temperature
    labels = torch.arange(sims.shape[0], device=sims.device)
    if p.id_feature and p.id_feature in input_batch:
      # Mask off-diagonal pairs that are actually positives (same id):
      # they must not be treated as negatives (reference id masking).
      ids = input_batch[p.id_feature].reshape(-1)
      dup = (ids[:, None] == ids[None, :]) &           ~torch.eye(len(ids), dtype=torch.bool, device=sims.device)
      sims = sims.masked_fill(dup, -1e30)
    loss_i2t = F.cross_entropy(sims, labels,
                               label_smoothing=p.label_smoothing)
    loss_t2i = F.cross_entropy(sims.t(), labels,
                               label_smoothing=p.label_smoothing)
    loss = 0.5 * (loss_i2t + loss_t2i)
    acc = (sims.argmax(-1) == labels).float().mean()
    w = torch.tensor(float(sims.shape[0]))
    metrics = NestedMap(loss=(loss, w),
                        retrieval_at_1=(acc.detach(), w),
                        num_samples_in_batch=(w, torch.ones(())))
    return metrics, NestedMap()

  def Decode(self, input_batch):
    with torch.no_grad():
      preds = self.ComputePredictions(self.theta, input_batch)
      sims = preds.image_emb @ preds.text_emb.t()
    return NestedMap(ranks=sims.argsort(-1, descending=True))

  def CreateDecoderMetrics(self) -> NestedMap:
    from lingvo_amd.core import metrics as metrics_lib
    return NestedMap(recall_at_1=metrics_lib.AverageMetric(),
                     recall_at_5=metrics_lib.AverageMetric(),
                     num_samples_in_batch=metrics_lib.AverageMetric())

  def PostProcessDecodeOut(self, decode_out, decode_metrics) -> None:
    """In-batch image->text retrieval recall@k (reference milan
    score/eval utilities)."""
    ranks = decode_out.ranks
    b = ranks.shape[0]
    labels = torch.arange(b, device=ranks.device).unsqueeze(1)
    pos = (ranks == labels).float().argmax(dim=1)
    decode_metrics.recall_at_1.Update(float((pos < 1).float().mean()), b)
    decode_metrics.recall_at_5.Update(float((pos < 5).float().mean()), b)
    decode_metrics.num_samples_in_batch.Update(float(b))
