"""MNIST image classification task (reference lingvo/tasks/image/classifier.py).

LeNet5 (reference tasks/image/params/mnist.py) is the CPU plumbing config
named by BASELINE.json.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch
import torch.nn.functional as F

from lingvo_amd.core import py_utils
from lingvo_amd.core.base_input_generator import BaseInputGenerator
from lingvo_amd.core.base_model import BaseTask
from lingvo_amd.core.nested_map import NestedMap
from lingvo_amd.layers import layers as lingvo_layers


class FakeMnistData(BaseInputGenerator):
  """Deterministic synthetic MNIST batches (reference
  tasks/image/input_generator.py:84 FakeMnistData)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.batch_size = 64
    p.Define('image_size', 28, 'Image side length.')
    p.Define('num_classes', 10, 'Classes.')
    return p

  def _InputBatch(self) -> NestedMap:
    p = self.p
    g = torch.Generator().manual_seed(1234 + self._batch_count)
    data = torch.rand(p.batch_size, p.image_size, p.image_size, 1,
                      generator=g)
    label = torch.randint(0, p.num_classes, (p.batch_size,), generator=g)
    return NestedMap(data=data, label=label,
                     weight=torch.ones(p.batch_size))


class BaseClassifier(BaseTask):
  """Image classifier task (reference tasks/image/classifier.py:48)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('softmax', lingvo_layers.SimpleFullSoftmax.Params(),
             'Softmax params.')
    return p


class ModelV1(BaseClassifier):
  """Conv-pool stack + FC + softmax (reference classifier.py:100)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('filter_shapes', [(5, 5, 1, 20), (5, 5, 20, 50)],
             'Conv filter shapes.')
    p.Define('window_shape', (2, 2), 'Pool window.')
    p.Define('hidden_dim', 300, 'FC hidden dim.')
    p.Define('image_size', 28, 'Input image side.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    conv_ps = []
    side = p.image_size
    for i, fs in enumerate(p.filter_shapes):
      conv_ps.append(lingvo_layers.Conv2DLayer.Params().Set(
          name=f'conv{i}', filter_shape=fs, filter_stride=(1, 1),
          padding='SAME', activation='RELU'))
      side = side // p.window_shape[0]
    self.CreateChildren('convs', conv_ps)
    self.CreateChild('pool', lingvo_layers.PoolingLayer.Params().Set(
        window_shape=p.window_shape, window_stride=p.window_shape))
    flat_dim = side * side * p.filter_shapes[-1][-1]
    self.CreateChild('fc', lingvo_layers.FCLayer.Params().Set(
        input_dim=flat_dim, output_dim=p.hidden_dim))
    softmax_p = p.softmax.Copy()
    softmax_p.input_dim = p.hidden_dim
    self.CreateChild('softmax', softmax_p)

  def ComputePredictions(self, theta: NestedMap,
                         input_batch: NestedMap) -> NestedMap:
    x = input_batch.data.to(self.fprop_dtype)
    for i, conv in enumerate(self.convs):
      x = conv.FProp(theta.convs[i], x)
      x = self.pool.FProp(theta.pool, x)
    x = x.reshape(x.shape[0], -1)
    x = self.fc.FProp(theta.fc, x)
    logits = self.softmax.Logits(theta.softmax, x)
    return NestedMap(logits=logits, activations=x)

  def ComputeLoss(self, theta: NestedMap, predictions: NestedMap,
                  input_batch: NestedMap):
    logits = predictions.logits
    labels = input_batch.label.long()
    weights = input_batch.weight.float()
    per_example = F.cross_entropy(logits.float(), labels, reduction='none')
    total_w = weights.sum()
    loss = (per_example * weights).sum() / total_w.clamp_min(1e-8)
    acc1 = ((logits.argmax(-1) == labels).float() * weights).sum() / \
        total_w.clamp_min(1e-8)
    top5 = logits.topk(min(5, logits.shape[-1]), dim=-1).indices
    acc5 = ((top5 == labels[:, None]).any(-1).float() * weights).sum() / \
        total_w.clamp_min(1e-8)
    metrics = NestedMap(
        loss=(loss, total_w),
        accuracy=(acc1.detach(), total_w),
        acc5=(acc5.detach(), total_w),
        num_samples_in_batch=(total_w.detach(), torch.ones(())))
    return metrics, NestedMap(loss=per_example.detach())

  def Decode(self, input_batch: NestedMap) -> NestedMap:
    with torch.no_grad():
      preds = self.ComputePredictions(self.theta, input_batch)
    return NestedMap(correct_top1=(
        preds.logits.argmax(-1) == input_batch.label.long()).float())

  def Inference(self) -> NestedMap:
    """Named inference subgraphs (reference base_model.py:943)."""

    def default(images):
      preds = self.ComputePredictions(self.theta,
                                      NestedMap(data=images))
      return NestedMap(logits=preds.logits,
                       probs=torch.softmax(preds.logits.float(), -1),
                       label=preds.logits.argmax(-1))

    return NestedMap(default=default)
