"""Multi-loss gradient combination (reference
lingvo/core/gradient_combiner.py:44 GradientCombiner; gradient surgery
arXiv:2001.06782, GradNorm arXiv:1711.02257).

Contract: `Combine(vmap, losses_and_gradients)` where
losses_and_gradients is {loss_name: NestedMap(loss_metric=(loss, w),
grads=NestedMap matching vmap)}. Returns a NestedMap of combined
gradients with vmap's structure.
"""

from __future__ import annotations

import torch

from lingvo_amd.core.base_layer import BaseLayer
from lingvo_amd.core.nested_map import NestedMap


class GradientCombiner(BaseLayer):
  """Abstract combiner (reference gradient_combiner.py:44)."""

  def Combine(self, vmap: NestedMap, losses_and_gradients) -> NestedMap:
    raise NotImplementedError(type(self))


class SumGradientCombiner(GradientCombiner):
  """Weighted sum of per-loss gradients (the default TF behavior)."""

  def Combine(self, vmap: NestedMap, losses_and_gradients) -> NestedMap:
    flat_out = None
    for entry in losses_and_gradients.values():
      _, weight = entry.loss_metric
      flat = entry.grads.Flatten()
      scaled = [None if g is None else g * weight for g in flat]
      if flat_out is None:
        flat_out = scaled
      else:
        flat_out = [a if b is None else (b if a is None else a + b)
                    for a, b in zip(flat_out, scaled)]
    return vmap.Pack(flat_out)


class PCGradCombiner(GradientCombiner):
  """Projecting Conflicting Gradients (arXiv:2001.06782): for each
  pair of task gradients with negative cosine similarity, project one
  onto the normal plane of the other before summing."""

  def Combine(self, vmap: NestedMap, losses_and_gradients) -> NestedMap:
    tasks = list(losses_and_gradients.values())
    orig = []
    for entry in tasks:
      _, weight = entry.loss_metric
      orig.append([None if g is None else g * weight
                   for g in entry.grads.Flatten()])
    # Projections use the ORIGINAL other-task gradients (PCGrad alg. 1):
    # after the pass, each projected g_i has non-negative dot with every
    # original g_j it conflicted with.
    flats = [[None if g is None else g.clone() for g in f] for f in orig]
    n = len(flats)
    for i in range(n):
      for j in range(n):
        if i == j:
          continue
        dot = sum((a * b).sum() for a, b in zip(flats[i], orig[j])
                  if a is not None and b is not None)
        if float(dot) >= 0.0:
          continue
        sq = sum((b * b).sum() for b in orig[j] if b is not None)
        coef = dot / sq.clamp_min(1e-12)
        for k, b in enumerate(orig[j]):
          if flats[i][k] is not None and b is not None:
            flats[i][k] = flats[i][k] - coef * b
    out = flats[0]
    for f in flats[1:]:
      out = [a if b is None else (b if a is None else a + b)
             for a, b in zip(out, f)]
    return vmap.Pack(out)
