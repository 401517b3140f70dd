"""Composable data sources (reference lingvo/core/datasource.py:38-838:
SimpleDataSource:85, cross-batch mixing:194, curriculum:253)."""

from __future__ import annotations

import random
from typing import List, Optional

from lingvo_amd.core.base_layer import BaseLayer
from lingvo_amd.core.nested_map import NestedMap


class DataSource(BaseLayer):

  def GetNext(self) -> NestedMap:
    raise NotImplementedError

  def Reset(self) -> None:
    pass


class SimpleDataSource(DataSource):
  """Wraps one input generator (reference datasource.py:85)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('input_generator', None, 'Input generator params.')
    return p

  def __init__(self, params):
    super().__init__(params)
    self.CreateChild('generator', self.p.input_generator)

  def GetNext(self) -> NestedMap:
    return self.generator.GetPreprocessedInputBatch()

  def Reset(self) -> None:
    self.generator.Reset()


class CrossBatchMixingDataSource(DataSource):
  """Samples whole batches from sub-sources with weights
  (reference datasource.py:194)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('sub', [], 'Sub-datasource params.')
    p.Define('weights', [], 'Sampling weights.')
    return p

  def __init__(self, params):
    super().__init__(params)
    assert len(self.p.sub) == len(self.p.weights)
    self.CreateChildren('sources', [sp.Copy() for sp in self.p.sub])
    self._rng = random.Random(self.p.random_seed or 301)

  def GetNext(self) -> NestedMap:
    i = self._rng.choices(range(len(self.sources)),
                          weights=self.p.weights, k=1)[0]
    batch = self.sources[i].GetNext()
    batch.source_id = i
    return batch


class CurriculumDataSource(DataSource):
  """Switches sources at step boundaries (reference datasource.py:253)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('sub', [], 'Sub-datasource params, one per stage.')
    p.Define('boundaries', [], 'Global-step boundaries between stages.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    if len(p.sub) != len(p.boundaries) + 1:
      raise ValueError(
          'Expected one more sub than boundaries; got %d sub, %d '
          'boundaries' % (len(p.sub), len(p.boundaries)))
    if list(p.boundaries) != sorted(p.boundaries):
      raise ValueError('boundaries must be monotonically increasing')
    self.CreateChildren('sources', [sp.Copy() for sp in p.sub])
    self._step = 0

  def SetStep(self, step: int) -> None:
    self._step = int(step)

  # Runner-facing alias (Trainer calls SetGlobalStep each step).
  SetGlobalStep = SetStep

  @property
  def current_stage(self) -> int:
    idx = 0
    for b in self.p.boundaries:
      if self._step >= b:
        idx += 1
    return idx

  def GetNext(self) -> NestedMap:
    return self.sources[self.current_stage].GetNext()

  def Reset(self) -> None:
    for src in self.sources:
      src.Reset()


class SequentialDataSource(DataSource):
  """Runs sub-sources to exhaustion in order (reference
  sequential_record_yielder + datasource chaining): source i+1 starts
  when source i raises StopIteration; raises StopIteration after the
  last (one pass)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('sub', [], 'Sub-datasource params, consumed in order.')
    return p

  def __init__(self, params):
    super().__init__(params)
    self.CreateChildren('sources', [sp.Copy() for sp in self.p.sub])
    self._idx = 0

  def GetNext(self) -> NestedMap:
    while self._idx < len(self.sources):
      try:
        batch = self.sources[self._idx].GetNext()
        batch.source_id = self._idx
        return batch
      except StopIteration:
        self._idx += 1
    raise StopIteration

  def Reset(self) -> None:
    self._idx = 0
    for s in self.sources:
      s.Reset()


class WithinBatchMixingDataSource(DataSource):
  """Example-level mixing: each output batch contains rows drawn from
  the sub-sources in (expected) weight proportion (reference
  base_input_generator.py:1216 within-batch mixing semantics, lifted to
  the datasource layer). Sub-batches must be structurally compatible."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('sub', [], 'Sub-datasource params.')
    p.Define('weights', [], 'Mixing weights.')
    p.Define('batch_size', 0, 'Rows per mixed batch (0 = first sub '
             'batch size).')
    return p

  def __init__(self, params):
    super().__init__(params)
    assert len(self.p.sub) == len(self.p.weights)
    self.CreateChildren('sources', [sp.Copy() for sp in self.p.sub])
    self._rng = random.Random(self.p.random_seed or 301)
    self._pools: List[Optional[NestedMap]] = [None] * len(self.p.sub)
    self._cursor = [0] * len(self.p.sub)

  def _Row(self, i: int) -> NestedMap:
    import torch
    if self._pools[i] is None or self._cursor[i] >= next(
        v.shape[0] for v in self._pools[i].Flatten()
        if isinstance(v, torch.Tensor)):
      self._pools[i] = self.sources[i].GetNext()
      self._cursor[i] = 0
    r = self._cursor[i]
    self._cursor[i] += 1
    return self._pools[i].Transform(
        lambda t: t[r:r + 1] if isinstance(t, torch.Tensor) else t)

  def GetNext(self) -> NestedMap:
    import torch
    n = self.p.batch_size
    if not n:
      if self._pools[0] is None:
        self._pools[0] = self.sources[0].GetNext()
        self._cursor[0] = 0
      n = next(v.shape[0] for v in self._pools[0].Flatten()
               if isinstance(v, torch.Tensor))
    rows = []
    srcs = []
    for _ in range(n):
      i = self._rng.choices(range(len(self.sources)),
                            weights=self.p.weights, k=1)[0]
      rows.append(self._Row(i))
      srcs.append(i)
    flat = [r.Flatten() for r in rows]
    out_vals = []
    for j in range(len(flat[0])):
      vals = [f[j] for f in flat]
      if isinstance(vals[0], torch.Tensor):
        out_vals.append(torch.cat(vals, dim=0))
      else:
        out_vals.append(vals[0])
    out = rows[0].Pack(out_vals)
    out.source_id = torch.tensor(srcs)
    return out

