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
    assert len(self.p.sub) == len(self.p.boundaries) + 1
    self.CreateChildren('sources', [sp.Copy() for sp in self.p.sub])
    self._step = 0

  def SetStep(self, step: int) -> None:
    self._step = step

  def GetNext(self) -> NestedMap:
    idx = 0
    for b in self.p.boundaries:
      if self._step >= b:
        idx += 1
    return self.sources[idx].GetNext()
