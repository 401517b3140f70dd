"""Input generators (reference lingvo/core/base_input_generator.py).

BaseInputGenerator yields NestedMap batches. On a GPU box, ToDevice()
moves batches host->HBM with pinned memory on a dedicated HIP copy stream
(the MI355X replacement for TPU infeed, reference
base_input_generator.py:446-671); on CPU it is a no-op.
"""

from __future__ import annotations

from typing import Iterator, List, Optional

import torch

from lingvo_amd.core.base_layer import BaseLayer
from lingvo_amd.core.nested_map import NestedMap


class BaseInputGenerator(BaseLayer):

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('batch_size', 16, 'Per-device batch size.')
    p.Define('num_samples', 0, 'Dataset size (0 = infinite/synthetic).')
    p.Define('resettable', False, 'Whether the source can be reset for eval.')
    return p

  def __init__(self, params):
    super().__init__(params)
    self._copy_stream = None
    self._batch_count = 0

  def InfeedBatchSize(self) -> int:
    return self.p.batch_size

  def _InputBatch(self) -> NestedMap:
    """Subclass: produce one CPU batch."""
    raise NotImplementedError

  def GetPreprocessedInputBatch(self) -> NestedMap:
    batch = self._InputBatch()
    self._batch_count += 1
    return batch

  def Reset(self) -> None:
    self._batch_count = 0

  def __iter__(self) -> Iterator[NestedMap]:
    while True:
      yield self.GetPreprocessedInputBatch()

  # ---- host->device (infeed replacement) --------------------------------
  def SplitInputBatch(self, batch: NestedMap,
                      num_splits: int) -> list:
    """Tower batch splitting (reference base_input_generator.py:1006
    SplitInputBatch / input_generator_helper.py ComputeSplits)."""
    from lingvo_amd.utils import helpers
    return helpers.SplitNestedMap(batch, num_splits)

  def ToDevice(self, batch: NestedMap, device) -> NestedMap:
    """Moves a batch to the device, overlapping the copy on a side stream."""
    if str(device).startswith('cuda') and torch.cuda.is_available():
      if self._copy_stream is None:
        self._copy_stream = torch.cuda.Stream(device=device)
      with torch.cuda.stream(self._copy_stream):
        moved = batch.Transform(
            lambda t: t.pin_memory().to(device, non_blocking=True)
            if isinstance(t, torch.Tensor) else t)
      torch.cuda.current_stream(device).wait_stream(self._copy_stream)
      return moved
    return batch.Transform(
        lambda t: t.to(device) if isinstance(t, torch.Tensor) else t)


class BaseSequenceInputGenerator(BaseInputGenerator):
  """Adds length-bucketing params (reference base_input_generator.py:1457)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('bucket_upper_bound', [], 'Bucket length upper bounds.')
    p.Define('bucket_batch_limit', [], 'Per-bucket batch sizes.')
    p.Define('source_max_length', None, 'Max source length.')
    p.Define('target_max_length', 300, 'Max target length.')
    p.Define('pad_to_max_seq_length', False, 'Static-shape padding.')
    return p

  def scaled_bucket_batch_limit(self) -> List[int]:
    return list(self.p.bucket_batch_limit)


class BaseInputGeneratorFromFiles(BaseSequenceInputGenerator):
  """File-pattern inputs with within-batch mixing (reference
  base_input_generator.py:1216 BaseInputGeneratorFromFiles).

  `file_pattern` is either one 'format:glob' string or a weighted list
  [(pattern, weight), ...]: records are then mixed EXAMPLE-level (every
  batch contains a weighted mix, unlike CrossBatchMixingDataSource which
  alternates whole batches). Subclasses implement ProcessRecord()
  (record bytes -> (NestedMap example, bucket key))."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('file_pattern', '', 'Pattern or [(pattern, weight), ...].')
    p.Define('file_random_seed', 301, 'Shuffle seed.')
    p.Define('file_buffer_size', 10000, 'Shuffle buffer records.')
    p.Define('file_parallelism', 4, 'Reader threads per pattern.')
    p.Define('num_batcher_threads', 2, 'Processor threads.')
    p.Define('repeat', True, 'Loop forever (False = one epoch).')
    return p

  def __init__(self, params):
    super().__init__(params)
    self._pipeline = None

  def ProcessRecord(self, record: bytes):
    """record -> (NestedMap example, int bucket_key) or None to drop."""
    raise NotImplementedError

  def _BuildPipeline(self):
    from lingvo_amd.core import generic_input
    from lingvo_amd.ops import _loader
    p = self.p
    ext = _loader.get_ext(required=True)
    patterns = p.file_pattern
    if isinstance(patterns, str):
      patterns = [(patterns, 1.0)]
    yielders = []
    weights = []
    for pattern, w in patterns:
      fmt, files = generic_input.ExpandFilePattern(pattern)
      yielders.append(ext.RecordYielder(
          files, fmt, p.file_random_seed, p.file_buffer_size,
          p.file_parallelism, p.repeat))
      weights.append(w)
    if len(yielders) == 1:
      yielder = yielders[0]
    else:
      # Within-batch mixing: record-level weighted sampling feeds ONE
      # batcher, so every batch mixes sources by weight.
      yielder = generic_input.WeightedMixYielder(
          yielders, weights, seed=p.file_random_seed)
    bounds = p.bucket_upper_bound or [2 ** 30]
    limits = p.bucket_batch_limit or [p.batch_size]
    return generic_input.RecordBatcher(
        yielder, lambda rec: self.ProcessRecord(rec), bounds, limits,
        num_threads=p.num_batcher_threads)

  def _InputBatch(self) -> NestedMap:
    if self._pipeline is None:
      self._pipeline = self._BuildPipeline()
    batch = self._pipeline.GetNext()
    if batch is None:
      raise StopIteration
    return batch

  def Reset(self) -> None:
    super().Reset()
    if self._pipeline is not None:
      self._pipeline.Stop()
      self._pipeline = None
