"""GenericInput: file patterns -> processor -> length-bucketed batches.

MI355X-native equivalent of the reference's main input entry
(lingvo/core/generic_input.py:104 + core/ops/generic_input_op_kernels.cc
+ record_batcher.{h,cc}): the C++ RecordYielder (lingvo_amd extension)
reads and shuffles records on native threads; the batcher runs the
user `processor` (record_bytes -> (NestedMap of tensors, bucket_key);
negative bucket keys drop the record, reference x_ops.cc:850-924) on a
thread pool and assembles per-bucket padded batches on a bounded queue.
"""

from __future__ import annotations

import glob
import queue
import threading
from typing import Callable, List, Optional, Sequence, Tuple

import torch

from lingvo_amd.core.nested_map import NestedMap


def ExpandFilePattern(file_pattern: str) -> Tuple[str, List[str]]:
  """'format:glob[,glob...]' -> (format, files). Default format text."""
  fmt = 'text'
  pattern = file_pattern
  if ':' in file_pattern.split(',')[0]:
    fmt, pattern = file_pattern.split(':', 1)
  files: List[str] = []
  for part in pattern.split(','):
    files.extend(sorted(glob.glob(part)))
  return fmt, files


class RecordBatcher:
  """Processor threadpool + bucketing (reference record_batcher.h:89)."""

  def __init__(self, yielder, processor: Callable,
               bucket_upper_bound: Sequence[int],
               bucket_batch_limit: Sequence[int],
               num_threads: int = 2, queue_depth: int = 8):
    assert len(bucket_upper_bound) == len(bucket_batch_limit)
    self._yielder = yielder
    self._processor = processor
    self._bounds = list(bucket_upper_bound)
    self._limits = list(bucket_batch_limit)
    self._buckets: List[List[NestedMap]] = [[] for _ in self._bounds]
    self._lock = threading.Lock()
    self._out: 'queue.Queue' = queue.Queue(maxsize=queue_depth)
    self._stop = False
    self._threads = [threading.Thread(target=self._Loop, daemon=True)
                     for _ in range(num_threads)]
    for t in self._threads:
      t.start()

  def _Loop(self):
    while not self._stop:
      try:
        record, source_id = self._yielder.yield_record()
      except StopIteration:
        # Flush partial buckets (eval tail; reference record_batcher
        # flush semantics) before signaling end-of-stream.
        with self._lock:
          tails = [b for b in self._buckets if b]
          self._buckets = [[] for _ in self._bounds]
        for tail in tails:
          self._out.put(self._Collate(tail))
        self._out.put(None)
        return
      out = self._processor(record)
      if out is None:
        continue
      example, key = out
      if key < 0:
        continue  # dropped (reference: negative bucketing key)
      batch = None
      with self._lock:
        for bi, bound in enumerate(self._bounds):
          if key <= bound:
            self._buckets[bi].append(example)
            if len(self._buckets[bi]) >= self._limits[bi]:
              batch = self._buckets[bi]
              self._buckets[bi] = []
            break
      if batch is not None:
        self._out.put(self._Collate(batch))

  @staticmethod
  def _Collate(examples: List[NestedMap]) -> NestedMap:
    """Pads each field's dim-0 to the max and stacks."""
    keys = [k for k, _ in examples[0].FlattenItems()]
    flat = [ex.Flatten() for ex in examples]
    out_vals = []
    for i in range(len(keys)):
      vals = [f[i] for f in flat]
      if isinstance(vals[0], torch.Tensor) and vals[0].dim() >= 1:
        maxlen = max(v.shape[0] for v in vals)
        padded = []
        for v in vals:
          if v.shape[0] < maxlen:
            pad_shape = (maxlen - v.shape[0],) + tuple(v.shape[1:])
            v = torch.cat([v, v.new_zeros(pad_shape)])
          padded.append(v)
        out_vals.append(torch.stack(padded))
      else:
        out_vals.append(torch.stack([torch.as_tensor(v) for v in vals]))
    return examples[0].Pack(out_vals)

  def GetNext(self, timeout: Optional[float] = 60.0) -> Optional[NestedMap]:
    item = self._out.get(timeout=timeout)
    return item

  def Stop(self):
    self._stop = True


class WeightedMixYielder:
  """Samples records from multiple yielders with given weights
  (reference weighted_mix_record_yielder.cc)."""

  def __init__(self, yielders, weights, seed: int = 301):
    import random
    assert len(yielders) == len(weights)
    self._yielders = yielders
    self._weights = [float(w) for w in weights]
    self._rng = random.Random(seed)

  def yield_record(self):
    i = self._rng.choices(range(len(self._yielders)),
                          weights=self._weights, k=1)[0]
    value, _ = self._yielders[i].yield_record()
    return value, i  # source_id = yielder index

  def stop(self):
    for y in self._yielders:
      y.stop()


def GenericInput(processor: Callable, file_pattern: str,
                 bucket_upper_bound: Sequence[int],
                 bucket_batch_limit: Sequence[int],
                 file_random_seed: int = 301,
                 file_buffer_size: int = 10000,
                 file_parallelism: int = 4,
                 num_batcher_threads: int = 2,
                 repeat: bool = True) -> RecordBatcher:
  """Builds the yielder+batcher pipeline (reference generic_input.py:104).

  processor(record_bytes) -> (NestedMap of tensors, int bucket_key);
  return None or a negative key to drop the record.
  """
  from lingvo_amd.ops import _loader
  ext = _loader.get_ext(required=True)
  fmt, files = ExpandFilePattern(file_pattern)
  yielder = ext.RecordYielder(files, fmt, file_random_seed,
                              file_buffer_size, file_parallelism, repeat)
  return RecordBatcher(yielder, processor, bucket_upper_bound,
                       bucket_batch_limit, num_threads=num_batcher_threads)


class SequentialYielder:
  """Strict-order record yielder (reference
  sequential_record_yielder.cc): files in the given order, records in
  file order, one epoch (eval/decode determinism — no shuffle buffer).
  Matches the native RecordYielder's yield_record interface."""

  def __init__(self, files, file_format: str = 'text'):
    assert file_format == 'text', 'sequential yielder: text only'
    self._files = list(files)
    self._gen = self._Iter()

  def _Iter(self):
    for src_id, path in enumerate(self._files):
      with open(path, 'rb') as f:
        for line in f:
          yield line.rstrip(b'\n'), src_id

  def yield_record(self):
    try:
      return next(self._gen)
    except StopIteration:
      raise StopIteration from None

  def current_epoch(self):
    return 1

  def stop(self):
    self._gen = iter(())
