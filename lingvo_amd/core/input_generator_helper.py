"""Batch-splitting helpers for tower/model splits
(reference lingvo/core/input_generator_helper.py:21-120)."""

from __future__ import annotations

from typing import Dict, List, Sequence, Tuple

import torch

from lingvo_amd.core.nested_map import NestedMap


def ComputeSplits(batch_size: int, num_splits: int) -> List[int]:
  """floor(batch/num_splits) each, remainder round-robined from the
  front: ComputeSplits(5, 3) == [2, 2, 1] (reference :21)."""
  base = batch_size // num_splits
  rem = batch_size % num_splits
  return [base + (1 if i < rem else 0) for i in range(num_splits)]


def SplitTensors(xs: Sequence[torch.Tensor],
                 num_splits: int) -> Tuple[List[torch.Tensor], ...]:
  """Splits each tensor along dim 0 by ComputeSplits; all tensors must
  share the first dimension (reference :54)."""
  b = xs[0].shape[0]
  for x in xs:
    assert x.shape[0] == b, 'first dim of tensors in xs must match'
  assert b >= num_splits, 'first dim must be >= num_splits'
  sizes = ComputeSplits(b, num_splits)
  return tuple(list(x.split(sizes, dim=0)) for x in xs)


def SplitDictOfTensors(t_dict: Dict[str, torch.Tensor],
                       num_splits: int) -> List[Dict[str, torch.Tensor]]:
  """Splits a dict of same-batch tensors into num_splits dicts
  (reference :90)."""
  keys = sorted(t_dict.keys())
  split_lists = SplitTensors([t_dict[k] for k in keys], num_splits)
  return [{k: split_lists[j][i] for j, k in enumerate(keys)}
          for i in range(num_splits)]


def SplitNestedMap(batch: NestedMap, num_splits: int) -> List[NestedMap]:
  """Structure-preserving split of a NestedMap batch along dim 0."""
  flat = batch.Flatten()
  tensors = [t for t in flat if isinstance(t, torch.Tensor)]
  assert tensors, 'batch has no tensors'
  split_lists = SplitTensors(tensors, num_splits)
  out = []
  for i in range(num_splits):
    vals = []
    ti = 0
    for v in flat:
      if isinstance(v, torch.Tensor):
        vals.append(split_lists[ti][i])
        ti += 1
      else:
        vals.append(v)
    out.append(batch.Pack(vals))
  return out
