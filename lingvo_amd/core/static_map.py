"""Static string<->int maps and call memoization (reference
lingvo/core/ops/static_map_op.cc:125 and
functional_ops_kernels.cc:127 CachedCall). In the torch runtime these
are host-side lookups, so plain-Python implementations keep the same
semantics without a custom op."""

from __future__ import annotations

from typing import Callable, Dict, List, Optional, Sequence

import torch


class StaticMapStringInt:
  """Bidirectional string<->int map with an unknown fallback."""

  def __init__(self, keys: Sequence[str],
               vals: Optional[Sequence[int]] = None,
               unk_int: int = -1, unk_str: str = ''):
    vals = list(vals) if vals is not None else list(range(len(keys)))
    assert len(keys) == len(vals)
    self._fwd: Dict[str, int] = dict(zip(keys, vals))
    self._bwd: Dict[int, str] = {v: k for k, v in zip(keys, vals)}
    self._unk_int = unk_int
    self._unk_str = unk_str

  def StringsToIds(self, strings: Sequence[str]) -> torch.Tensor:
    return torch.tensor(
        [self._fwd.get(s, self._unk_int) for s in strings],
        dtype=torch.int32)

  def IdsToStrings(self, ids) -> List[str]:
    return [self._bwd.get(int(i), self._unk_str) for i in ids]


class CachedCall:
  """Memoizes a nullary tensor-producing function (reference
  CachedCall): the wrapped fn runs once; later calls return the cached
  tensors."""

  def __init__(self, fn: Callable[[], object]):
    self._fn = fn
    self._cached = None
    self._called = False

  def __call__(self):
    if not self._called:
      self._cached = self._fn()
      self._called = True
    return self._cached
