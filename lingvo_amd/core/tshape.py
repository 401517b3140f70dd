"""Symbolic tensor shapes (reference lingvo/core/tshape.py:21 Shape):
a shape whose dims may be symbolic expressions; used by FPropMeta-style
cost metadata (see parallel/pipeline.py PartitionByCost for the
consumer)."""

from __future__ import annotations

from lingvo_amd.core import symbolic


class Shape:

  def __init__(self, dims):
    self._dims = list(dims)

  def __getitem__(self, i):
    if isinstance(i, slice):
      return Shape(self._dims[i])
    return self._dims[i]

  def __len__(self):
    return len(self._dims)

  def __add__(self, other):
    """Concatenation (reference Shape + Shape)."""
    other_dims = other._dims if isinstance(other, Shape) else list(other)
    return Shape(self._dims + other_dims)

  @property
  def rank(self):
    return len(self._dims)

  def num_elements(self):
    out = 1
    for d in self._dims:
      out = out * d
    return out

  def ToTensorShape(self):
    """Concrete dims under the active SymbolToValueMap."""
    return [symbolic.EvalExpr(d) for d in self._dims]

  def __repr__(self):
    return f'Shape({self._dims})'
