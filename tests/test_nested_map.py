"""Tests for NestedMap (reference nested_map_test.py capability)."""

import pytest
import torch

from lingvo_amd.core.nested_map import NestedMap


def test_attr_access():
  m = NestedMap(a=1)
  m.b = 2
  assert m.a == 1 and m['b'] == 2
  del m.a
  assert 'a' not in m
  with pytest.raises(AttributeError):
    _ = m.zzz


def test_invalid_keys():
  with pytest.raises(ValueError):
    NestedMap(**{'bad key': 1})
  m = NestedMap()
  with pytest.raises(ValueError):
    m['items'] = 3  # shadows dict method


def test_flatten_pack_order():
  m = NestedMap(b=NestedMap(y=2, x=1), a=0, c=[3, 4])
  flat = m.Flatten()
  assert flat == [0, 1, 2, 3, 4]
  packed = m.Pack([10, 11, 12, 13, 14])
  assert packed.a == 10 and packed.b.x == 11 and packed.b.y == 12
  assert packed.c == [13, 14]


def test_flatten_items_keys():
  m = NestedMap(a=NestedMap(b=[NestedMap(c=5)]))
  items = m.FlattenItems()
  assert items == [('a.b[0].c', 5)]


def test_transform_filter():
  m = NestedMap(a=1, b=NestedMap(c=2, d=3))
  t = m.Transform(lambda v: v * 10)
  assert t.b.c == 20
  f = m.Filter(lambda v: v % 2 == 1)
  assert 'a' in f and 'd' in f.b and 'c' not in f.b


def test_get_set_dotted():
  m = NestedMap()
  m.Set('x.y.z', 7)
  assert m.Get('x.y.z') == 7
  assert m.Get('x.q', 'dflt') == 'dflt'
  m.Set('arr[1].v', 5)
  assert m.Get('arr[1].v') == 5


def test_is_compatible():
  a = NestedMap(x=1, y=NestedMap(z=2))
  b = NestedMap(x='s', y=NestedMap(z=None))
  c = NestedMap(x=1)
  assert a.IsCompatible(b)
  assert not a.IsCompatible(c)


def test_tensors_as_leaves():
  m = NestedMap(w=torch.ones(2), sub=NestedMap(v=torch.zeros(3)))
  flat = m.Flatten()
  assert len(flat) == 2
  packed = m.Pack([t + 1 for t in flat])
  assert torch.equal(packed.sub.v, torch.ones(3))


def test_from_nested_dict_union():
  m = NestedMap.FromNestedDict({'a': {'b': 1}})
  assert isinstance(m.a, NestedMap)
  u = m.Union(NestedMap(a=NestedMap(c=2)))
  assert u.a.b == 1 and u.a.c == 2
