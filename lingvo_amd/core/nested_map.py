"""NestedMap: the universal nested container for batches, theta and states.

Provides the capability surface of the reference's NestedMap
(lingvo/core/nested_map.py:81): a dict subclass with attribute access,
dotted-path Get/Set, and structure-preserving Flatten/Pack/Transform/Filter.
Written from scratch for the MI355X-native framework; values are typically
torch.Tensors but any object is allowed.
"""

from __future__ import annotations

import copy as _copy
import re
from typing import Any, Callable, Iterable, List, Optional, Tuple

_VALID_KEY_RE = re.compile(r'^[a-zA-Z_][a-zA-Z0-9_]*$')
_RESERVED = frozenset(dir(dict)) | {'_VALID_KEY_RE'}


class NestedMap(dict):
  """A dict with attribute access and structure-aware utilities.

  Keys must be valid Python identifiers not shadowing dict methods.
  Flatten/Pack order is deterministic (sorted by key) so two NestedMaps
  with the same structure flatten to aligned lists.
  """

  __slots__ = ()

  def __init__(self, *args, **kwargs):
    super().__init__(*args, **kwargs)
    for key in self.keys():
      self.CheckKey(key)

  # ---- attribute access -------------------------------------------------
  def __getattr__(self, name: str) -> Any:
    try:
      return self[name]
    except KeyError as e:
      raise AttributeError(
          f'{name}; available attributes: {sorted(self.keys())}') from e

  def __setattr__(self, name: str, value: Any) -> None:
    self.CheckKey(name)
    self[name] = value

  def __delattr__(self, name: str) -> None:
    try:
      del self[name]
    except KeyError as e:
      raise AttributeError(name) from e

  def __setitem__(self, key: str, value: Any) -> None:
    self.CheckKey(key)
    super().__setitem__(key, value)

  @staticmethod
  def CheckKey(key: str) -> None:
    if not isinstance(key, str) or not _VALID_KEY_RE.match(key):
      raise ValueError(f'Invalid NestedMap key {key!r}')
    if key in _RESERVED:
      raise ValueError(f'NestedMap key {key!r} shadows a dict attribute')

  # ---- copies -----------------------------------------------------------
  def copy(self) -> 'NestedMap':  # shallow
    return NestedMap(self)

  def __deepcopy__(self, memo) -> 'NestedMap':
    ret = NestedMap()
    memo[id(self)] = ret
    for k, v in self.items():
      ret[k] = _copy.deepcopy(v, memo)
    return ret

  def DeepCopy(self) -> 'NestedMap':
    """Structure-deep copy; leaf values are shared (tensors not cloned)."""
    return self.Pack(self.Flatten())

  # ---- dotted-path access ----------------------------------------------
  def Get(self, path: str, default: Any = None) -> Any:
    """Returns self.a.b[3].c for path 'a.b[3].c', or default if missing."""
    cur: Any = self
    for part in path.split('.'):
      m = re.match(r'^([a-zA-Z_][a-zA-Z0-9_]*)((\[\d+\])*)$', part)
      if not m:
        return default
      name, idxs = m.group(1), m.group(2)
      if isinstance(cur, dict):
        if name not in cur:
          return default
        cur = cur[name]
      else:
        return default
      if idxs:
        for im in re.finditer(r'\[(\d+)\]', idxs):
          i = int(im.group(1))
          if not isinstance(cur, (list, tuple)) or i >= len(cur):
            return default
          cur = cur[i]
    return cur

  def Set(self, path: str, value: Any) -> None:
    """Sets a dotted path, creating intermediate NestedMaps/lists."""
    parts = path.split('.')
    cur: Any = self
    for pi, part in enumerate(parts):
      is_last = pi == len(parts) - 1
      m = re.match(r'^([a-zA-Z_][a-zA-Z0-9_]*)((\[\d+\])*)$', part)
      if not m:
        raise ValueError(f'Invalid path element {part!r}')
      name, idxs = m.group(1), m.group(2)
      idx_list = [int(im.group(1)) for im in re.finditer(r'\[(\d+)\]', idxs)]
      if not idx_list:
        if is_last:
          cur[name] = value
        else:
          if name not in cur or not isinstance(cur[name], dict):
            cur[name] = NestedMap()
          cur = cur[name]
      else:
        if name not in cur or not isinstance(cur[name], list):
          cur[name] = []
        seq = cur[name]
        for ii, idx in enumerate(idx_list):
          last_idx = ii == len(idx_list) - 1
          while len(seq) <= idx:
            seq.append(NestedMap() if (is_last and not last_idx) or
                       not last_idx else None)
          if last_idx:
            if is_last:
              seq[idx] = value
            else:
              if not isinstance(seq[idx], dict):
                seq[idx] = NestedMap()
              cur = seq[idx]
          else:
            if not isinstance(seq[idx], list):
              seq[idx] = []
            seq = seq[idx]

  # ---- structure traversal ----------------------------------------------
  def _RecKeysAndValues(self) -> List[Tuple[str, Any]]:
    out: List[Tuple[str, Any]] = []

    def recurse(prefix: str, val: Any):
      if isinstance(val, dict):
        for k in sorted(val.keys()):
          recurse(f'{prefix}.{k}' if prefix else str(k), val[k])
      elif isinstance(val, (list, tuple)):
        for i, v in enumerate(val):
          recurse(f'{prefix}[{i}]', v)
      else:
        out.append((prefix, val))

    recurse('', self)
    return out

  def Flatten(self) -> List[Any]:
    """Leaf values in deterministic (sorted-key) order."""
    return [v for _, v in self._RecKeysAndValues()]

  def FlattenItems(self) -> List[Tuple[str, Any]]:
    """List of (dotted_key, leaf_value) in deterministic order."""
    return self._RecKeysAndValues()

  def Pack(self, values: Iterable[Any]) -> 'NestedMap':
    """Returns a NestedMap with self's structure and `values` as leaves."""
    values = list(values)
    n_expected = len(self.Flatten())
    if len(values) != n_expected:
      raise ValueError(f'Pack expects {n_expected} values, got {len(values)}')
    it = iter(values)

    def recurse(val: Any) -> Any:
      if isinstance(val, dict):
        return NestedMap(
            {k: recurse(val[k]) for k in sorted(val.keys())})
      if isinstance(val, (list, tuple)):
        seq = [recurse(v) for v in val]
        return type(val)(seq) if isinstance(val, tuple) else seq
      return next(it)

    return recurse(self)

  def Transform(self, fn: Callable[[Any], Any]) -> 'NestedMap':
    """Applies fn to every leaf, preserving structure."""
    return self.Pack([fn(v) for v in self.Flatten()])

  def TransformWithKey(self, fn: Callable[[str, Any], Any]) -> 'NestedMap':
    return self.Pack([fn(k, v) for k, v in self.FlattenItems()])

  def Filter(self, pred: Callable[[Any], bool]) -> 'NestedMap':
    """Keeps only leaves where pred(value); prunes empty subtrees."""
    return self.FilterKeyVal(lambda _, v: pred(v))

  def FilterKeyVal(self, pred: Callable[[str, Any], bool]) -> 'NestedMap':
    def recurse(prefix: str, val: Any) -> Tuple[Any, bool]:
      if isinstance(val, dict):
        ret = NestedMap()
        for k in sorted(val.keys()):
          sub, keep = recurse(f'{prefix}.{k}' if prefix else str(k), val[k])
          if keep:
            ret[k] = sub
        return ret, bool(ret)
      if isinstance(val, (list, tuple)):
        seq = []
        for i, v in enumerate(val):
          sub, keep = recurse(f'{prefix}[{i}]', v)
          if keep:
            seq.append(sub)
        return seq, bool(seq)
      return val, pred(prefix, val)

    ret, _ = recurse('', self)
    return ret

  def IsCompatible(self, other: 'NestedMap') -> bool:
    """True if self and other have identical nested structure."""
    def sig(val: Any) -> Any:
      if isinstance(val, dict):
        return {k: sig(val[k]) for k in sorted(val.keys())}
      if isinstance(val, (list, tuple)):
        return [sig(v) for v in val]
      return None

    return sig(self) == sig(other)

  def GetKeys(self) -> List[str]:
    return [k for k, _ in self.FlattenItems()]

  def Union(self, other: 'NestedMap') -> 'NestedMap':
    ret = self.DeepCopy()

    def merge(dst, src):
      for k, v in src.items():
        if k in dst and isinstance(dst[k], dict) and isinstance(v, dict):
          merge(dst[k], v)
        else:
          dst[k] = v

    merge(ret, other)
    return ret

  def DebugString(self) -> str:
    return '\n'.join(f'{k}: {v!r}' for k, v in self.FlattenItems())

  @staticmethod
  def FromNestedDict(d: Any) -> Any:
    """Converts plain (nested) dicts to NestedMaps."""
    if isinstance(d, dict):
      return NestedMap({k: NestedMap.FromNestedDict(v) for k, v in d.items()})
    if isinstance(d, (list, tuple)):
      seq = [NestedMap.FromNestedDict(v) for v in d]
      return type(d)(seq) if isinstance(d, tuple) else seq
    return d
