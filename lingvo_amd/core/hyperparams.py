"""Params: typed hierarchical hyperparameter trees.

Re-creates the capability of the reference's hyperparams
(lingvo/core/hyperparams.py:266): Define/Get/Set/Copy/Freeze, dotted-path
access, text round-trip (ToText/FromText), TextDiff, and
InstantiableParams(cls).Instantiate(). Implemented from scratch in pure
Python for the MI355X-native framework.
"""

from __future__ import annotations

import ast
import copy as _copy
import dataclasses
import enum
import inspect
import re
from typing import Any, Callable, Dict, List, Optional, Tuple, Type


class _SortedDict(dict):
  pass


def _QuoteString(s: str) -> str:
  return repr(s)


class _Param:
  """One named hyperparameter."""

  __slots__ = ('name', 'value', 'description')

  def __init__(self, name: str, default: Any, description: str):
    self.name = name
    self.value = default
    self.description = description

  def __deepcopy__(self, memo):
    try:
      value = _copy.deepcopy(self.value, memo)
    except TypeError:
      # Runtime handles (e.g. torch.distributed ProcessGroup) are not
      # copyable — share them by reference (reference hyperparams
      # behavior for non-copyable leaf objects).
      value = self.value
    p = _Param(self.name, value, self.description)
    memo[id(self)] = p
    return p


def _IsNamedTuple(x) -> bool:
  return isinstance(x, tuple) and hasattr(x, '_fields')


class Params:
  """A nested, typed hyperparameter container.

  Usage:
    p = Params()
    p.Define('learning_rate', 0.1, 'LR.')
    p.learning_rate = 0.2
    q = p.Copy()
  """

  _immutable: bool

  def __init__(self):
    self.__dict__['_immutable'] = False
    self.__dict__['_params'] = {}  # name -> _Param

  # ---- define/get/set ---------------------------------------------------
  def Define(self, name: str, default: Any, description: str) -> None:
    if self._immutable:
      raise TypeError('This Params instance is immutable.')
    if not re.match(r'^[a-z][a-z0-9_]*$', name):
      raise AttributeError(f'Invalid param name: {name!r}')
    if name in self._params:
      raise AttributeError(f'Parameter {name!r} is already defined')
    self._params[name] = _Param(name, default, description)

  def Undefine(self, name: str) -> None:
    if self._immutable:
      raise TypeError('This Params instance is immutable.')
    del self._params[name]

  def Freeze(self) -> None:
    self.__dict__['_immutable'] = True

  def IsImmutable(self) -> bool:
    return self._immutable

  def __setattr__(self, name: str, value: Any) -> None:
    if self._immutable:
      raise TypeError('This Params instance is immutable.')
    if name not in self._params:
      raise AttributeError(self._KeyErrorString(name))
    self._params[name].value = value

  def __getattr__(self, name: str) -> Any:
    if name.startswith('_'):
      raise AttributeError(name)
    try:
      return self.__dict__['_params'][name].value
    except KeyError:
      raise AttributeError(self._KeyErrorString(name)) from None

  def _KeyErrorString(self, name: str) -> str:
    similar = [k for k in self._params if name in k or k in name]
    return f'{name} (did you mean: {sorted(similar)}; known: ' \
           f'{sorted(self._params)})'

  def __dir__(self):
    return sorted(self._params.keys())

  def __contains__(self, name: str) -> bool:
    return name in self._params

  def __len__(self) -> int:
    return len(self._params)

  def __eq__(self, other) -> bool:
    return isinstance(other, Params) and self.ToText() == other.ToText()

  def __ne__(self, other) -> bool:
    return not self == other

  def IterParams(self):
    for name, p in self._params.items():
      yield name, p.value

  def GetKeys(self) -> List[str]:
    return sorted(self._params.keys())

  # ---- dotted access ----------------------------------------------------
  def _NavigateTo(self, path: str) -> Tuple['Params', str]:
    parts = path.split('.')
    cur: Any = self
    for part in parts[:-1]:
      m = re.match(r'^([a-z][a-z0-9_]*)(\[(\d+)\])?$', part)
      if not m:
        raise AttributeError(f'Invalid path element {part!r}')
      cur = getattr(cur, m.group(1))
      if m.group(3) is not None:
        cur = cur[int(m.group(3))]
    return cur, parts[-1]

  def Get(self, path: str) -> Any:
    cur, leaf = self._NavigateTo(path)
    m = re.match(r'^([a-z][a-z0-9_]*)(\[(\d+)\])?$', leaf)
    val = getattr(cur, m.group(1))
    if m.group(3) is not None:
      val = val[int(m.group(3))]
    return val

  def Set(self, **kwargs) -> 'Params':
    """Sets multiple params (dotted names use __ as separator not allowed;
    use SetPath for dotted paths). Returns self for chaining."""
    for name, value in kwargs.items():
      if self._immutable:
        raise TypeError('This Params instance is immutable.')
      if name not in self._params:
        raise AttributeError(self._KeyErrorString(name))
      self._params[name].value = value
    return self

  def SetPath(self, path: str, value: Any) -> 'Params':
    cur, leaf = self._NavigateTo(path)
    setattr(cur, leaf, value)
    return self

  def Delete(self, *names: str) -> 'Params':
    if self._immutable:
      raise TypeError('This Params instance is immutable.')
    for name in names:
      del self._params[name]
    return self

  # ---- copy -------------------------------------------------------------
  def Copy(self) -> 'Params':
    return _copy.deepcopy(self)

  def __deepcopy__(self, memo):
    ret = type(self).__new__(type(self))
    ret.__dict__['_immutable'] = False
    ret.__dict__['_params'] = {
        k: _copy.deepcopy(v, memo) for k, v in self._params.items()
    }
    for k, v in self.__dict__.items():
      if k not in ('_immutable', '_params'):
        ret.__dict__[k] = _copy.deepcopy(v, memo)
    memo[id(self)] = ret
    return ret

  # ---- text round trip --------------------------------------------------
  def ToText(self, prefix: str = '') -> str:
    """Serializes to 'dotted.path : value' lines, sorted."""
    lines: List[str] = []

    def fmt(val: Any) -> str:
      if isinstance(val, str):
        return _QuoteString(val)
      if isinstance(val, enum.Enum):
        return f'{type(val).__name__}.{val.name}'
      if isinstance(val, type):
        return f'type/{val.__module__}/{val.__qualname__}'
      if callable(val) and hasattr(val, '__qualname__'):
        return f'fn/{getattr(val, "__module__", "?")}/{val.__qualname__}'
      try:
        import torch
        if isinstance(val, torch.dtype):
          return str(val)
      except ImportError:
        pass
      return repr(val)

    def recurse(p: Any, pref: str):
      if isinstance(p, Params):
        for name in sorted(p._params):
          recurse(p._params[name].value, f'{pref}{name}.')
      elif isinstance(p, (list, tuple)) and any(
          isinstance(v, Params) for v in p):
        for i, v in enumerate(p):
          recurse(v, f'{pref[:-1]}[{i}].')
      else:
        lines.append(f'{pref[:-1]} : {fmt(p)}')

    recurse(self, prefix)
    return '\n'.join(lines) + '\n'

  def FromText(self, text: str) -> 'Params':
    """Applies 'dotted.path : value' lines to this Params tree in place.

    Only literal values (via ast.literal_eval) are restored; class/fn
    values must already be structurally present (a design shared with the
    reference, which needs type hints for non-literals).
    """
    if self._immutable:
      raise TypeError('This Params instance is immutable.')
    for line in text.splitlines():
      line = line.strip()
      if not line or line.startswith('#'):
        continue
      key, _, val = line.partition(' : ')
      key = key.strip()
      val = val.strip()
      try:
        parsed = ast.literal_eval(val)
      except (ValueError, SyntaxError):
        continue  # non-literal (class/fn/dtype): keep existing value
      try:
        self.SetPath(key, parsed)
      except AttributeError:
        continue
    return self

  def ToProto(self):
    """Serializes to a google.protobuf Struct (reference
    hyperparams.py:529 ToProto — the MI355X-native equivalent uses the
    well-known Struct type instead of a custom Hyperparam schema). Each
    leaf path maps to {'t': tag, 'v': value}; classes/functions/dtypes
    are stored by qualified name and restored by import in FromProto."""
    from google.protobuf import struct_pb2

    def encode(val):
      if val is None:
        return 'none', 0
      if isinstance(val, bool):
        return 'bool', val
      if isinstance(val, int):
        return 'int', val
      if isinstance(val, float):
        return 'float', val
      if isinstance(val, str):
        return 'str', val
      if isinstance(val, enum.Enum):
        return 'enum', (f'{type(val).__module__}/'
                        f'{type(val).__qualname__}/{val.name}')
      if isinstance(val, type):
        return 'type', f'{val.__module__}/{val.__qualname__}'
      try:
        import torch
        if isinstance(val, torch.dtype):
          return 'dtype', str(val)
      except ImportError:
        pass
      if callable(val) and hasattr(val, '__qualname__'):
        return 'fn', (f'{getattr(val, "__module__", "?")}/'
                      f'{val.__qualname__}')
      return 'literal', repr(val)

    proto = struct_pb2.Struct()

    def recurse(p, pref):
      if isinstance(p, Params):
        for name in sorted(p._params):
          recurse(p._params[name].value, f'{pref}{name}.')
      elif isinstance(p, (list, tuple)) and any(
          isinstance(v, Params) for v in p):
        for i, v in enumerate(p):
          recurse(v, f'{pref[:-1]}[{i}].')
      else:
        tag, enc = encode(p)
        entry = struct_pb2.Struct()
        entry.fields['t'].string_value = tag
        if tag in ('int', 'float'):
          entry.fields['v'].number_value = enc
        elif tag == 'bool':
          entry.fields['v'].bool_value = enc
        elif tag == 'none':
          entry.fields['v'].null_value = 0
        else:
          entry.fields['v'].string_value = enc
        proto.fields[pref[:-1]].struct_value.CopyFrom(entry)

    recurse(self, '')
    return proto

  def FromProto(self, proto) -> 'Params':
    """Applies a ToProto() Struct to this tree in place; classes,
    functions, dtypes and enums are restored by importing their
    qualified names (reference hyperparams.py:611 FromProto)."""
    import importlib

    def resolve(qual):
      mod, _, rest = qual.partition('/')
      obj = importlib.import_module(mod)
      for part in rest.split('.'):
        obj = getattr(obj, part)
      return obj

    for key, entry in proto.fields.items():
      s = entry.struct_value
      tag = s.fields['t'].string_value
      v = s.fields['v']
      if tag == 'none':
        val = None
      elif tag == 'bool':
        val = v.bool_value
      elif tag == 'int':
        val = int(v.number_value)
      elif tag == 'float':
        val = v.number_value
      elif tag == 'str':
        val = v.string_value
      elif tag == 'dtype':
        import torch
        val = getattr(torch, v.string_value.split('.')[-1])
      elif tag in ('type', 'fn'):
        val = resolve(v.string_value)
      elif tag == 'enum':
        mod_qual, _, member = v.string_value.rpartition('/')
        val = getattr(resolve(mod_qual), member)
      else:  # literal
        val = ast.literal_eval(v.string_value)
      try:
        self.SetPath(key, val)
      except AttributeError:
        continue
    return self

  def TextDiff(self, other: 'Params') -> str:
    """Returns a unified human-readable diff of two Params trees."""
    mine = dict(
        l.split(' : ', 1) for l in self.ToText().splitlines() if ' : ' in l)
    theirs = dict(
        l.split(' : ', 1) for l in other.ToText().splitlines() if ' : ' in l)
    out = []
    for k in sorted(set(mine) | set(theirs)):
      a, b = mine.get(k), theirs.get(k)
      if a != b:
        if a is not None:
          out.append(f'< {k} : {a}')
        if b is not None:
          out.append(f'> {k} : {b}')
    return '\n'.join(out) + ('\n' if out else '')

  def __str__(self) -> str:
    return self.ToText()

  def __repr__(self) -> str:
    return self.ToText()


class InstantiableParams(Params):
  """Params bound to a class; Instantiate() constructs it."""

  def __init__(self, cls: Optional[Type] = None):
    super().__init__()
    self.Define('cls', cls, 'Class to instantiate.')

  def Instantiate(self, **kwargs):
    assert self.cls is not None, 'Params.cls is unset'
    return self.cls(self, **kwargs)


def CopyParamsTo(from_p: Params, to_p: Params,
                 skip: Optional[List[str]] = None) -> Params:
  """Copies fields defined in both from_p and to_p (reference
  hyperparams.py:197 CopyFieldsTo)."""
  skip = set(skip or [])
  for name, value in from_p.IterParams():
    if name in skip or name == 'cls':
      continue
    if name in to_p:
      setattr(to_p, name, _copy.deepcopy(value))
  return to_p
