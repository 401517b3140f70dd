"""Symbolic dimensions (reference lingvo/core/symbolic.py:21).

The reference uses sympy for FPropMeta flop/shape estimation; sympy is
not in this environment, so this is a self-contained expression tree
supporting +,-,*,//,/ and substitution — the full API surface the
layer code needs (Symbol, NewSymbol, EvalExpr, IsExpr,
SymbolToValueMap context)."""

from __future__ import annotations

import threading
from typing import Dict, Union

_LOCAL = threading.local()


class Expr:
  """Arithmetic over symbols builds an expression tree."""

  def _eval(self, env):
    raise NotImplementedError

  def __add__(self, o):
    return _Bin('+', self, o)

  def __radd__(self, o):
    return _Bin('+', o, self)

  def __sub__(self, o):
    return _Bin('-', self, o)

  def __rsub__(self, o):
    return _Bin('-', o, self)

  def __mul__(self, o):
    return _Bin('*', self, o)

  def __rmul__(self, o):
    return _Bin('*', o, self)

  def __floordiv__(self, o):
    return _Bin('//', self, o)

  def __truediv__(self, o):
    return _Bin('/', self, o)

  def __repr__(self):
    return self._repr()

  def _repr(self):
    return 'Expr'


class Symbol(Expr):
  """A named symbolic dimension (reference symbolic.py Symbol)."""

  def __init__(self, name: str):
    self.name = name

  def _eval(self, env):
    if self in env:
      return env[self]
    if self.name in env:
      return env[self.name]
    raise KeyError(f'unbound symbol {self.name}')

  def _repr(self):
    return self.name

  def __hash__(self):
    return hash(('sym', self.name))

  def __eq__(self, other):
    # Identity-style equality on the name keeps dict lookups sane while
    # arithmetic still builds trees (match reference Symbol semantics).
    return isinstance(other, Symbol) and other.name == self.name


def NewSymbol(name: str) -> Symbol:
  return Symbol(name)


class _Bin(Expr):

  def __init__(self, op, a, b):
    self.op, self.a, self.b = op, a, b

  def _eval(self, env):
    a = self.a._eval(env) if isinstance(self.a, Expr) else self.a
    b = self.b._eval(env) if isinstance(self.b, Expr) else self.b
    if self.op == '+':
      return a + b
    if self.op == '-':
      return a - b
    if self.op == '*':
      return a * b
    if self.op == '//':
      return a // b
    return a / b

  def _repr(self):
    ra = self.a._repr() if isinstance(self.a, Expr) else repr(self.a)
    rb = self.b._repr() if isinstance(self.b, Expr) else repr(self.b)
    return f'({ra} {self.op} {rb})'


def IsExpr(x) -> bool:
  return isinstance(x, Expr)


class SymbolToValueMap:
  """Context manager binding symbols to values (reference
  symbolic.py SymbolToValueMap; STATIC_VALUES semantics — bindings
  nest and restore)."""

  def __init__(self, values: Dict[Union[Symbol, str], int]):
    self._values = dict(values)

  def __enter__(self):
    stack = getattr(_LOCAL, 'stack', None)
    if stack is None:
      stack = _LOCAL.stack = []
    stack.append(self._values)
    return self

  def __exit__(self, *a):
    _LOCAL.stack.pop()
    return False

  @staticmethod
  def Current() -> Dict:
    merged: Dict = {}
    for frame in getattr(_LOCAL, 'stack', []):
      merged.update(frame)
    return merged


def EvalExpr(x):
  """Evaluates expressions under the current SymbolToValueMap; plain
  numbers (and nested lists/tuples) pass through (reference
  symbolic.py:95 EvalExpr)."""
  if isinstance(x, (list, tuple)):
    out = [EvalExpr(v) for v in x]
    return type(x)(out)
  if isinstance(x, Expr):
    return x._eval(SymbolToValueMap.Current())
  return x
