"""Activation registry (reference lingvo/core/activations.py)."""

from __future__ import annotations

import math
from typing import Callable, Dict

import torch
import torch.nn.functional as F

_ACTIVATIONS: Dict[str, Callable] = {
    'RELU': F.relu,
    'RELU6': lambda x: torch.clamp(x, 0.0, 6.0),
    'SIGMOID': torch.sigmoid,
    'TANH': torch.tanh,
    'GELU': lambda x: F.gelu(x, approximate='none'),
    'GELU_APPROXIMATE': lambda x: F.gelu(x, approximate='tanh'),
    'SWISH': F.silu,
    'SILU': F.silu,
    'SOFTPLUS': F.softplus,
    'NONE': lambda x: x,
}


def GetFn(name: str) -> Callable:
  key = name.upper()
  if key not in _ACTIVATIONS:
    raise KeyError(f'Unknown activation {name!r}; known: '
                   f'{sorted(_ACTIVATIONS)}')
  return _ACTIVATIONS[key]


def Register(name: str, fn: Callable) -> None:
  _ACTIVATIONS[name.upper()] = fn
