"""Test utilities (reference lingvo/core/test_utils.py:278 TestCase,
:434 CompareToGoldenSingleFloat, :468 ComputeNumericGradient)."""

from __future__ import annotations

from typing import Callable

import torch


def ComputeNumericGradient(fn: Callable[[torch.Tensor], torch.Tensor],
                           x: torch.Tensor,
                           eps: float = 1e-3) -> torch.Tensor:
  """Central-difference gradient of scalar fn at x
  (reference test_utils.py:468). fn must return a scalar tensor."""
  x = x.detach().double()
  grad = torch.zeros_like(x)
  flat = x.reshape(-1)
  gflat = grad.reshape(-1)
  for i in range(flat.numel()):
    orig = flat[i].item()
    flat[i] = orig + eps
    fp = float(fn(x.reshape(x.shape)))
    flat[i] = orig - eps
    fm = float(fn(x.reshape(x.shape)))
    flat[i] = orig
    gflat[i] = (fp - fm) / (2 * eps)
  return grad


def CompareToGoldenSingleFloat(test_value: float, golden: float,
                               rtol: float = 1e-5,
                               atol: float = 1e-6) -> None:
  """Asserts a scalar matches its golden value (reference
  test_utils.py:434; goldens here are literals in the test source —
  update them by hand when a deliberate numeric change lands)."""
  if abs(test_value - golden) > atol + rtol * abs(golden):
    raise AssertionError(
        f'value {test_value!r} != golden {golden!r} '
        f'(diff {abs(test_value - golden):.3e})')


def AssertAllClose(a: torch.Tensor, b: torch.Tensor, rtol: float = 1e-5,
                   atol: float = 1e-6, msg: str = '') -> None:
  if not torch.allclose(a, b, rtol=rtol, atol=atol):
    diff = (a - b).abs().max().item()
    raise AssertionError(f'tensors differ (max {diff:.3e}) {msg}')
