"""Functional recurrence (reference lingvo/core/recurrent.py:985 Recurrent).

`Recurrent(theta, state0, inputs, cell_fn)` scans cell_fn over the leading
time dim of every tensor in `inputs`, threading state. The reference's
hand-written backward (re-running cell_fn per step without stashing
intermediates, recurrent.py:1023-1026) maps to per-step
torch.utils.checkpoint when `remat=True`; plain autograd otherwise.
"""

from __future__ import annotations

from typing import Callable, Optional, Tuple

import torch

from lingvo_amd.core.nested_map import NestedMap


def _SliceT(inputs: NestedMap, t: int) -> NestedMap:
  return inputs.Transform(lambda x: x[t] if isinstance(x, torch.Tensor)
                          else x)


def Recurrent(theta: NestedMap, state0: NestedMap, inputs: NestedMap,
              cell_fn: Callable[[NestedMap, NestedMap, NestedMap],
                                Tuple[NestedMap, NestedMap]],
              remat: bool = False,
              stop_fn=None,
              check_stateful_ops: bool = False
              ) -> Tuple[NestedMap, NestedMap]:
  """Returns (acc_states, final_state).

  cell_fn(theta, state, inputs_t) -> (new_state, extras). acc_states
  stacks every new_state along a leading time dim.

  check_stateful_ops (reference recurrent.py:1046 /
  StatefulRandomOpsInDefun py_utils.py:5106): the remat backward
  RE-RUNS cell_fn, so hidden statefulness (global RNG, in-place
  mutation of captured state) silently corrupts gradients. With the
  flag on, the first step runs cell_fn twice and asserts identical
  outputs — a runtime stand-in for the reference's graph-time scan.
  """
  t_max = None
  for v in inputs.Flatten():
    if isinstance(v, torch.Tensor):
      t_max = v.shape[0] if t_max is None else min(t_max, v.shape[0])
  assert t_max is not None, 'Recurrent needs at least one tensor input'

  if check_stateful_ops and t_max > 0:
    probe_in = _SliceT(inputs, 0)
    out_a, _ = cell_fn(theta, state0, probe_in)
    out_b, _ = cell_fn(theta, state0, probe_in)
    for va, vb in zip(out_a.Flatten(), out_b.Flatten()):
      if isinstance(va, torch.Tensor) and not torch.equal(va, vb):
        raise RuntimeError(
            'Recurrent cell_fn is stateful (two runs differ) — '
            'recompute-based backward would be wrong. Use the '
            'deterministic step-seed RNG (py_utils.GraphSafeUniform) '
            'inside scanned cells.')
  state = state0
  acc = []
  for t in range(t_max):
    inp_t = _SliceT(inputs, t)
    if remat and torch.is_grad_enabled():
      flat_state = state.Flatten()
      flat_inp = inp_t.Flatten()
      n_state = len(flat_state)

      def step(*tensors, _state_tpl=state, _inp_tpl=inp_t):
        st = _state_tpl.Pack(tensors[:n_state])
        it = _inp_tpl.Pack(tensors[n_state:])
        new_state, _ = cell_fn(theta, st, it)
        return tuple(new_state.Flatten())

      out = torch.utils.checkpoint.checkpoint(
          step, *(flat_state + flat_inp), use_reentrant=False)
      state = state.Pack(list(out))
    else:
      state, _ = cell_fn(theta, state, inp_t)
    acc.append(state)
    if stop_fn is not None and stop_fn(t, theta, state):
      break

  # Stack accumulated states along time.
  stacked = acc[0].Pack([
      torch.stack([s.Flatten()[i] for s in acc])
      for i in range(len(acc[0].Flatten()))
  ])
  return stacked, state
