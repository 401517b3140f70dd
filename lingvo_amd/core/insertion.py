"""Insertion-based sequence generation (reference lingvo/core/insertion.py
SymbolInsertionLayer; KERMIT / Insertion Transformer, Chan et al. 2019).

Training samples a partial "canvas" (a random subsequence of the
target); the model learns to predict, for every canvas slot, which
symbols must still be inserted there. Generation then inserts symbols
in parallel until every slot emits end-of-slot.
"""

from __future__ import annotations

import torch

from lingvo_amd.core import py_utils
from lingvo_amd.core.base_layer import BaseLayer
from lingvo_amd.core.nested_map import NestedMap


class SymbolInsertionLayer(BaseLayer):
  """Creates insertion-training canvases and targets.

  FProp(x [B, T] ids, paddings [B, T]) returns NestedMap:
    canvas           [B, C]  sampled subsequence (order preserved)
    canvas_paddings  [B, C]
    target_indices   [N, 3]  (batch, slot, symbol) insertion targets:
                             symbol must be inserted BEFORE canvas
                             position `slot` (slot == canvas len means
                             append at the end)
    target_weights   [N]
  Every non-canvas token appears exactly once as a target, keyed to the
  slot holding the next canvas token after it — so (canvas, targets)
  reconstructs x exactly.
  """

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('rollin_policy', 'uniform',
             "'uniform': canvas size ~ U[0, len]; 'oracle': fixed frac.")
    p.Define('oracle_frac', 0.5, 'Canvas fraction for oracle policy.')
    return p

  def FProp(self, theta: NestedMap, x: torch.Tensor,
            paddings: torch.Tensor) -> NestedMap:
    p = self.p
    b, t = x.shape
    lens = py_utils.LengthsFromPaddings(paddings).long()  # [B]
    device = x.device
    # per-sequence random keep-count k in [0, len]
    if p.rollin_policy == 'uniform':
      u = py_utils.GraphSafeUniform((b,), device)
      k = (u * (lens + 1).float()).long().clamp(max=lens)
    else:
      k = (lens.float() * p.oracle_frac).long()
    # sample k positions uniformly: random scores, keep top-k, sorted
    scores = py_utils.GraphSafeUniform((b, t), device)
    scores = scores.masked_fill(paddings > 0.5, -1.0)
    order = scores.argsort(dim=1, descending=True)
    keep = torch.zeros(b, t, dtype=torch.bool, device=device)
    arange_t = torch.arange(t, device=device)
    for i in range(b):  # small python loop over batch; shapes are tiny
      keep[i, order[i, :k[i]]] = True
    canvas_len = int(k.max().item())
    canvas = torch.zeros(b, max(1, canvas_len), dtype=x.dtype,
                         device=device)
    canvas_paddings = torch.ones(b, max(1, canvas_len), device=device)
    tgt = []
    for i in range(b):
      kept_pos = arange_t[keep[i]]
      canvas[i, :len(kept_pos)] = x[i, kept_pos]
      canvas_paddings[i, :len(kept_pos)] = 0.0
      # every dropped token inserts before the next kept slot
      slot_of_pos = torch.searchsorted(kept_pos, arange_t, right=False)
      for pos in range(int(lens[i])):
        if not keep[i, pos]:
          tgt.append((i, int(slot_of_pos[pos]), int(x[i, pos])))
    if tgt:
      target_indices = torch.tensor(tgt, dtype=torch.long, device=device)
      target_weights = torch.ones(len(tgt), device=device)
    else:
      target_indices = torch.zeros(0, 3, dtype=torch.long, device=device)
      target_weights = torch.zeros(0, device=device)
    return NestedMap(canvas=canvas, canvas_paddings=canvas_paddings,
                     target_indices=target_indices,
                     target_weights=target_weights)


def ReconstructFromCanvas(canvas: torch.Tensor,
                          canvas_paddings: torch.Tensor,
                          target_indices: torch.Tensor) -> list:
  """Inverse of SymbolInsertionLayer (for tests/decoding): merge the
  insertion targets back into each canvas row. Targets for the same
  slot insert in their listed order."""
  b = canvas.shape[0]
  out = []
  for i in range(b):
    clen = int((canvas_paddings[i] < 0.5).sum())
    slots = [[] for _ in range(clen + 1)]
    for bi, slot, sym in target_indices.tolist():
      if bi == i:
        slots[slot].append(sym)
    seq = []
    for s in range(clen):
      seq.extend(slots[s])
      seq.append(int(canvas[i, s]))
    seq.extend(slots[clen])
    out.append(seq)
  return out
