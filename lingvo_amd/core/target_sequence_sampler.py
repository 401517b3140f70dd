"""Temperature / top-k / top-p sampling decode
(reference lingvo/core/target_sequence_sampler.py:35)."""

from __future__ import annotations

from typing import Callable, Optional

import torch

from lingvo_amd.core.hyperparams import Params
from lingvo_amd.core.nested_map import NestedMap


class TargetSequenceSampler:

  @classmethod
  def Params(cls) -> Params:
    p = Params()
    p.Define('target_sos_id', 1, 'SOS.')
    p.Define('target_eos_id', 2, 'EOS.')
    p.Define('max_steps', 128, 'Max length.')
    p.Define('temperature', 1.0, 'Softmax temperature.')
    p.Define('top_k', 0, 'If >0, sample from the top-k logits.')
    p.Define('top_p', 0.0, 'If >0, nucleus sampling mass.')
    p.Define('random_seed', None, 'Seed.')
    return p

  def __init__(self, params: Params):
    self.p = params

  def Sample(self, batch: int, init_fn, step_fn) -> NestedMap:
    """step_fn(state, prev_ids) -> (logits [B, V], state)."""
    p = self.p
    state = init_fn(batch, 1)
    device = next(iter(s for s in state.Flatten()
                       if isinstance(s, torch.Tensor))).device
    gen = torch.Generator(device=device)
    gen.manual_seed(p.random_seed or 90210)
    prev = torch.full((batch,), p.target_sos_id, dtype=torch.long,
                      device=device)
    done = torch.zeros(batch, dtype=torch.bool, device=device)
    ids = []
    for _ in range(p.max_steps):
      logits, state = step_fn(state, prev)
      logits = logits.float()
      if p.temperature != 1.0:
        logits = logits / max(p.temperature, 1e-6)
      if p.top_k > 0:
        kth = logits.topk(p.top_k, dim=-1).values[:, -1:]
        logits = logits.masked_fill(logits < kth, -1e30)
      if p.top_p > 0.0:
        sorted_logits, order = logits.sort(dim=-1, descending=True)
        probs = torch.softmax(sorted_logits, dim=-1)
        cum = probs.cumsum(dim=-1)
        cut = cum - probs > p.top_p
        sorted_logits = sorted_logits.masked_fill(cut, -1e30)
        logits = torch.full_like(logits, -1e30).scatter(
            -1, order, sorted_logits)
      probs = torch.softmax(logits, dim=-1)
      prev = torch.multinomial(probs, 1, generator=gen).squeeze(-1)
      prev = torch.where(done, torch.full_like(prev, p.target_eos_id),
                         prev)
      done = done | (prev == p.target_eos_id)
      ids.append(prev)
      if bool(done.all()):
        break
    out = torch.stack(ids, dim=1)
    lens = (out != p.target_eos_id).long().sum(-1) + 1
    return NestedMap(ids=out, lens=lens.clamp_max(out.shape[1]))
