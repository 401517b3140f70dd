"""Speculative decoding for LM serving (beyond the reference, which
has no serving-side decode acceleration; Leviathan et al. 2023).

A small draft LM proposes `lookahead` tokens autoregressively; the
target LM scores the whole proposed block in ONE forward pass (the
GEMM-shaped call MI355X wants — verification is batched, only the
draft runs token-by-token). Greedy mode accepts the longest prefix
where the target's argmax agrees, then takes the target's correction
token — the output is IDENTICAL to target-only greedy decoding, just
in fewer target forward passes.
"""

from __future__ import annotations

import torch

from lingvo_amd.core.nested_map import NestedMap


class SpeculativeDecoder:

  def __init__(self, target_lm, target_theta, draft_lm, draft_theta,
               lookahead: int = 4, eos_id: int = 2):
    self.target = target_lm
    self.t_theta = target_theta
    self.draft = draft_lm
    self.d_theta = draft_theta
    self.k = lookahead
    self.eos = eos_id
    self.stats = dict(target_calls=0, draft_calls=0, accepted=0,
                      proposed=0)

  def _Logits(self, lm, theta, ids):
    pads = torch.zeros(ids.shape, dtype=torch.float32,
                       device=ids.device)
    act = lm.FProp(theta, ids, pads)
    return lm.softmax.Logits(theta.softmax, act)

  @torch.no_grad()
  def Generate(self, prefix: torch.Tensor, max_new: int = 64
               ) -> NestedMap:
    """prefix [B, T0] -> NestedMap(ids [B, T0+n], new_tokens n).
    Greedy; stops early when every row has emitted EOS."""
    ids = prefix.clone()
    b = ids.shape[0]
    done = torch.zeros(b, dtype=torch.bool, device=ids.device)
    new = 0
    while new < max_new and not bool(done.all()):
      # 1) draft proposes k tokens autoregressively
      draft_ids = ids
      for _ in range(min(self.k, max_new - new)):
        logits = self._Logits(self.draft, self.d_theta, draft_ids)
        self.stats['draft_calls'] += 1
        nxt = logits[:, -1].argmax(-1, keepdim=True)
        draft_ids = torch.cat([draft_ids, nxt], dim=1)
      prop = draft_ids[:, ids.shape[1]:]              # [B, k']
      kp = prop.shape[1]
      self.stats['proposed'] += kp * b
      # 2) ONE target pass over prefix + proposal scores all positions
      logits = self._Logits(self.target, self.t_theta, draft_ids)
      self.stats['target_calls'] += 1
      # target's argmax at position t predicts token t+1
      t0 = ids.shape[1]
      tgt_pred = logits[:, t0 - 1:t0 + kp - 1].argmax(-1)  # [B, k']
      agree = (tgt_pred == prop)
      # longest agreed prefix per row
      n_acc = (agree.cumprod(dim=1)).sum(dim=1)            # [B]
      self.stats['accepted'] += int(n_acc.sum())
      # 3) emit accepted tokens + the target's correction token
      min_acc = int(n_acc.min())
      step_tokens = []
      for j in range(min_acc):
        step_tokens.append(prop[:, j])
      # the first disagreement position (or kp) gets the target token
      corr_pos = t0 - 1 + n_acc.clamp(max=kp)
      corr = logits[torch.arange(b, device=ids.device),
                    corr_pos].argmax(-1)
      # rows that accepted more than min_acc: their extra accepted
      # tokens match the target anyway; for simplicity advance by
      # min_acc + 1 each round (still exact — every emitted token is
      # the target's greedy choice).
      nxt = prop[:, min_acc] if min_acc < kp else corr
      nxt = torch.where(n_acc > min_acc, nxt, corr)
      step_tokens.append(nxt)
      for tok in step_tokens:
        tok = torch.where(done, torch.full_like(tok, self.eos), tok)
        ids = torch.cat([ids, tok.unsqueeze(1)], dim=1)
        done = done | (tok == self.eos)
        new += 1
        if new >= max_new:
          break
    return NestedMap(ids=ids, new_tokens=new, stats=dict(self.stats))


def GreedyReference(lm, theta, prefix: torch.Tensor, max_new: int,
                    eos_id: int = 2) -> torch.Tensor:
  """Token-by-token target-only greedy decode (the oracle)."""
  ids = prefix.clone()
  b = ids.shape[0]
  done = torch.zeros(b, dtype=torch.bool, device=ids.device)
  with torch.no_grad():
    for _ in range(max_new):
      pads = torch.zeros(ids.shape, dtype=torch.float32,
                         device=ids.device)
      act = lm.FProp(theta, ids, pads)
      logits = lm.softmax.Logits(theta.softmax, act)
      tok = logits[:, -1].argmax(-1)
      tok = torch.where(done, torch.full_like(tok, eos_id), tok)
      ids = torch.cat([ids, tok.unsqueeze(1)], dim=1)
      done = done | (tok == eos_id)
      if bool(done.all()):
        break
  return ids
