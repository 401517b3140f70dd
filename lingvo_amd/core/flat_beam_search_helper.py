"""Flat (fully-tensorized) beam search.

MI355X-native re-implementation of the reference's
lingvo/core/flat_beam_search_helper.py (:69 flat_beam_search, :52
update_nbest): the whole search — extension top-k, EOS handling and the
n-best pool — is expressed as batched tensor ops with NO host-side
python over hypotheses, so the decode loop stays on-device (and can be
captured in a hipGraph). Contrast with beam_search_helper.py, whose
n-best update loops over (batch, hyp) on the host.

Policy note: unlike BeamSearchHelper there is no
valid_eos_max_logit_delta gate — every step's EOS candidates enter the
n-best pool and compete on normalized score (keeping the update fully
tensorized).

Same callback contract as BeamSearchHelper:
  state = init_fn(batch, K)
  log_probs [B*K, V], state = step_fn(state, prev_ids [B*K])
  state = reorder_fn(state, gather_idx [B*K])
"""

from __future__ import annotations

from typing import Callable

import torch

from lingvo_amd.core.hyperparams import Params
from lingvo_amd.core.nested_map import NestedMap


def _UpdateNbest(nbest_scores, nbest_ids, nbest_lens, cand_scores,
                 cand_ids, cand_lens):
  """Tensorized n-best pool update (reference update_nbest:52): merge
  the candidate set into the pool and keep the top-K per beam, all via
  one sort. Shapes: pool [B, K(, L)], candidates [B, C(, L)]."""
  scores = torch.cat([nbest_scores, cand_scores], dim=1)   # [B, K+C]
  ids = torch.cat([nbest_ids, cand_ids], dim=1)
  lens = torch.cat([nbest_lens, cand_lens], dim=1)
  k = nbest_scores.shape[1]
  top, idx = scores.topk(k, dim=1)
  gather3 = idx.unsqueeze(-1).expand(-1, -1, ids.shape[-1])
  return top, ids.gather(1, gather3), lens.gather(1, idx)


class FlatBeamSearchHelper:

  @classmethod
  def Params(cls) -> Params:
    p = Params()
    p.Define('num_hyps_per_beam', 8, 'Beam width K.')
    p.Define('max_steps', 128, 'Max decode length.')
    p.Define('target_sos_id', 1, 'SOS.')
    p.Define('target_eos_id', 2, 'EOS.')
    p.Define('length_norm_alpha', 0.8, 'GNMT length norm alpha.')
    return p

  def __init__(self, params: Params):
    self.p = params

  def _Norm(self, length):
    a = self.p.length_norm_alpha
    return ((5.0 + length.float()) / 6.0) ** a

  @torch.no_grad()
  def BeamSearchDecode(self, batch: int, init_fn: Callable,
                       step_fn: Callable,
                       reorder_fn: Callable) -> NestedMap:
    p = self.p
    k = p.num_hyps_per_beam
    state = init_fn(batch, k)
    device = next(iter(t for t in state.Flatten()
                       if isinstance(t, torch.Tensor))).device
    bk = batch * k
    prev_ids = torch.full((bk,), p.target_sos_id, dtype=torch.long,
                          device=device)
    cum = torch.full((batch, k), -1e30, device=device)
    cum[:, 0] = 0.0
    hist = torch.zeros(batch, k, p.max_steps, dtype=torch.long,
                       device=device)
    nbest_scores = torch.full((batch, k), -1e30, device=device)
    nbest_ids = torch.zeros(batch, k, p.max_steps, dtype=torch.long,
                            device=device)
    nbest_lens = torch.zeros(batch, k, dtype=torch.long, device=device)
    barange = torch.arange(batch, device=device)

    for t in range(p.max_steps):
      log_probs, state = step_fn(state, prev_ids)
      v = log_probs.shape[-1]
      total = cum.reshape(bk, 1) + log_probs.float()

      # EOS candidates from every live hyp — tensorized pool update.
      eos_scores = total[:, p.target_eos_id].reshape(batch, k)
      eos_norm = eos_scores / self._Norm(
          torch.full((batch, k), t + 1, device=device))
      cand_ids = hist.clone()
      cand_ids[:, :, t] = p.target_eos_id
      cand_lens = torch.full((batch, k), t + 1, dtype=torch.long,
                             device=device)
      nbest_scores, nbest_ids, nbest_lens = _UpdateNbest(
          nbest_scores, nbest_ids, nbest_lens, eos_norm, cand_ids,
          cand_lens)

      # Extend: top-k over (hyp, vocab) without EOS.
      no_eos = total.clone()
      no_eos[:, p.target_eos_id] = -1e30
      top, idx = no_eos.reshape(batch, k * v).topk(k, dim=1)
      prev_hyp = idx // v
      new_tok = idx % v
      gather = (barange.unsqueeze(1) * k + prev_hyp).reshape(bk)
      hist = hist.reshape(bk, -1)[gather].reshape(batch, k, -1)
      hist[:, :, t] = new_tok
      cum = top
      prev_ids = new_tok.reshape(bk)
      state = reorder_fn(state, gather)

      # all-tensor early stop: live upper bound below the worst kept.
      live_bound = cum.max(dim=1).values / self._Norm(
          torch.full((batch,), t + 1, device=device))
      if bool((live_bound < nbest_scores.min(dim=1).values).all()):
        break

    # Terminated hyps take precedence (matching BeamSearchHelper):
    # still-live hyps only compete for beams whose pool has empty slots.
    live_norm = cum / self._Norm(
        torch.full((batch, k), p.max_steps, device=device))
    has_empty = (nbest_scores < -1e29).any(dim=1, keepdim=True)
    live_norm = torch.where(has_empty, live_norm,
                            torch.full_like(live_norm, -1e30))
    live_lens = torch.full((batch, k), p.max_steps, dtype=torch.long,
                           device=device)
    nbest_scores, nbest_ids, nbest_lens = _UpdateNbest(
        nbest_scores, nbest_ids, nbest_lens, live_norm, hist, live_lens)
    return NestedMap(topk_ids=nbest_ids, topk_lens=nbest_lens,
                     topk_scores=nbest_scores)
