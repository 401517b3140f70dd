"""Faithful beam-search step semantics (reference C++ op port in Python).

Implements the per-step pruning of the reference's
`lingvo/core/ops/beam_search_step_op_kernels.cc` (ComputeTopK :111,
Compute :681, UpdateAllDone :845) and its attrs at
`lingvo/core/ops/x_ops.cc:116-125`:

  - `valid_eos_max_logit_delta` — EOS terminates a hyp only if its
    global score is within delta of that hyp's best extension (and its
    local score exceeds `local_eos_threshold`).
  - `merge_paths` — epsilon-emitting models (RNN-T/NT): candidate hyps
    whose epsilon-stripped label sequences are identical are merged,
    with global scores combined by log-sum-exp
    (InsertHypWithEpsilonDedupe, kernels.h:138).
  - `ensure_full_beam` — a beam only counts as done once it holds
    num_hyps_per_beam terminated hyps AND every live hyp is below
    best_score - beam_size.
  - `force_eos_in_last_step` — at the final step, EOS terminations are
    accepted regardless of the score thresholds (hyps that still fail
    to emit EOS are dropped, matching the reference).

Hypothesis tables are kept in the reference's layout: N = K*B hyps,
hyp-major (hyp i belongs to beam i % B), token/prev-hyp tables [T, N].
The vocab top-k runs as a batched torch op (GPU when scores are on
GPU); per-beam merge and done-hyp bookkeeping are host-side, mirroring
the reference's CPU op (SURVEY §3.4 device->host boundary).
"""

from __future__ import annotations

import math
from dataclasses import dataclass, field
from typing import List, Optional, Tuple

import torch

from lingvo_amd.core.nested_map import NestedMap


def _log_sum_exp(a: float, b: float) -> float:
  m = max(a, b)
  return m + math.log(math.exp(a - m) + math.exp(b - m))


@dataclass
class Hyp:
  beam_id: int
  hyp_id: int
  word_id: int
  local_score: float
  global_score: float
  prev_labels: Tuple[int, ...] = ()


@dataclass
class DoneHyp:
  """A terminated hypothesis (reference Hypothesis proto payload)."""
  beam_id: int
  ids: List[int]
  scores: List[float]
  global_score: float
  step: int


# Test knob: force the generic merge-capable loop even when the
# vectorized fast path applies (equivalence fuzzing).
_DISABLE_FAST_PATH = False


@dataclass
class BeamSearchState:
  """Mutable search state across steps (the reference op's in/out
  tensors)."""
  num_beams: int
  k: int
  max_steps: int
  best_scores: List[float] = field(default_factory=list)      # [B]
  cumulative_scores: Optional[torch.Tensor] = None            # [N]
  hyps: Optional[torch.Tensor] = None                         # [T, N]
  prev_hyps: Optional[torch.Tensor] = None                    # [T, N]
  step_scores: Optional[torch.Tensor] = None                  # [T, N]
  done_hyps: List[DoneHyp] = field(default_factory=list)
  beam_done: List[bool] = field(default_factory=list)
  all_done: bool = False

  @classmethod
  def Init(cls, num_beams: int, k: int, max_steps: int) -> 'BeamSearchState':
    n = num_beams * k
    return cls(
        num_beams=num_beams, k=k, max_steps=max_steps,
        best_scores=[-1e30] * num_beams,
        cumulative_scores=torch.zeros(n),
        hyps=torch.zeros(max_steps, n, dtype=torch.long),
        prev_hyps=torch.zeros(max_steps, n, dtype=torch.long),
        step_scores=torch.zeros(max_steps, n),
        beam_done=[False] * num_beams)


def _AssemblePrevLabels(state: BeamSearchState, t: int,
                        eoc_id: int) -> List[Tuple[int, ...]]:
  """Walks prev_hyps back from step t to recover each hyp's
  (epsilon-stripped) label sequence (reference AssembleHyps :648)."""
  n = state.num_beams * state.k
  hyps = state.hyps[:t].tolist() if t else []
  prev = state.prev_hyps[:t].tolist() if t else []
  out = []
  for i in range(n):
    chain = [0] * t
    h = i
    for j in range(t - 1, -1, -1):
      chain[j] = h
      h = prev[j][h]
    labels = []
    for j in range(t):
      tok = hyps[j][chain[j]]
      if tok != eoc_id:
        labels.append(tok)
    out.append(tuple(labels))
  return out


def _TraceIds(state: BeamSearchState, hyp_id: int, t: int
              ) -> Tuple[List[int], List[float]]:
  """Token ids + per-step local scores of hyp ending at (t-1, hyp_id)."""
  chain = [0] * t
  h = hyp_id
  for j in range(t - 1, -1, -1):
    chain[j] = h
    h = int(state.prev_hyps[j, h])
  ids = [int(state.hyps[j, chain[j]]) for j in range(t)]
  scores = [float(state.step_scores[j, chain[j]]) for j in range(t)]
  return ids, scores


def BeamSearchStep(scores: torch.Tensor, state: BeamSearchState, t: int,
                   eos_id: int = 2, eoc_id: int = -1,
                   beam_size: float = 3.0,
                   valid_eos_max_logit_delta: float = 5.0,
                   local_eos_threshold: float = -100.0,
                   merge_paths: bool = False,
                   ensure_full_beam: bool = False,
                   force_eos_in_last_step: bool = False,
                   force_eos_in_top_k: bool = False,
                   allow_empty_terminated_hyp: bool = True,
                   is_last_chunk: Optional[torch.Tensor] = None,
                   beam_independence: bool = False) -> torch.Tensor:
  """One step: prunes `scores` [N, V] (log-probs of extending each live
  hyp) into the next K hyps per beam; returns gather indices [N] (the
  prev hyp each new slot continues; used to reorder decoder states).
  Mutates `state` in place."""
  b = state.num_beams
  k = state.k
  n = b * k
  assert scores.shape[0] == n
  is_last = t == state.max_steps - 1
  last_step_force = is_last and force_eos_in_last_step
  skip_beam = [beam_independence and state.beam_done[i] for i in range(b)]

  prev_labels = (_AssemblePrevLabels(state, t, eoc_id)
                 if merge_paths else [()] * n)
  cum = state.cumulative_scores

  # Device-side per-hyp top-(k+2): covers k survivors + eos (+eoc).
  topk_size = min(k + 2, scores.shape[1])
  if scores.is_cuda and topk_size <= 32:
    # HIP top-k kernel (K12): prune [N, V] on the GPU, ship only
    # [N, k+2] to the host hyp bookkeeping below.
    from lingvo_amd.ops import _loader
    ext = _loader.get_ext(required=True)
    total_dev = cum.to(scores.device).float().unsqueeze(1) + \
        scores.float()
    tv, ti = ext.topk_rows(total_dev.contiguous(), topk_size)
    eos_cols = total_dev[:, eos_id]
    scr_eos = scores[:, eos_id].float()
    if eoc_id >= 0:
      eoc_cols = total_dev[:, eoc_id]
      scr_eoc = scores[:, eoc_id].float()
    top_vals, top_idx = tv.cpu(), ti.cpu().long()
    eos_global = eos_cols.cpu()
    eos_local = scr_eos.cpu()
    if eoc_id >= 0:
      eoc_global = eoc_cols.cpu()
      eoc_local = scr_eoc.cpu()
    scores_f = None
  else:
    scores_f = scores.float().cpu()
    total = cum.unsqueeze(1) + scores_f                   # [N, V]
    top_vals, top_idx = total.topk(topk_size, dim=-1)
    eos_local = scores_f[:, eos_id]
    eos_global = total[:, eos_id]
    if eoc_id >= 0:
      eoc_local = scores_f[:, eoc_id]
      eoc_global = total[:, eoc_id]

  # Host-side merge per beam (reference merged_topk_vec).
  merged: List[List[Hyp]] = [[] for _ in range(b)]
  eos_done: List[Optional[Tuple[Hyp, int]]] = [None] * n

  # Vectorized fast path (no path merging, no epsilon): candidate
  # selection and table writes as tensor ops; only TERMINATED hyps
  # materialize python Hyp objects. Identical semantics to the generic
  # loop below (sort key (-global, word, hyp); EOS entries never join
  # the live pool; t==0 admits only hyp 0 per beam).
  if not merge_paths and eoc_id < 0 and not _DISABLE_FAST_PATH:
    k2 = top_vals.shape[1]
    vals = top_vals.clone()                        # [N, K2] global
    words = top_idx.clone()                        # [N, K2]
    if force_eos_in_top_k:
      no_eos = (words != eos_id).all(dim=1)
      if bool(no_eos.any()):
        rows = no_eos.nonzero(as_tuple=True)[0]
        words[rows, -1] = eos_id
        vals[rows, -1] = eos_global[rows]
    locals_ = vals - cum.unsqueeze(1)
    active = torch.tensor(
        [not skip_beam[i % b] and (t > 0 or i < b) for i in range(n)])
    # --- EOS terminations: first qualifying eos entry per row.
    best_global = vals[:, 0]
    thr = best_global - valid_eos_max_logit_delta
    is_eos = words == eos_id
    ok = torch.ones_like(vals, dtype=torch.bool) if last_step_force \
        else ((vals > thr.unsqueeze(1)) &
              (locals_ > local_eos_threshold))
    eos_ok = is_eos & ok & active.unsqueeze(1)
    has_eos = eos_ok.any(dim=1)
    first_eos = eos_ok.float().argmax(dim=1)
    for i in has_eos.nonzero(as_tuple=True)[0].tolist():
      c = int(first_eos[i])
      eos_done[i] = (Hyp(i % b, i, eos_id, float(locals_[i, c]),
                         float(vals[i, c]), ()), eos_id)
    # --- Live pool per beam: layout (j asc, col asc) == python order.
    NEG = -1e30
    pool_ok = (~is_eos) & active.unsqueeze(1)      # [N, K2]
    # [N, K2] -> [K, b, K2] -> [b, K*K2]
    def to_pool(x):
      return x.reshape(k, b, k2).permute(1, 0, 2).reshape(b, k * k2)
    pg = to_pool(vals.clone())
    pw = to_pool(words.clone())
    pl = to_pool(locals_.clone())
    pm = to_pool(pool_ok)
    hyp_ids = torch.arange(n).unsqueeze(1).expand(n, k2)
    ph = to_pool(hyp_ids.clone())
    pg = torch.where(pm, pg, torch.full_like(pg, NEG))
    big = torch.iinfo(torch.long).max
    pw = torch.where(pm, pw, torch.full_like(pw, big))
    ph = torch.where(pm, ph, torch.full_like(ph, big))
    #

    # Lexicographic stable sort: minor keys first.
    o1 = torch.argsort(ph, dim=1, stable=True)
    pg, pw, pl, ph, pm = (x.gather(1, o1) for x in (pg, pw, pl, ph, pm))
    o2 = torch.argsort(pw, dim=1, stable=True)
    pg, pw, pl, ph, pm = (x.gather(1, o2) for x in (pg, pw, pl, ph, pm))
    o3 = torch.argsort(-pg, dim=1, stable=True)
    pg, pw, pl, ph, pm = (x.gather(1, o3) for x in (pg, pw, pl, ph, pm))

    gather = torch.arange(n, dtype=torch.long)
    new_cum = cum.clone()
    for beam_id in range(b):
      if skip_beam[beam_id]:
        continue
      valid = int(pm[beam_id].sum())
      for j in range(k):
        i = j * b + beam_id
        if j < valid:
          state.hyps[t, i] = int(pw[beam_id, j])
          state.prev_hyps[t, i] = int(ph[beam_id, j])
          state.step_scores[t, i] = float(pl[beam_id, j])
          new_cum[i] = float(pg[beam_id, j])
          gather[i] = int(ph[beam_id, j])
        else:
          state.hyps[t, i] = eos_id
          state.prev_hyps[t, i] = beam_id
          state.step_scores[t, i] = -1e30
          new_cum[i] = -1e30
          gather[i] = beam_id
    state.cumulative_scores = new_cum
    _RecordTerminations(state, eos_done, t, b, k, n, merge_paths,
                        beam_size, ensure_full_beam, new_cum)
    return gather


  def _insert(beam: List[Hyp], h: Hyp):
    if merge_paths:
      for i, other in enumerate(beam):
        if _IsDuplicate(h, other, eoc_id):
          better = h if _Higher(h, other) else other
          beam[i] = Hyp(better.beam_id, better.hyp_id, better.word_id,
                        better.local_score,
                        _log_sum_exp(h.global_score, other.global_score),
                        better.prev_labels)
          return
    beam.append(h)

  top_vals_l = top_vals.tolist()
  top_idx_l = top_idx.tolist()
  cum_l = cum.tolist()
  eos_local_l = eos_local.tolist()
  eos_global_l = eos_global.tolist()

  for i in range(n):
    beam_id = i % b
    if skip_beam[beam_id]:
      continue
    if t == 0 and i >= b:
      # First step: only hyp 0 of each beam (reference :171).
      continue
    entries = [Hyp(beam_id, i, int(w), float(gs) - cum_l[i], float(gs),
                   prev_labels[i])
               for gs, w in zip(top_vals_l[i], top_idx_l[i])]
    if force_eos_in_top_k and all(e.word_id != eos_id for e in entries):
      entries[-1] = Hyp(beam_id, i, eos_id, eos_local_l[i],
                        eos_global_l[i], prev_labels[i])
    best_global = entries[0].global_score
    eos_threshold = best_global - valid_eos_max_logit_delta
    for e in entries:
      if e.word_id == eos_id:
        ok = last_step_force or (
            e.global_score > eos_threshold and
            e.local_score > local_eos_threshold)
        if ok and eos_done[i] is None:
          eos_done[i] = (e, eos_id)
      elif (eoc_id >= 0 and is_last_chunk is not None and
            bool(is_last_chunk[i]) and e.word_id == eoc_id):
        if (e.global_score > eos_threshold and
            e.local_score > local_eos_threshold and
            (allow_empty_terminated_hyp or e.prev_labels)):
          if eos_done[i] is None:
            eos_done[i] = (e, eoc_id)
        else:
          _insert(merged[beam_id], e)
      else:
        _insert(merged[beam_id], e)

  # Select next K live hyps per beam; write tables.
  gather = torch.arange(n, dtype=torch.long)
  new_cum = cum.clone()
  for beam_id in range(b):
    if skip_beam[beam_id]:
      continue
    ranked = sorted(merged[beam_id], key=_SortKey)
    for j in range(k):
      i = j * b + beam_id
      if j < len(ranked):
        h = ranked[j]
        state.hyps[t, i] = h.word_id
        state.prev_hyps[t, i] = h.hyp_id
        state.step_scores[t, i] = h.local_score
        new_cum[i] = h.global_score
        gather[i] = h.hyp_id
      else:  # fewer than K live continuations
        state.hyps[t, i] = eos_id
        state.prev_hyps[t, i] = beam_id
        state.step_scores[t, i] = -1e30
        new_cum[i] = -1e30
        gather[i] = beam_id
  state.cumulative_scores = new_cum
  _RecordTerminations(state, eos_done, t, b, k, n, merge_paths,
                      beam_size, ensure_full_beam, new_cum)
  return gather


def _RecordTerminations(state, eos_done, t, b, k, n, merge_paths,
                        beam_size, ensure_full_beam, new_cum):
  """Appends DoneHyps, updates best scores and beam_done/all_done
  (reference UpdateAllDone :845). Shared by the vectorized fast path
  and the generic merge-capable loop."""
  for i in range(n):
    if eos_done[i] is None:
      continue
    h, terminal = eos_done[i]
    beam_id = h.beam_id
    if h.global_score > state.best_scores[beam_id]:
      state.best_scores[beam_id] = h.global_score
    ids, step_scores = _TraceIds(state, h.hyp_id, t)
    ids.append(terminal)
    if merge_paths:
      avg = h.global_score / (t + 1)
      step_scores = [avg] * t + [avg]
    else:
      step_scores = step_scores + [h.local_score]
    state.done_hyps.append(DoneHyp(beam_id, ids, step_scores,
                                   h.global_score, t))

  for beam_id in range(b):
    if state.beam_done[beam_id]:
      continue
    if ensure_full_beam:
      num_done = sum(1 for d in state.done_hyps if d.beam_id == beam_id)
      if num_done < k:
        state.beam_done[beam_id] = False
        continue
    live = [float(new_cum[j * b + beam_id]) for j in range(k)]
    state.beam_done[beam_id] = all(
        s < state.best_scores[beam_id] - beam_size for s in live)
  state.all_done = all(state.beam_done)


def _Higher(x: Hyp, y: Hyp) -> bool:
  """HigherScore comparator (kernels.h:55)."""
  if x.global_score != y.global_score:
    return x.global_score > y.global_score
  if x.word_id != y.word_id:
    return x.word_id < y.word_id
  return x.hyp_id < y.hyp_id


def _SortKey(h: Hyp):
  return (-h.global_score, h.word_id, h.hyp_id)


def _IsDuplicate(cur: Hyp, other: Hyp, epsilon_id: int) -> bool:
  """IsDuplicateHyp (kernels.cc:47): epsilon-stripped sequences equal."""
  a, b = cur.prev_labels, other.prev_labels
  if cur.word_id == other.word_id:
    return a == b
  if cur.word_id == epsilon_id:
    return len(a) == len(b) + 1 and a[-1] == other.word_id and a[:-1] == b
  if other.word_id == epsilon_id:
    return len(b) == len(a) + 1 and b[-1] == cur.word_id and b[:-1] == a
  return False


def TopKTerminatedHyps(state: BeamSearchState, num_hyps_per_beam: int,
                       length_normalization: float = 0.0,
                       coverage_penalty: float = 0.0,
                       target_seq_length_ratio: float = 1.0) -> NestedMap:
  """Ranks each beam's terminated hyps by normalized score (reference
  TopKTerminatedHypsOp, x_ops.cc:322). Returns topk_ids [B, K, L],
  topk_lens [B, K], topk_scores [B, K] (normalized)."""
  b = state.num_beams
  max_len = max([len(d.ids) for d in state.done_hyps], default=1)

  def norm_score(d: DoneHyp) -> float:
    length = len(d.ids)
    norm = ((length + 5.0) / 6.0) ** length_normalization
    return d.global_score / norm

  ids = torch.zeros(b, num_hyps_per_beam, max_len, dtype=torch.long)
  lens = torch.zeros(b, num_hyps_per_beam, dtype=torch.long)
  out_scores = torch.full((b, num_hyps_per_beam), -1e30)
  for beam_id in range(b):
    beam_done = [d for d in state.done_hyps if d.beam_id == beam_id]
    beam_done.sort(key=norm_score, reverse=True)
    for j, d in enumerate(beam_done[:num_hyps_per_beam]):
      ids[beam_id, j, :len(d.ids)] = torch.tensor(d.ids)
      lens[beam_id, j] = len(d.ids)
      out_scores[beam_id, j] = norm_score(d)
  return NestedMap(topk_ids=ids, topk_lens=lens, topk_scores=out_scores)


def UnpackHyp(done: DoneHyp) -> NestedMap:
  """Per-hyp fields (reference UnpackHypOp, x_ops.cc:367)."""
  return NestedMap(
      beam_id=done.beam_id,
      ids=list(done.ids),
      scores=list(done.scores),
      normalized_score=done.global_score,
      seq_len=len(done.ids))
