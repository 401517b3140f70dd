"""Host-side eval metrics (reference lingvo/core/metrics.py)."""

from __future__ import annotations

import collections
import math
from typing import Dict, List, Optional

import torch


class BaseMetric:

  @property
  def value(self) -> float:
    raise NotImplementedError

  def Summary(self) -> Dict[str, float]:
    return {'value': self.value}


class AverageMetric(BaseMetric):
  """Weighted average (reference metrics.py:79)."""

  def __init__(self):
    self._total = 0.0
    self._weight = 0.0

  def Update(self, value: float, weight: float = 1.0) -> None:
    self._total += float(value) * float(weight)
    self._weight += float(weight)

  @property
  def value(self) -> float:
    return self._total / self._weight if self._weight else 0.0

  @property
  def total_weight(self) -> float:
    return self._weight


class UniqueAverageMetric(AverageMetric):
  """Average that de-duplicates by key (reference metrics.py)."""

  def __init__(self):
    super().__init__()
    self._seen = set()

  def Update(self, value: float, weight: float = 1.0, key=None) -> None:
    if key is not None:
      if key in self._seen:
        return
      self._seen.add(key)
    super().Update(value, weight)


class F1Metric(BaseMetric):
  """(reference metrics.py:183)."""

  def __init__(self):
    self._tp = 0.0
    self._fp = 0.0
    self._fn = 0.0

  def UpdateTruePositive(self, count: float = 1.0):
    self._tp += count

  def UpdateFalsePositive(self, count: float = 1.0):
    self._fp += count

  def UpdateFalseNegative(self, count: float = 1.0):
    self._fn += count

  @property
  def value(self) -> float:
    prec = self._tp / (self._tp + self._fp) if self._tp + self._fp else 0.0
    rec = self._tp / (self._tp + self._fn) if self._tp + self._fn else 0.0
    return 2 * prec * rec / (prec + rec) if prec + rec else 0.0


class CorpusBleuMetric(BaseMetric):
  """Corpus BLEU-4 with brevity penalty (reference metrics.py:240 /
  scorers.py). Tokenized on whitespace."""

  def __init__(self, max_order: int = 4):
    self._max_order = max_order
    self._match = [0] * max_order
    self._count = [0] * max_order
    self._ref_len = 0
    self._hyp_len = 0

  @staticmethod
  def _Ngrams(tokens: List[str], order: int):
    c = collections.Counter()
    for i in range(len(tokens) - order + 1):
      c[tuple(tokens[i:i + order])] += 1
    return c

  def Update(self, ref: str, hyp: str) -> None:
    ref_t, hyp_t = ref.split(), hyp.split()
    self._ref_len += len(ref_t)
    self._hyp_len += len(hyp_t)
    for n in range(self._max_order):
      rn = self._Ngrams(ref_t, n + 1)
      hn = self._Ngrams(hyp_t, n + 1)
      self._match[n] += sum((rn & hn).values())
      self._count[n] += max(0, len(hyp_t) - n)

  @property
  def value(self) -> float:
    if not all(self._count):
      return 0.0
    precisions = [self._match[n] / self._count[n] if self._count[n] else 0.0
                  for n in range(self._max_order)]
    if min(precisions) <= 0:
      return 0.0
    log_p = sum(math.log(p) for p in precisions) / self._max_order
    bp = 1.0 if self._hyp_len > self._ref_len else math.exp(
        1 - self._ref_len / max(1, self._hyp_len))
    return bp * math.exp(log_p)


class WerMetric(BaseMetric):
  """Word error rate via edit distance (ASR decoder metric; reference
  tasks/asr/decoder_metrics.py)."""

  def __init__(self):
    self._errors = 0.0
    self._words = 0.0

  @staticmethod
  def EditDistance(ref: List[str], hyp: List[str]) -> int:
    dp = list(range(len(hyp) + 1))
    for i in range(1, len(ref) + 1):
      prev = dp[0]
      dp[0] = i
      for j in range(1, len(hyp) + 1):
        cur = dp[j]
        dp[j] = min(dp[j] + 1, dp[j - 1] + 1,
                    prev + (ref[i - 1] != hyp[j - 1]))
        prev = cur
    return dp[-1]

  def Update(self, ref: str, hyp: str) -> None:
    ref_t, hyp_t = ref.split(), hyp.split()
    self._errors += self.EditDistance(ref_t, hyp_t)
    self._words += len(ref_t)

  @property
  def value(self) -> float:
    return self._errors / self._words if self._words else 0.0


class AUCMetric(BaseMetric):
  """ROC-AUC by rank statistic (reference metrics.py:461)."""

  def __init__(self):
    self._scores: List[float] = []
    self._labels: List[int] = []

  def Update(self, labels, scores, weights=None) -> None:
    self._scores.extend(float(s) for s in scores)
    self._labels.extend(int(l) for l in labels)

  @property
  def value(self) -> float:
    n_pos = sum(self._labels)
    n_neg = len(self._labels) - n_pos
    if not n_pos or not n_neg:
      return 0.0
    # Mann-Whitney with AVERAGE ranks for tied scores (a plain sort
    # would tie-break by label and bias the statistic).
    order = sorted(range(len(self._scores)),
                   key=lambda i: self._scores[i])
    ranks = [0.0] * len(order)
    i = 0
    while i < len(order):
      j = i
      while j + 1 < len(order) and \
          self._scores[order[j + 1]] == self._scores[order[i]]:
        j += 1
      avg = (i + j) / 2.0 + 1.0
      for k in range(i, j + 1):
        ranks[order[k]] = avg
      i = j + 1
    rank_sum = sum(r for r, l in zip(ranks, self._labels) if l)
    return (rank_sum - n_pos * (n_pos + 1) / 2) / (n_pos * n_neg)


def PackMetrics(metrics) -> torch.Tensor:
  """Packs {name: (value, weight)} into a flat [2N] tensor for a single
  all-reduce (reference TpuEvalMetrics, metrics.py:258)."""
  vals = []
  for name in sorted(metrics.keys()):
    v, w = metrics[name]
    vals.append(torch.as_tensor(v, dtype=torch.float32).flatten()[0] *
                torch.as_tensor(w, dtype=torch.float32).flatten()[0])
    vals.append(torch.as_tensor(w, dtype=torch.float32).flatten()[0])
  return torch.stack(vals) if vals else torch.zeros(0)


def UnpackMetrics(metrics_keys, packed: torch.Tensor):
  """Inverse of PackMetrics after sum-all-reduce: returns weighted means."""
  from lingvo_amd.core.nested_map import NestedMap
  out = NestedMap()
  for i, name in enumerate(sorted(metrics_keys)):
    vw = packed[2 * i]
    w = packed[2 * i + 1]
    out[name] = (vw / w.clamp_min(1e-8), w)
  return out
