"""Corpus-level scorers (reference lingvo/core/scorers.py:35
Unsegmenter, :65 BleuScorer)."""

from __future__ import annotations

import collections
import math
from typing import List


def NGrams(lst: List[str], order: int):
  """All n-grams of the given order (reference scorers.py:30)."""
  return [tuple(lst[i:i + order]) for i in range(len(lst) - order + 1)]


class Unsegmenter:
  """Merges BPE/WPM/SPM-segmented strings back to surface form
  (reference scorers.py:35)."""

  _BPE_SEPARATOR = '@@ '
  _WPM_SEPARATOR = '▁'  # same for SPM

  def __init__(self, separator_type=None):
    self._separator_type = separator_type

  def __call__(self, line: str) -> str:
    if self._separator_type == 'bpe':
      return line.replace(self._BPE_SEPARATOR, '').strip()
    if self._separator_type in ('wpm', 'spm'):
      return line.replace(' ', '').replace(self._WPM_SEPARATOR,
                                           ' ').strip()
    return line


class BleuScorer:
  """Corpus BLEU with brevity penalty (reference scorers.py:65).
  Accumulates (reference, hypothesis) sentence pairs; `ComputeOverall`
  returns 4-gram BLEU."""

  def __init__(self, max_ngram: int = 4,
               separator_type=None):
    self._max_ngram = max_ngram
    self._unseg = Unsegmenter(separator_type)
    self._matches = [0] * max_ngram
    self._hyp_counts = [0] * max_ngram
    self._ref_len = 0
    self._hyp_len = 0

  @property
  def unsegmenter(self):
    return self._unseg

  def AddSentence(self, ref_str: str, hyp_str: str) -> None:
    ref = self._unseg(ref_str).split()
    hyp = self._unseg(hyp_str).split()
    self._ref_len += len(ref)
    self._hyp_len += len(hyp)
    for n in range(self._max_ngram):
      ref_counts = collections.Counter(NGrams(ref, n + 1))
      hyp_counts = collections.Counter(NGrams(hyp, n + 1))
      self._hyp_counts[n] += max(len(hyp) - n, 0)
      for g, c in hyp_counts.items():
        self._matches[n] += min(c, ref_counts.get(g, 0))

  def ComputeOverallScore(self) -> float:
    log_precision = 0.0
    for n in range(self._max_ngram):
      if self._matches[n] == 0 or self._hyp_counts[n] == 0:
        return 0.0
      log_precision += math.log(self._matches[n] / self._hyp_counts[n])
    log_precision /= self._max_ngram
    bp = 1.0
    if self._hyp_len < self._ref_len and self._hyp_len > 0:
      bp = math.exp(1.0 - self._ref_len / self._hyp_len)
    return bp * math.exp(log_precision)
