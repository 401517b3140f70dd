"""Tokenizers (reference lingvo/core/tokenizers.py:26-449 —
AsciiTokenizer, VocabFileTokenizer, WpmTokenizer; C++ ops
ascii_tokenizer.cc / tokenizer_ops_kernels.cc).

StringsToIds returns (ids [B, maxlen] with SOS prefix, labels [B, maxlen]
with EOS suffix, paddings) following the reference contract.
"""

from __future__ import annotations

import re

from typing import Dict, List, Optional, Sequence, Tuple

import torch

from lingvo_amd.core.base_layer import BaseLayer
from lingvo_amd.core.nested_map import NestedMap


class BaseTokenizer(BaseLayer):

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('vocab_size', 0, 'Vocab size.')
    p.Define('target_sos_id', 1, 'SOS id.')
    p.Define('target_eos_id', 2, 'EOS id.')
    p.Define('target_unk_id', 0, 'UNK id.')
    return p

  def _TokensToIds(self, text: str) -> List[int]:
    raise NotImplementedError

  def _IdsToTokens(self, ids: Sequence[int]) -> str:
    raise NotImplementedError

  def StringsToIds(self, strs: Sequence[str], max_length: int
                   ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    p = self.p
    b = len(strs)
    ids = torch.full((b, max_length), p.target_eos_id, dtype=torch.long)
    labels = torch.full((b, max_length), p.target_eos_id, dtype=torch.long)
    paddings = torch.ones(b, max_length)
    for i, s in enumerate(strs):
      toks = self._TokensToIds(s)[:max_length - 1]
      n = len(toks)
      ids[i, 0] = p.target_sos_id
      ids[i, 1:n + 1] = torch.tensor(toks, dtype=torch.long)
      labels[i, :n] = torch.tensor(toks, dtype=torch.long)
      labels[i, n] = p.target_eos_id
      paddings[i, :n + 1] = 0.0
    return ids, labels, paddings

  def IdsToStrings(self, ids: torch.Tensor,
                   lens: Optional[torch.Tensor] = None) -> List[str]:
    p = self.p
    out = []
    for i in range(ids.shape[0]):
      row = ids[i].tolist()
      if lens is not None:
        row = row[:int(lens[i])]
      row = [t for t in row if t not in
             (p.target_sos_id, p.target_eos_id)]
      out.append(self._IdsToTokens(row))
    return out


class AsciiTokenizer(BaseTokenizer):
  """Char-level (reference ascii_tokenizer.cc): ids 0=unk,1=sos,2=eos,
  3=' ', then lowercase chars/digits/punct."""

  CHARS = ' abcdefghijklmnopqrstuvwxyz0123456789' \
          '!"\'&.,:;%/?+-=()[]$#@'

  @classmethod
  def Params(cls):
    p = super().Params()
    p.vocab_size = 76
    return p

  def _TokensToIds(self, text: str) -> List[int]:
    base = 3
    out = []
    for ch in text.lower():
      idx = self.CHARS.find(ch)
      out.append(base + idx if idx >= 0 else self.p.target_unk_id)
    return out

  def _IdsToTokens(self, ids: Sequence[int]) -> str:
    base = 3
    chars = []
    for t in ids:
      j = t - base
      chars.append(self.CHARS[j] if 0 <= j < len(self.CHARS) else '?')
    return ''.join(chars)


class VocabFileTokenizer(BaseTokenizer):
  """Whitespace tokens looked up in a vocab list (reference
  tokenizers.py VocabFileTokenizer / simple_vocab.cc)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('token_vocab_filepath', None, 'One token per line.')
    p.Define('tokens', None, 'Inline token list (overrides file).')
    return p

  def __init__(self, params):
    super().__init__(params)
    toks = self.p.tokens
    if toks is None and self.p.token_vocab_filepath:
      with open(self.p.token_vocab_filepath) as f:
        toks = [l.rstrip('\n') for l in f]
    self._vocab = {t: i for i, t in enumerate(toks or [])}
    self._inv = {i: t for t, i in self._vocab.items()}

  def _TokensToIds(self, text: str) -> List[int]:
    return [self._vocab.get(w, self.p.target_unk_id)
            for w in text.split()]

  def _IdsToTokens(self, ids: Sequence[int]) -> str:
    return ' '.join(self._inv.get(t, '<unk>') for t in ids)


class WpmTokenizer(VocabFileTokenizer):
  """Greedy longest-match wordpiece (reference core/wpm_encoder.py).
  Word-internal continuation pieces carry no marker; word starts are
  prefixed with '▁' (sentencepiece-style).

  When the native extension is built, encoding runs in the C++
  `WpmEncoder` (hip/wpm_tokenizer.cpp — GIL-released, multi-threaded
  batch path, the MI355X-native stand-in for the reference's
  tokenizer_ops_kernels.cc); the Python scan below is the fallback and
  the numerics oracle for its tests."""

  WORD_MARK = '▁'

  def __init__(self, params):
    super().__init__(params)
    self._native = None
    from lingvo_amd.ops import _loader
    ext = _loader.get_ext()
    if ext is not None and hasattr(ext, 'WpmEncoder'):
      pieces = [self._inv.get(i, '\0<gap>') for i in
                range(max(self._inv) + 1)] if self._inv else []
      self._native = ext.WpmEncoder(pieces, self.p.target_unk_id)

  def EncodeBatch(self, lines: Sequence[str],
                  num_threads: int = 4) -> List[List[int]]:
    """Batch encode; C++ multi-threaded when the extension is built."""
    if self._native is not None:
      return self._native.encode_batch(list(lines), num_threads)
    return [self._TokensToIds(l) for l in lines]

  def _EncodeWord(self, word: str) -> List[int]:
    p = self.p
    pieces = []
    s = self.WORD_MARK + word
    i = 0
    while i < len(s):
      j = len(s)
      while j > i:
        if s[i:j] in self._vocab:
          pieces.append(self._vocab[s[i:j]])
          break
        j -= 1
      else:
        pieces.append(p.target_unk_id)
        j = i + 1
      i = j
    return pieces

  # Tokenization splits on ASCII whitespace ONLY (space/tab/CR/LF/FF/VT)
  # — an explicit contract shared with the C++ encoder. (Python's
  # str.split() also treats \x1c-\x1f and Unicode spaces as separators,
  # which the byte-level native path must not.)
  _WS = re.compile(r'[ \t\n\r\f\v]+')

  def _TokensToIds(self, text: str) -> List[int]:
    if self._native is not None:
      return list(self._native.encode(text))
    out = []
    for w in self._WS.split(text):
      if w:
        out.extend(self._EncodeWord(w))
    return out

  def _IdsToTokens(self, ids: Sequence[int]) -> str:
    s = ''.join(self._inv.get(t, '?') for t in ids)
    return s.replace(self.WORD_MARK, ' ').strip()


class BpeTokenizer(BaseTokenizer):
  """Byte-pair-encoding tokenizer (reference core/tokenizers.py:305
  BpeTokenizer + BpeWordsToIds / BpeIdsToWords ops, x_ops.cc:613-849).

  `codes`: merge rules in priority order, each 'a b' (the pair merged
  into 'ab'); words end with the '</w>' marker. `tokens`: subword ->
  id vocabulary list (index = id). Common words can be pre-cached via
  `words_to_ids` {word: [ids]} (the reference's words_to_ids file)."""

  WORD_END = '</w>'

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('codes', [], "Merge rules ['a b', ...] in priority order.")
    p.Define('codes_filepath', None, 'File of merge rules (one per line).')
    p.Define('tokens', [], 'Subword vocab (index = id).')
    p.Define('vocab_filepath', None, 'File of subwords (one per line).')
    p.Define('words_to_ids', {}, 'Optional pre-tokenized word cache.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    codes = list(p.codes)
    if p.codes_filepath:
      with open(p.codes_filepath) as f:
        codes = [ln.rstrip('\n') for ln in f
                 if ln.strip() and not ln.startswith('#')]
    self._ranks = {}
    for i, line in enumerate(codes):
      a, b = line.split()
      self._ranks[(a, b)] = i
    tokens = list(p.tokens)
    if p.vocab_filepath:
      with open(p.vocab_filepath) as f:
        tokens = [ln.rstrip('\n') for ln in f if ln.rstrip('\n')]
    self._vocab = {t: i for i, t in enumerate(tokens)}
    self._inv = {i: t for t, i in self._vocab.items()}
    self._cache = {w: list(ids) for w, ids in p.words_to_ids.items()}

  def _BpeWord(self, word: str) -> List[str]:
    pieces = list(word[:-1]) + [word[-1] + self.WORD_END]
    while len(pieces) > 1:
      best, best_rank = None, None
      for i in range(len(pieces) - 1):
        r = self._ranks.get((pieces[i], pieces[i + 1]))
        if r is not None and (best_rank is None or r < best_rank):
          best, best_rank = i, r
      if best is None:
        break
      pieces = (pieces[:best] + [pieces[best] + pieces[best + 1]] +
                pieces[best + 2:])
    return pieces

  def _TokensToIds(self, text: str) -> List[int]:
    unk = self.p.target_unk_id
    out: List[int] = []
    for w in text.split():
      if not w:
        continue
      cached = self._cache.get(w)
      if cached is None:
        cached = [self._vocab.get(piece, unk)
                  for piece in self._BpeWord(w)]
        self._cache[w] = cached
      out.extend(cached)
    return out

  def _IdsToTokens(self, ids: Sequence[int]) -> str:
    s = ''.join(self._inv.get(t, '?') for t in ids)
    return s.replace(self.WORD_END, ' ').strip()
