"""Builds a WPM vocabulary from a text corpus (the offline counterpart
of the reference's wordpiece tooling; BPE-style pair merging).

  python tools/build_wpm_vocab.py --input corpus.txt --size 8000 \
      --output vocab.txt

Output: one piece per line, index = token id, starting with the
reserved <unk>/<s>/</s> ids the tokenizers expect. Word-start pieces
carry the '▁' mark, matching WpmTokenizer/WpmEncoder semantics.
"""

from __future__ import annotations

import argparse
import collections
from typing import Dict, List, Tuple

RESERVED = ['<unk>', '<s>', '</s>']
WORD_MARK = '▁'


def TrainWpmVocab(lines, vocab_size: int,
                  min_pair_count: int = 2) -> List[str]:
  """Greedy BPE: start from characters, repeatedly merge the most
  frequent adjacent pair until vocab_size pieces exist."""
  word_counts = collections.Counter()
  for line in lines:
    for w in line.split():
      word_counts[WORD_MARK + w] += 1
  # each word as a tuple of current pieces
  words: List[Tuple[List[str], int]] = [
      (list(w), c) for w, c in word_counts.items()]
  pieces = set(RESERVED)
  for segs, _ in words:
    pieces.update(segs)
  while len(pieces) < vocab_size:
    pair_counts: Dict[Tuple[str, str], int] = collections.Counter()
    for segs, c in words:
      for a, b in zip(segs, segs[1:]):
        pair_counts[(a, b)] += c
    if not pair_counts:
      break
    (a, b), count = pair_counts.most_common(1)[0]
    if count < min_pair_count:
      break
    merged = a + b
    pieces.add(merged)
    for segs, _ in words:
      i = 0
      while i < len(segs) - 1:
        if segs[i] == a and segs[i + 1] == b:
          segs[i:i + 2] = [merged]
        else:
          i += 1
  # order: reserved, then by length desc (longest-match friendliness is
  # handled by the encoder; ordering here just needs determinism)
  rest = sorted(p for p in pieces if p not in RESERVED)
  return RESERVED + rest


def main(argv=None):
  ap = argparse.ArgumentParser()
  ap.add_argument('--input', required=True)
  ap.add_argument('--output', required=True)
  ap.add_argument('--size', type=int, default=8000)
  args = ap.parse_args(argv)
  with open(args.input) as f:
    vocab = TrainWpmVocab(f, args.size)
  with open(args.output, 'w') as f:
    f.write('\n'.join(vocab) + '\n')
  print(f'wrote {len(vocab)} pieces to {args.output}')


if __name__ == '__main__':
  main()
