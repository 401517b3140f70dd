"""Aggregates metrics.jsonl files under a logdir into one table.

  python tools/summarize_metrics.py --logdir /tmp/run [--last N]

Walks every program/runner output dir (train/, eval_*/, decoder_*/,
executor_metrics.jsonl) and prints the latest records side by side —
the quick-look counterpart of the reference's TensorBoard dirs.
"""

from __future__ import annotations

import argparse
import glob
import json
import os


def Summarize(logdir: str, last: int = 3) -> str:
  lines = []
  paths = sorted(glob.glob(os.path.join(logdir, '**', '*.jsonl'),
                           recursive=True))
  for path in paths:
    rel = os.path.relpath(path, logdir)
    with open(path) as f:
      recs = [json.loads(l) for l in f if l.strip()]
    if not recs:
      continue
    lines.append(f'== {rel} ({len(recs)} records)')
    for rec in recs[-last:]:
      kv = '  '.join(
          f'{k}={v:.5g}' if isinstance(v, float) else f'{k}={v}'
          for k, v in rec.items())
      lines.append(f'   {kv}')
  return '\n'.join(lines)


def main(argv=None):
  ap = argparse.ArgumentParser()
  ap.add_argument('--logdir', required=True)
  ap.add_argument('--last', type=int, default=3)
  args = ap.parse_args(argv)
  print(Summarize(args.logdir, args.last))


if __name__ == '__main__':
  main()
