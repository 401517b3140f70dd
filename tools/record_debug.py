#!/usr/bin/env python3
"""Dumps records from an input shard (reference
lingvo/core/ops/record_debug.cc:76 CLI). Supports the framework's file
formats: 'text:<path>' (one record per line), 'bytes:<path>'
(length-prefixed binary), 'tfrecord:<path>'.

Usage: python tools/record_debug.py text:/path/to/file --limit 5
"""
import argparse
import sys

sys.path.insert(0, __file__.rsplit('/', 2)[0])

from lingvo_amd.ops import _loader  # noqa: E402


def main():
  ap = argparse.ArgumentParser()
  ap.add_argument('pattern', help='typed path, e.g. text:/data/f.txt')
  ap.add_argument('--limit', type=int, default=10)
  args = ap.parse_args()
  ext = _loader.get_ext(required=True)
  y = ext.RecordYielder([args.pattern], seed=0, buffer_size=1,
                        parallelism=1) \
      if hasattr(ext, 'RecordYielder') else None
  if y is None:
    print('RecordYielder not available in extension', file=sys.stderr)
    return 1
  for i in range(args.limit):
    try:
      rec = y.Yield()
    except StopIteration:
      break
    body = rec if isinstance(rec, (bytes, str)) else getattr(
        rec, 'value', rec)
    if isinstance(body, bytes):
      shown = body[:200]
      print(f'[{i}] {len(body)} bytes: {shown!r}')
    else:
      print(f'[{i}] {body!r:.200}')
  return 0


if __name__ == '__main__':
  sys.exit(main())
