"""Writes length-framed binary ASR shards for the native AsrFrameBatcher.

Record layout (record_batcher.cpp AsrFrameBatcher): uint32 record length,
then int32 T, int32 D, int32 L, float32 frames[T*D], int32 tokens[L].

  python tools/make_asr_shards.py --out /tmp/shards --num 4 \
      --records 1000 --frame-dim 80

Synthetic content (no network for the real corpus); the shard format is
the on-disk contract, exercised by tests/test_record_batcher.py and the
input benchmark.
"""

from __future__ import annotations

import argparse
import os
import random
import struct


def write_shard(path: str, records: int, frame_dim: int, rng: random.Random,
                min_frames: int = 200, max_frames: int = 1200,
                max_tokens: int = 64, vocab: int = 1024) -> None:
  with open(path, 'wb') as f:
    for _ in range(records):
      t = rng.randint(min_frames, max_frames)
      l = rng.randint(4, max_tokens)
      frames = [rng.uniform(-3, 3) for _ in range(t * frame_dim)]
      tokens = [rng.randint(3, vocab - 1) for _ in range(l)]
      rec = struct.pack('<iii', t, frame_dim, l)
      rec += struct.pack(f'<{len(frames)}f', *frames)
      rec += struct.pack(f'<{l}i', *tokens)
      f.write(struct.pack('<I', len(rec)))
      f.write(rec)


def main():
  ap = argparse.ArgumentParser()
  ap.add_argument('--out', required=True)
  ap.add_argument('--num', type=int, default=4, help='Shard count.')
  ap.add_argument('--records', type=int, default=1000, help='Per shard.')
  ap.add_argument('--frame-dim', type=int, default=80)
  ap.add_argument('--min-frames', type=int, default=200)
  ap.add_argument('--max-frames', type=int, default=1200)
  ap.add_argument('--seed', type=int, default=301)
  args = ap.parse_args()
  os.makedirs(args.out, exist_ok=True)
  rng = random.Random(args.seed)
  for i in range(args.num):
    path = os.path.join(args.out, f'shard-{i:05d}-of-{args.num:05d}.bin')
    write_shard(path, args.records, args.frame_dim, rng,
                args.min_frames, args.max_frames)
    print(path)


if __name__ == '__main__':
  main()
