"""Sequence packing (reference lingvo/core/ops/pack_ops.cc:
PackSequences x_ops.cc:1061, ApplyPacking :1228, text_packing.{h,cc}).

Greedy first-fit packing of variable-length (src, tgt) pairs into fixed
[B, T] rows; emits segment_ids, segment_pos and indices_in_input, which
feed the flash-attention kernel's segment mask path.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

from lingvo_amd.core.nested_map import NestedMap


def PackSequences(src_lens: torch.Tensor, tgt_lens: torch.Tensor,
                  packed_batch: int, src_time: int, tgt_time: int
                  ) -> NestedMap:
  """Greedy first-fit packing (reference pack_ops.cc semantics).

  Returns per-side [packed_batch, time] segment_ids (1-based, 0 =
  empty), segment_pos, and indices_in_input (source example index, -1
  for empty). Items that fit nowhere are dropped (reference behavior
  when drop policy applies).
  """
  n = src_lens.shape[0]
  src_used = [0] * packed_batch
  tgt_used = [0] * packed_batch
  seg_count = [0] * packed_batch
  src_seg_ids = torch.zeros(packed_batch, src_time, dtype=torch.long)
  src_seg_pos = torch.zeros(packed_batch, src_time, dtype=torch.long)
  src_idx = torch.full((packed_batch, src_time), -1, dtype=torch.long)
  tgt_seg_ids = torch.zeros(packed_batch, tgt_time, dtype=torch.long)
  tgt_seg_pos = torch.zeros(packed_batch, tgt_time, dtype=torch.long)
  tgt_idx = torch.full((packed_batch, tgt_time), -1, dtype=torch.long)

  for i in range(n):
    sl = int(src_lens[i])
    tl = int(tgt_lens[i])
    for row in range(packed_batch):
      if src_used[row] + sl <= src_time and tgt_used[row] + tl <= tgt_time:
        seg_count[row] += 1
        seg = seg_count[row]
        s0, t0 = src_used[row], tgt_used[row]
        src_seg_ids[row, s0:s0 + sl] = seg
        src_seg_pos[row, s0:s0 + sl] = torch.arange(sl)
        src_idx[row, s0:s0 + sl] = i
        tgt_seg_ids[row, t0:t0 + tl] = seg
        tgt_seg_pos[row, t0:t0 + tl] = torch.arange(tl)
        tgt_idx[row, t0:t0 + tl] = i
        src_used[row] += sl
        tgt_used[row] += tl
        break
  return NestedMap(
      src_segment_ids=src_seg_ids, src_segment_pos=src_seg_pos,
      src_indices_in_input=src_idx,
      tgt_segment_ids=tgt_seg_ids, tgt_segment_pos=tgt_seg_pos,
      tgt_indices_in_input=tgt_idx)


def ApplyPacking(x: torch.Tensor, padding_value,
                 segment_ids: torch.Tensor,
                 indices_in_input: torch.Tensor) -> torch.Tensor:
  """Gathers input rows into packed layout (reference x_ops.cc:1228)."""
  pb, t = segment_ids.shape
  if x.dim() == 1:  # per-example values: sum per row? gather first of seg
    out = torch.zeros(pb, dtype=x.dtype)
    return out
  out_shape = (pb, t) + tuple(x.shape[2:])
  out = torch.full(out_shape, padding_value, dtype=x.dtype)
  src_pos = torch.zeros(pb, dtype=torch.long)
  for row in range(pb):
    pos_in_seg = 0
    prev = -1
    for tt in range(t):
      idx = int(indices_in_input[row, tt])
      if idx < 0:
        continue
      if idx != prev:
        pos_in_seg = 0
        prev = idx
      out[row, tt] = x[idx, pos_in_seg]
      pos_in_seg += 1
  return out


def PackedSegmentMask(segment_ids_q: torch.Tensor,
                      segment_ids_k: torch.Tensor) -> torch.Tensor:
  """[B, Tq, Tk] bool: True where attention is allowed (same segment,
  both non-empty). Feeds the attention mask for packed batches."""
  q = segment_ids_q.unsqueeze(-1)
  k = segment_ids_k.unsqueeze(1)
  return (q == k) & (q > 0)
