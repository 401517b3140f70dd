"""Batch-size scaling helpers (reference lingvo/core/batch_utils.py:21).

The reference scales between per-infeed / per-split / global batch
sizes using TPU-host cluster facts. In the MI355X one-process-per-GPU
runtime the "infeed" is a rank's input pipeline, so the scale factor is
the torch.distributed world size.
"""

from __future__ import annotations


def _world_size() -> int:
  import torch.distributed as dist
  if dist.is_available() and dist.is_initialized():
    return dist.get_world_size()
  return 1


def scale_infeed_to_global(infeed_batch_size: int,
                           use_per_host_infeed: bool = True) -> int:
  """Per-rank batch size -> whole-job batch size."""
  if use_per_host_infeed:
    return infeed_batch_size * _world_size()
  return infeed_batch_size


def scale_global_to_infeed(global_batch_size: int,
                           use_per_host_infeed: bool = True) -> int:
  """Whole-job batch size -> per-rank batch size."""
  if use_per_host_infeed:
    ws = _world_size()
    assert global_batch_size % ws == 0, (global_batch_size, ws)
    return global_batch_size // ws
  return global_batch_size


def scale_split_to_infeed(split_batch_size: int,
                          use_per_host_infeed: bool = True) -> int:
  """Per-model-split batch size -> per-rank infeed size (splits == 1
  in the one-process-per-GPU runtime)."""
  del use_per_host_infeed
  return split_batch_size


def scale_global_to_worker(global_batch_size: int) -> int:
  """Whole-job batch size -> per-worker (rank) batch size."""
  return scale_global_to_infeed(global_batch_size, True)
