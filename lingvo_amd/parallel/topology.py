"""Process-grid topology for composed parallelism (PP x DP, TP x DP).

The reference composes parallelism through TF device placement and XLA
sharding; here the composition is explicit process groups over RCCL:
ranks form a grid, each axis gets its own communicator, and the same
rank participates in one group per axis (pipeline P2P along the stage
axis, gradient all-reduce along the replica axis).

Layout (num_stages contiguous so pipeline neighbors share an xGMI hop):
    rank = dp_idx * num_stages + stage_idx
"""

from __future__ import annotations

from typing import Optional

import torch.distributed as dist


class PpDpTopology:
  """Pipeline x data-parallel grid.

  Every rank must construct this with identical arguments (new_group is
  collective). Exposes this rank's pipe/dp groups, grid coordinates and
  the GLOBAL ranks of the previous/next pipeline stages.
  """

  def __init__(self, num_stages: int, world: Optional[int] = None,
               rank: Optional[int] = None):
    world = world if world is not None else dist.get_world_size()
    rank = rank if rank is not None else dist.get_rank()
    assert world % num_stages == 0, (world, num_stages)
    self.num_stages = num_stages
    self.dp_degree = world // num_stages
    self.stage_idx = rank % num_stages
    self.dp_idx = rank // num_stages
    # all ranks create ALL groups, in the same order
    pipe_groups = [
        dist.new_group(list(range(d * num_stages, (d + 1) * num_stages)))
        for d in range(self.dp_degree)]
    dp_groups = [dist.new_group(list(range(s, world, num_stages)))
                 for s in range(num_stages)]
    self.pipe_group = pipe_groups[self.dp_idx]
    self.dp_group = dp_groups[self.stage_idx]
    self.prev_rank = rank - 1 if self.stage_idx > 0 else None
    self.next_rank = rank + 1 if self.stage_idx < num_stages - 1 else None
    self.is_first_stage = self.stage_idx == 0
    self.is_last_stage = self.stage_idx == num_stages - 1

  def MakeRunner(self, num_micro_batches: int, device: str = 'cpu'):
    from lingvo_amd.parallel.pipeline import GPipeRunner
    return GPipeRunner(self.stage_idx, self.num_stages,
                       num_micro_batches, group=self.pipe_group,
                       device=device, prev_rank=self.prev_rank,
                       next_rank=self.next_rank)

  def AllReduceStageGrads(self, module) -> None:
    """Average this stage's gradients across its DP replicas (called
    after RunStep, before the stage optimizer)."""
    if self.dp_degree == 1:
      return
    for param in module.parameters():
      if param.grad is not None:
        dist.all_reduce(param.grad, group=self.dp_group)
        param.grad /= self.dp_degree


class TpDpTopology:
  """Tensor x data-parallel grid: TP groups are CONTIGUOUS ranks (same
  node / max xGMI locality, where the per-block all-reduce lives), DP
  groups stride across them.

      rank = dp_idx * tp_degree + tp_idx
  """

  def __init__(self, tp_degree: int, world: Optional[int] = None,
               rank: Optional[int] = None):
    world = world if world is not None else dist.get_world_size()
    rank = rank if rank is not None else dist.get_rank()
    assert world % tp_degree == 0, (world, tp_degree)
    self.tp_degree = tp_degree
    self.dp_degree = world // tp_degree
    self.tp_idx = rank % tp_degree
    self.dp_idx = rank // tp_degree
    tp_groups = [dist.new_group(list(range(d * tp_degree,
                                           (d + 1) * tp_degree)))
                 for d in range(self.dp_degree)]
    dp_groups = [dist.new_group(list(range(t, world, tp_degree)))
                 for t in range(tp_degree)]
    self.tp_group = tp_groups[self.dp_idx]
    self.dp_group = dp_groups[self.tp_idx]

  def MakeGradSync(self, module, **kwargs):
    """DP gradient sync over the replica axis only (TP-sharded weights
    differ per tp rank; their replicas live across the DP axis)."""
    from lingvo_amd.parallel.ddp import GradSync
    return GradSync(module, process_group=self.dp_group, **kwargs)
