"""Registered Waymo Open Dataset car params (reference
lingvo/tasks/car/params waymo configs, README.md:312-325): same
pillars detector family as KITTI at the larger Waymo scene extent /
grid resolution. Synthetic point clouds stand in for the dataset (no
network access)."""

from __future__ import annotations

from lingvo_amd.core import registry
from lingvo_amd.models.params.car.kitti import StarNetPillars


@registry.RegisterSingleTaskModel
class WaymoPillars(StarNetPillars):
  """Pillars on Waymo-scale scenes: 75 m extent, finer BEV grid."""

  def Train(self):
    return super().Train().Set(num_points=4096)

  def Task(self):
    p = super().Task()
    p.name = 'waymo_pillars'
    p.grid_size = 96
    p.scene_extent = 75.0
    p.backbone_channels = [64, 128, 256]
    return p
