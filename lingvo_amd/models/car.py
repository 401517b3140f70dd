"""Car task: point-cloud 3D detection (reference lingvo/tasks/car:
pillars/starnet models, README.md:312-325; C++ ops in tasks/car/ops
covered by models/car_ops.py).

PointPillars-style model: points are binned into a BEV grid
(PointToGrid, car_ops.cc), a per-pillar PointNet featurizes them, a 2D
conv backbone runs over the grid, and an anchor-free head predicts
per-cell occupancy + box residuals. Synthetic scenes stand in for
KITTI/Waymo (no network for the datasets).
"""

from __future__ import annotations

import math
from typing import Optional, Tuple

import torch
import torch.nn.functional as F

from lingvo_amd.core import py_utils
from lingvo_amd.core.base_input_generator import BaseInputGenerator
from lingvo_amd.core.base_model import BaseTask
from lingvo_amd.core.nested_map import NestedMap
from lingvo_amd.layers import layers as lingvo_layers
from lingvo_amd.models import car_ops


class SyntheticPointCloudInput(BaseInputGenerator):
  """Scenes with a handful of boxes and points sampled on/off objects."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.batch_size = 4
    p.Define('num_points', 2048, 'Points per scene.')
    p.Define('max_boxes', 8, 'Max ground-truth boxes.')
    p.Define('scene_extent', 40.0, 'Half-extent of the square scene (m).')
    return p

  def _InputBatch(self) -> NestedMap:
    p = self.p
    g = torch.Generator().manual_seed(3100 + self._batch_count)
    b, n = p.batch_size, p.num_points
    ext = p.scene_extent
    points = (torch.rand(b, n, 3, generator=g) - 0.5) * 2 * ext
    points[..., 2] = torch.rand(b, n, generator=g) * 3.0 - 1.0
    boxes = torch.zeros(b, p.max_boxes, 7)
    nboxes = torch.randint(2, p.max_boxes + 1, (b,), generator=g)
    for i in range(b):
      k = int(nboxes[i])
      ctr = (torch.rand(k, 2, generator=g) - 0.5) * 1.6 * ext
      boxes[i, :k, 0:2] = ctr
      boxes[i, :k, 2] = 0.5
      boxes[i, :k, 3:6] = torch.tensor([4.0, 1.8, 1.6])
      boxes[i, :k, 6] = torch.rand(k, generator=g) * math.pi
      # Drop ~30% of points onto the boxes so there is signal.
      per_box = n // (4 * k)
      idx = 0
      for j in range(k):
        sl = slice(idx, idx + per_box)
        local = (torch.rand(per_box, 3, generator=g) - 0.5)
        points[i, sl, 0] = ctr[j, 0] + local[:, 0] * 4.0
        points[i, sl, 1] = ctr[j, 1] + local[:, 1] * 1.8
        points[i, sl, 2] = 0.5 + local[:, 2] * 1.6
        idx += per_box
    return NestedMap(points=points, gt_boxes=boxes,
                     num_boxes=nboxes)


class PointsToGridFeaturizer(torch.nn.Module):
  """PointToGrid + per-pillar PointNet (reference car_ops.cc PointToGrid
  + pillars featurizer), implemented with scatter ops."""

  def __init__(self, grid: int, extent: float, feat_dim: int):
    super().__init__()
    self.grid = grid
    self.extent = extent
    self.feat = feat_dim

  def forward(self, points: torch.Tensor, mlp) -> torch.Tensor:
    """points [B, N, 3] -> grid features [B, feat, G, G]."""
    b, n, _ = points.shape
    g = self.grid
    cell = 2 * self.extent / g
    ij = ((points[..., :2] + self.extent) / cell).long().clamp(0, g - 1)
    flat = ij[..., 0] * g + ij[..., 1]  # [B, N]
    # Per-point features: xyz + offset within cell.
    centers = (ij.float() + 0.5) * cell - self.extent
    feats = torch.cat([points, points[..., :2] - centers], dim=-1)
    feats = mlp(feats)  # [B, N, F]
    out = feats.new_zeros(b, g * g, feats.shape[-1])
    out.scatter_reduce_(1, flat.unsqueeze(-1).expand_as(feats), feats,
                        reduce='amax', include_self=False)
    out = out.nan_to_num(0.0).clamp_min(-1e4)
    return out.reshape(b, g, g, -1).permute(0, 3, 1, 2)


class PillarsModel(BaseTask):
  """Anchor-free BEV detector (StarNet/pillars capability surface)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('grid_size', 64, 'BEV grid G.')
    p.Define('scene_extent', 40.0, 'Scene half-extent (m).')
    p.Define('point_feat_dim', 64, 'PointNet feature dim.')
    p.Define('backbone_channels', [64, 128], 'Conv backbone channels.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    self.CreateChild('point_mlp', lingvo_layers.FeedForwardNet.Params()
                     .Set(input_dim=5,
                          hidden_layer_dims=[32, p.point_feat_dim],
                          activation='RELU'))
    self._featurizer = PointsToGridFeaturizer(
        p.grid_size, p.scene_extent, p.point_feat_dim)
    convs = []
    cin = p.point_feat_dim
    for i, ch in enumerate(p.backbone_channels):
      convs.append(lingvo_layers.Conv2DLayer.Params().Set(
          name=f'bb{i}', filter_shape=(3, 3, cin, ch),
          filter_stride=(1, 1), activation='RELU'))
      cin = ch
    self.CreateChildren('backbone', convs)
    # Head: occupancy logit + (dx, dy, z, logl, logw, logh, sin, cos).
    self.CreateChild('head', lingvo_layers.ProjectionLayer.Params().Set(
        input_dim=cin, output_dim=9, has_bias=True))

  def _Forward(self, theta, points):
    feats = self._featurizer(
        points.to(self.fprop_dtype),
        lambda x: self.point_mlp.FProp(theta.point_mlp, x))
    x = feats.permute(0, 2, 3, 1)  # NHWC for Conv2DLayer
    for i, conv in enumerate(self.backbone):
      x = conv.FProp(theta.backbone[i], x)
    return self.head.FProp(theta.head, x)  # [B, G, G, 9]

  def _CellTargets(self, gt_boxes, num_boxes, device):
    p = self.p
    g = p.grid_size
    cell = 2 * p.scene_extent / g
    b = gt_boxes.shape[0]
    occ = torch.zeros(b, g, g, device=device)
    reg = torch.zeros(b, g, g, 8, device=device)
    for i in range(b):
      for j in range(int(num_boxes[i])):
        box = gt_boxes[i, j]
        gx = int((box[0] + p.scene_extent) / cell)
        gy = int((box[1] + p.scene_extent) / cell)
        if 0 <= gx < g and 0 <= gy < g:
          occ[i, gx, gy] = 1.0
          cx = (gx + 0.5) * cell - p.scene_extent
          cy = (gy + 0.5) * cell - p.scene_extent
          reg[i, gx, gy] = torch.tensor(
              [float(box[0]) - cx, float(box[1]) - cy, float(box[2]),
               math.log(float(box[3])), math.log(float(box[4])),
               math.log(float(box[5])), math.sin(float(box[6])),
               math.cos(float(box[6]))], device=device)
    return occ, reg

  def ComputePredictions(self, theta, input_batch):
    out = self._Forward(theta, input_batch.points)
    return NestedMap(head_out=out)

  def ComputeLoss(self, theta, predictions, input_batch):
    out = predictions.head_out.float()
    occ_logit = out[..., 0]
    reg_pred = out[..., 1:]
    occ, reg = self._CellTargets(input_batch.gt_boxes,
                                 input_batch.num_boxes, out.device)
    occ_loss = F.binary_cross_entropy_with_logits(occ_logit, occ)
    mask = occ.unsqueeze(-1)
    denom = mask.sum().clamp_min(1.0)
    reg_loss = (F.smooth_l1_loss(reg_pred * mask, reg * mask,
                                 reduction='sum') / denom)
    loss = occ_loss + reg_loss
    w = torch.tensor(float(out.shape[0]))
    metrics = NestedMap(loss=(loss, w),
                        occ_loss=(occ_loss.detach(), w),
                        reg_loss=(reg_loss.detach(), w),
                        num_samples_in_batch=(w, torch.ones(())))
    return metrics, NestedMap()

  def Decode(self, input_batch) -> NestedMap:
    """Head -> boxes -> NMS3D -> AP3D against ground truth."""
    p = self.p
    with torch.no_grad():
      out = self._Forward(self.theta, input_batch.points).float()
    g = p.grid_size
    cell = 2 * p.scene_extent / g
    scores_all, boxes_all, aps = [], [], []
    for i in range(out.shape[0]):
      probs = torch.sigmoid(out[i, ..., 0]).reshape(-1)
      topk = probs.topk(32)
      cells = topk.indices
      gx = (cells // g).float()
      gy = (cells % g).float()
      r = out[i].reshape(-1, 9)[cells]
      cx = (gx + 0.5) * cell - p.scene_extent + r[:, 1]
      cy = (gy + 0.5) * cell - p.scene_extent + r[:, 2]
      boxes = torch.stack([
          cx, cy, r[:, 3], r[:, 4].exp().clamp(0.1, 20),
          r[:, 5].exp().clamp(0.1, 20), r[:, 6].exp().clamp(0.1, 20),
          torch.atan2(r[:, 7], r[:, 8])], dim=-1).cpu()
      keep = car_ops.NonMaxSuppression3D(boxes, topk.values.cpu(),
                                         iou_threshold=0.3, max_boxes=16)
      kb = boxes[keep]
      ks = topk.values.cpu()[keep]
      nb = int(input_batch.num_boxes[i])
      ap = car_ops.AveragePrecision3D(
          input_batch.gt_boxes[i, :nb].cpu(), kb, ks, iou_threshold=0.25)
      aps.append(ap)
    return NestedMap(ap=torch.tensor(aps))

  def CreateDecoderMetrics(self) -> NestedMap:
    from lingvo_amd.core import metrics as metrics_lib
    return NestedMap(ap3d=metrics_lib.AverageMetric(),
                     num_samples_in_batch=metrics_lib.AverageMetric())

  def PostProcessDecodeOut(self, decode_out, decode_metrics) -> None:
    for ap in decode_out.ap.tolist():
      decode_metrics.ap3d.Update(ap)
    decode_metrics.num_samples_in_batch.Update(
        float(decode_out.ap.shape[0]))


class StarNetModel(BaseTask):
  """Point-based 3D detector (reference tasks/car StarNet,
  starnet.py: sampled anchor centers + per-cell featurizer): furthest-
  point-sample C centers, gather the K nearest points per center,
  featurize each cell with a shared MLP + max-pool, and predict an
  occupancy logit + box residuals per center. Contrast with
  PillarsModel's dense BEV grid: compute follows the points, not the
  scene area."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('num_centers', 64, 'Sampled anchor centers C.')
    p.Define('num_neighbors', 64, 'Points gathered per center K.')
    p.Define('feat_dim', 64, 'Cell feature dim.')
    p.Define('match_radius', 3.0, 'Center-to-gt match distance (m).')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    self.CreateChild('cell_mlp', lingvo_layers.FeedForwardNet.Params().Set(
        input_dim=3, hidden_layer_dims=[32, p.feat_dim]))
    self.CreateChild('head', lingvo_layers.FeedForwardNet.Params().Set(
        input_dim=p.feat_dim, hidden_layer_dims=[p.feat_dim, 9],
        activation=['RELU', 'NONE']))

  def _SampleCenters(self, points):
    from lingvo_amd.models import car_ops
    b = points.shape[0]
    centers = []
    for i in range(b):
      idx = car_ops.SamplePoints(points[i], self.p.num_centers,
                                 seed=1234 + i)
      centers.append(points[i, idx])
    return torch.stack(centers)  # [B, C, 3]

  def _Forward(self, theta, points):
    p = self.p
    centers = self._SampleCenters(points)              # [B, C, 3]
    d = torch.cdist(centers, points)                   # [B, C, N]
    knn = d.topk(p.num_neighbors, largest=False).indices
    gathered = torch.gather(
        points.unsqueeze(1).expand(-1, centers.shape[1], -1, -1), 2,
        knn.unsqueeze(-1).expand(-1, -1, -1, 3))       # [B, C, K, 3]
    rel = gathered - centers.unsqueeze(2)
    feats = self.cell_mlp.FProp(theta.cell_mlp, rel).max(dim=2).values
    out = self.head.FProp(theta.head, feats)           # [B, C, 9]
    return centers, out

  def _CenterTargets(self, centers, gt_boxes, num_boxes):
    p = self.p
    b, c, _ = centers.shape
    occ = torch.zeros(b, c, device=centers.device)
    reg = torch.zeros(b, c, 8, device=centers.device)
    for i in range(b):
      k = int(num_boxes[i])
      if k == 0:
        continue
      boxes = gt_boxes[i, :k]
      d = torch.cdist(centers[i, :, :2], boxes[:, :2])  # [C, k]
      best = d.argmin(dim=1)
      near = d.gather(1, best.unsqueeze(1)).squeeze(1) < p.match_radius
      for ci in torch.nonzero(near, as_tuple=True)[0].tolist():
        box = boxes[best[ci]]
        occ[i, ci] = 1.0
        reg[i, ci] = torch.stack([
            box[0] - centers[i, ci, 0], box[1] - centers[i, ci, 1],
            box[2], box[3].log(), box[4].log(), box[5].log(),
            box[6].sin(), box[6].cos()])
    return occ, reg

  def ComputePredictions(self, theta, input_batch):
    centers, out = self._Forward(theta, input_batch.points)
    return NestedMap(centers=centers, head_out=out)

  def ComputeLoss(self, theta, predictions, input_batch):
    out = predictions.head_out.float()
    occ_logit = out[..., 0]
    reg_pred = out[..., 1:]
    occ, reg = self._CenterTargets(predictions.centers,
                                   input_batch.gt_boxes,
                                   input_batch.num_boxes)
    occ_loss = F.binary_cross_entropy_with_logits(occ_logit, occ)
    mask = occ.unsqueeze(-1)
    denom = mask.sum().clamp_min(1.0)
    reg_loss = (F.smooth_l1_loss(reg_pred * mask, reg * mask,
                                 reduction='sum') / denom)
    loss = occ_loss + reg_loss
    w = torch.tensor(float(out.shape[0]))
    metrics = NestedMap(loss=(loss, w),
                        occ_loss=(occ_loss.detach(), w),
                        reg_loss=(reg_loss.detach(), w),
                        num_samples_in_batch=(w, torch.ones(())))
    return metrics, NestedMap()

  def Decode(self, input_batch) -> NestedMap:
    from lingvo_amd.models import car_ops
    with torch.no_grad():
      preds = self.ComputePredictions(self.theta, input_batch)
    out = preds.head_out.float()
    scores = torch.sigmoid(out[..., 0])
    boxes_all, scores_all = [], []
    for i in range(out.shape[0]):
      ctr = preds.centers[i]
      r = out[i, :, 1:]
      boxes = torch.stack([
          ctr[:, 0] + r[:, 0], ctr[:, 1] + r[:, 1], r[:, 2],
          r[:, 3].exp(), r[:, 4].exp(), r[:, 5].exp(),
          torch.atan2(r[:, 6], r[:, 7])], dim=1)
      conf = scores[i] > 0.3
      boxes_c, scores_c = boxes[conf], scores[i][conf]
      keep = car_ops.NonMaxSuppression3D(boxes_c, scores_c,
                                         iou_threshold=0.3)
      boxes_all.append(boxes_c[keep])
      scores_all.append(scores_c[keep])
    return NestedMap(boxes=boxes_all, scores=scores_all,
                     gt_boxes=input_batch.gt_boxes,
                     num_boxes=input_batch.num_boxes)
