"""Point-cloud 3D detection ops (reference lingvo/tasks/car/ops:
PairwiseIou3D / NonMaxSuppression3D / AveragePrecision3D / SamplePoints
at car_ops.cc:22-189, nms_3d_op.cc, sampling_ops.cc).

Torch implementations of the core geometry (axis-aligned + yaw-rotated
BEV overlap via polygon clipping); furthest-point sampling for
SamplePoints. The car task model zoo itself is round-2 scope; these ops
cover the reusable compute surface.
"""

from __future__ import annotations

import math
from typing import Tuple

import torch


def _BoxCorners2D(boxes: torch.Tensor) -> torch.Tensor:
  """boxes [N, 5] (cx, cy, dx, dy, yaw) -> corners [N, 4, 2]."""
  cx, cy, dx, dy, yaw = boxes.unbind(-1)
  cos, sin = torch.cos(yaw), torch.sin(yaw)
  hx, hy = dx / 2, dy / 2
  base = torch.stack([
      torch.stack([hx, hy], -1), torch.stack([-hx, hy], -1),
      torch.stack([-hx, -hy], -1), torch.stack([hx, -hy], -1)
  ], dim=1)  # [N, 4, 2]
  rot = torch.stack([torch.stack([cos, -sin], -1),
                     torch.stack([sin, cos], -1)], dim=1)  # [N, 2, 2]
  return torch.einsum('nij,nkj->nki', rot, base) + \
      torch.stack([cx, cy], -1)[:, None, :]


def _PolygonArea(poly) -> float:
  n = len(poly)
  if n < 3:
    return 0.0
  s = 0.0
  for i in range(n):
    x1, y1 = poly[i]
    x2, y2 = poly[(i + 1) % n]
    s += x1 * y2 - x2 * y1
  return abs(s) / 2.0


def _ClipPolygon(subject, clip):
  """Sutherland-Hodgman clipping (convex clip polygon)."""
  def inside(pt, a, b):
    return (b[0] - a[0]) * (pt[1] - a[1]) - (b[1] - a[1]) * \
        (pt[0] - a[0]) >= -1e-9

  def intersect(p1, p2, a, b):
    dx1, dy1 = p2[0] - p1[0], p2[1] - p1[1]
    dx2, dy2 = b[0] - a[0], b[1] - a[1]
    denom = dx1 * dy2 - dy1 * dx2
    if abs(denom) < 1e-12:
      return p2
    t = ((a[0] - p1[0]) * dy2 - (a[1] - p1[1]) * dx2) / denom
    return (p1[0] + t * dx1, p1[1] + t * dy1)

  output = list(subject)
  for i in range(len(clip)):
    a, b = clip[i], clip[(i + 1) % len(clip)]
    input_list, output = output, []
    if not input_list:
      break
    s = input_list[-1]
    for e in input_list:
      if inside(e, a, b):
        if not inside(s, a, b):
          output.append(intersect(s, e, a, b))
        output.append(e)
      elif inside(s, a, b):
        output.append(intersect(s, e, a, b))
      s = e
  return output


def PairwiseIou3D(boxes_a: torch.Tensor,
                  boxes_b: torch.Tensor) -> torch.Tensor:
  """boxes [N, 7] (cx, cy, cz, dx, dy, dz, yaw) -> IoU [N, M]
  (reference car_ops.cc PairwiseIou3D)."""
  n, m = boxes_a.shape[0], boxes_b.shape[0]
  out = torch.zeros(n, m)
  ca = _BoxCorners2D(boxes_a[:, [0, 1, 3, 4, 6]]).tolist()
  cb = _BoxCorners2D(boxes_b[:, [0, 1, 3, 4, 6]]).tolist()
  for i in range(n):
    za0 = float(boxes_a[i, 2] - boxes_a[i, 5] / 2)
    za1 = float(boxes_a[i, 2] + boxes_a[i, 5] / 2)
    va = float(boxes_a[i, 3] * boxes_a[i, 4] * boxes_a[i, 5])
    for j in range(m):
      zb0 = float(boxes_b[j, 2] - boxes_b[j, 5] / 2)
      zb1 = float(boxes_b[j, 2] + boxes_b[j, 5] / 2)
      zo = max(0.0, min(za1, zb1) - max(za0, zb0))
      if zo <= 0:
        continue
      inter2d = _PolygonArea(_ClipPolygon(ca[i], cb[j]))
      inter = inter2d * zo
      vb = float(boxes_b[j, 3] * boxes_b[j, 4] * boxes_b[j, 5])
      union = va + vb - inter
      if union > 0:
        out[i, j] = inter / union
  return out


def NonMaxSuppression3D(boxes: torch.Tensor, scores: torch.Tensor,
                        iou_threshold: float = 0.5,
                        max_boxes: int = 100) -> torch.Tensor:
  """Returns kept indices, best-score first (reference nms_3d_op.cc)."""
  order = scores.argsort(descending=True)
  keep = []
  iou = PairwiseIou3D(boxes, boxes)
  suppressed = torch.zeros(boxes.shape[0], dtype=torch.bool)
  for idx in order.tolist():
    if suppressed[idx]:
      continue
    keep.append(idx)
    if len(keep) >= max_boxes:
      break
    suppressed |= iou[idx] > iou_threshold
  return torch.tensor(keep, dtype=torch.long)


def AveragePrecision3D(gt_boxes: torch.Tensor, pred_boxes: torch.Tensor,
                       pred_scores: torch.Tensor,
                       iou_threshold: float = 0.7) -> float:
  """11-point interpolated AP (reference average_precision_3d_op.cc)."""
  if pred_boxes.shape[0] == 0:
    return 0.0
  order = pred_scores.argsort(descending=True)
  iou = PairwiseIou3D(pred_boxes, gt_boxes)
  matched = torch.zeros(gt_boxes.shape[0], dtype=torch.bool)
  tps = []
  for idx in order.tolist():
    best_j, best = -1, iou_threshold
    for j in range(gt_boxes.shape[0]):
      if not matched[j] and iou[idx, j] >= best:
        best, best_j = iou[idx, j], j
    if best_j >= 0:
      matched[best_j] = True
      tps.append(1.0)
    else:
      tps.append(0.0)
  tps_t = torch.tensor(tps)
  cum_tp = tps_t.cumsum(0)
  precision = cum_tp / torch.arange(1, len(tps) + 1)
  recall = cum_tp / max(1, gt_boxes.shape[0])
  ap = 0.0
  for r in [i / 10 for i in range(11)]:
    mask = recall >= r
    ap += (precision[mask].max().item() if mask.any() else 0.0) / 11
  return ap


def SamplePoints(points: torch.Tensor, num_samples: int,
                 seed: int = 0) -> torch.Tensor:
  """Furthest-point sampling -> indices [num_samples]
  (reference sampling_ops.cc)."""
  n = points.shape[0]
  g = torch.Generator().manual_seed(seed)
  first = int(torch.randint(0, n, (1,), generator=g))
  chosen = [first]
  dists = (points - points[first]).pow(2).sum(-1)
  for _ in range(min(num_samples, n) - 1):
    nxt = int(dists.argmax())
    chosen.append(nxt)
    dists = torch.minimum(dists,
                          (points - points[nxt]).pow(2).sum(-1))
  return torch.tensor(chosen, dtype=torch.long)


def PointToGrid(points: torch.Tensor, num_points_per_cell: int,
                x_intervals: int, y_intervals: int, z_intervals: int,
                x_range, y_range, z_range, seed: int = 0):
  """Bins points into an equally spaced 3-D grid (reference
  tasks/car/ops/car_ops.cc:38 PointToGrid / point_grid_op.cc).

  points [n, d] (first 3 dims = xyz). Returns (output_points
  [gx, gy, gz, P, d], grid_centers [gx, gy, gz, 3], num_points
  [gx, gy, gz]). Cells beyond capacity drop random points (shuffled);
  short cells are padded with the cell center on xyz and zeros on the
  remaining features.
  """
  n, d = points.shape
  gx, gy, gz = x_intervals, y_intervals, z_intervals
  device = points.device
  sizes = torch.tensor(
      [(x_range[1] - x_range[0]) / gx, (y_range[1] - y_range[0]) / gy,
       (z_range[1] - z_range[0]) / gz], device=device)
  lo = torch.tensor([x_range[0], y_range[0], z_range[0]], device=device)
  cell = ((points[:, :3] - lo) / sizes).floor().long()
  in_range = ((cell >= 0) & (cell < torch.tensor([gx, gy, gz],
                                                 device=device))).all(-1)
  flat = (cell[:, 0] * gy + cell[:, 1]) * gz + cell[:, 2]
  flat = torch.where(in_range, flat, torch.full_like(flat, -1))

  # Grid centers.
  ix = torch.arange(gx, device=device)
  iy = torch.arange(gy, device=device)
  iz = torch.arange(gz, device=device)
  cx, cy, cz = torch.meshgrid(ix, iy, iz, indexing='ij')
  centers = torch.stack(
      [(cx + 0.5) * sizes[0] + lo[0], (cy + 0.5) * sizes[1] + lo[1],
       (cz + 0.5) * sizes[2] + lo[2]], dim=-1).float()

  p = num_points_per_cell
  out = torch.zeros(gx * gy * gz, p, d, device=device)
  out[:, :, :3] = centers.reshape(-1, 1, 3)  # padding = cell center
  counts = torch.zeros(gx * gy * gz, dtype=torch.int32, device=device)
  g = torch.Generator(device='cpu').manual_seed(seed)
  order = torch.randperm(n, generator=g).to(device)  # shuffle per doc
  slots = {}
  for i in order.tolist():
    c = int(flat[i])
    if c < 0:
      continue
    k = slots.get(c, 0)
    if k < p:
      out[c, k] = points[i]
      slots[c] = k + 1
  for c, k in slots.items():
    counts[c] = k
  return (out.reshape(gx, gy, gz, p, d), centers,
          counts.reshape(gx, gy, gz))


def BallQuery(points: torch.Tensor, centers: torch.Tensor, radius: float,
              num_neighbors: int) -> torch.Tensor:
  """Fixed-radius neighborhood gather (reference ps_utils.cc
  neighborhood sampling): for each center, up to `num_neighbors` point
  indices within `radius` (first index repeated as padding).
  points [n, 3], centers [m, 3] -> [m, num_neighbors] long."""
  d2 = (centers[:, None, :] - points[None, :, :]).pow(2).sum(-1)
  within = d2 <= radius * radius
  idx = torch.zeros(centers.shape[0], num_neighbors, dtype=torch.long,
                    device=points.device)
  for i in range(centers.shape[0]):
    cand = within[i].nonzero(as_tuple=True)[0]
    if cand.numel() == 0:
      continue
    take = cand[:num_neighbors]
    idx[i, :take.numel()] = take
    if take.numel() < num_neighbors:
      idx[i, take.numel():] = take[0]
  return idx


def FarthestPointSampler(points: torch.Tensor, padding: torch.Tensor,
                         num_sampled_points: int,
                         precomputed_squared_distance=None,
                         num_seeded_points: int = 0, random_seed=None):
  """Batched farthest-point sampling (reference car_lib.py:244).

  points [N, P, dims]; padding [N, P] (1 = padded). Returns
  (sampled_idx [N, S] long, closest_idx [N, P] long) where closest_idx
  maps every input point to its nearest sampled point (PCNN pooling).
  The first num_seeded_points are taken as-is (assumed unpadded).
  """
  n, p1, _ = points.shape
  s_count = min(num_sampled_points, p1)
  g = torch.Generator().manual_seed(
      random_seed if random_seed is not None else 0)
  big = torch.finfo(torch.float32).max

  def pair_dist(idx):
    # squared distance from every point to the sampled point idx [N].
    if precomputed_squared_distance is not None:
      return precomputed_squared_distance[
          torch.arange(n), idx]  # [N, P]
    sel = points[torch.arange(n), idx].unsqueeze(1)  # [N, 1, dims]
    return (points.float() - sel.float()).pow(2).sum(-1)

  pad_mask = padding > 0.5
  sampled = torch.zeros(n, s_count, dtype=torch.long)
  closest = torch.zeros(n, p1, dtype=torch.long)
  min_dist = torch.full((n, p1), big)

  for step in range(s_count):
    if step < num_seeded_points:
      idx = torch.full((n,), step, dtype=torch.long)
    elif step == 0:
      # Random unpadded start per batch row.
      r = torch.rand(n, p1, generator=g).masked_fill(pad_mask, -1.0)
      idx = r.argmax(dim=1)
    else:
      scores = min_dist.masked_fill(pad_mask, -big)
      idx = scores.argmax(dim=1)
    sampled[:, step] = idx
    d = pair_dist(idx)
    improved = d < min_dist
    closest = torch.where(improved,
                          torch.full_like(closest, step), closest)
    min_dist = torch.minimum(min_dist, d)
  return sampled, closest
