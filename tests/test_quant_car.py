"""Quantization utils + car 3D ops tests."""

import math

import pytest
import torch

from lingvo_amd.core import quant_utils
from lingvo_amd.models import car_ops


def test_fake_quant_layer():
  p = quant_utils.QuantizedProjectionLayer.Params().Set(
      name='q', input_dim=8, output_dim=8, random_seed=1,
      qdomain_default=quant_utils.QDomain.Params().Set(bits=8))
  layer = p.Instantiate()
  x = torch.randn(4, 8)
  out = layer.FProp(layer.theta, x)
  assert out.shape == (4, 8)
  # straight-through gradient flows
  x2 = torch.randn(4, 8, requires_grad=True)
  layer.FProp(layer.theta, x2).sum().backward()
  assert layer.w.grad is not None
  # start_step gating: before start, identity
  p2 = p.Copy().Set(name='q2')
  p2.qdomain_default.start_step = 100
  layer2 = p2.Instantiate()
  w = layer2.theta.w
  assert torch.equal(layer2.QWeight(w), w)


def test_pairwise_iou_3d():
  a = torch.tensor([[0., 0., 0., 2., 2., 2., 0.]])
  b = torch.tensor([[0., 0., 0., 2., 2., 2., 0.],      # identical
                    [1., 0., 0., 2., 2., 2., 0.],      # half x overlap
                    [10., 0., 0., 2., 2., 2., 0.],     # disjoint
                    [0., 0., 0., 2., 2., 2., math.pi / 2]])  # rotated 90
  iou = car_ops.PairwiseIou3D(a, b)
  assert abs(iou[0, 0] - 1.0) < 1e-4
  assert abs(iou[0, 1] - (4.0 / 12.0)) < 1e-3  # inter 1x2x2, union 12
  assert iou[0, 2] == 0.0
  assert abs(iou[0, 3] - 1.0) < 1e-3  # square rotated 90 == itself


def test_nms_3d():
  boxes = torch.tensor([[0., 0., 0., 2., 2., 2., 0.],
                        [0.1, 0., 0., 2., 2., 2., 0.],
                        [5., 0., 0., 2., 2., 2., 0.]])
  scores = torch.tensor([0.9, 0.8, 0.7])
  keep = car_ops.NonMaxSuppression3D(boxes, scores, iou_threshold=0.5)
  assert keep.tolist() == [0, 2]


def test_average_precision_3d():
  gt = torch.tensor([[0., 0., 0., 2., 2., 2., 0.]])
  pred = torch.tensor([[0., 0., 0., 2., 2., 2., 0.],
                       [5., 5., 0., 2., 2., 2., 0.]])
  scores = torch.tensor([0.9, 0.8])
  ap = car_ops.AveragePrecision3D(gt, pred, scores)
  assert ap > 0.9  # perfect first detection


def test_furthest_point_sampling():
  pts = torch.tensor([[0., 0., 0.], [0.1, 0., 0.], [10., 0., 0.],
                      [0., 10., 0.]])
  idx = car_ops.SamplePoints(pts, 3, seed=1)
  assert len(set(idx.tolist())) == 3
  # the far points should be picked
  assert 2 in idx.tolist() and 3 in idx.tolist()


def test_pillars_model_train_and_decode():
  from lingvo_amd.core import registry
  p = registry.GetParams('car.kitti.StarNetPillars', 'Train')
  p.task.random_seed = 6
  p.task.grid_size = 32
  p.input.Set(batch_size=2, num_points=512)
  task = p.Instantiate().GetTask()
  losses = []
  for _ in range(3):
    m = task.TrainStep(task.GetInputBatch())
    losses.append(float(m['loss'][0]))
  assert all(l == l for l in losses)
  task.eval()
  out = task.Decode(task.GetInputBatch())
  assert out.ap.shape == (2,)
  dm = task.CreateDecoderMetrics()
  task.PostProcessDecodeOut(out, dm)
  assert 0.0 <= dm.ap3d.value <= 1.0


def test_fake_quant_schedule_and_asym_domain():
  import torch
  from lingvo_amd.core import quant_utils
  sched = quant_utils.FakeQuantizationSchedule.Params().Set(
      name='s', clip_start_step=10, clip_end_step=20,
      quant_start_step=15, start_cap=8.0, end_cap=1.0).Instantiate()
  assert sched.CurrentCap(0) == 8.0
  assert sched.CurrentCap(25) == 1.0
  assert 1.0 < sched.CurrentCap(15) < 8.0
  assert not sched.ShouldQuantize(14) and sched.ShouldQuantize(15)

  dom = quant_utils.PassiveAsymQDomain.Params().Set(
      name='d', bits=8, decay=0.0).Instantiate()
  dom.train()
  dom.SetStep(1)
  x = torch.linspace(-1.0, 3.0, 101)
  q = dom.QuantizeNamedTensor('act', x)
  # asymmetric range covers [-1, 3]; quantization error <= scale/2
  scale = 4.0 / 255
  assert (q - x).abs().max() <= scale * 0.51 + 1e-6
  # straight-through grad
  x2 = x.clone().requires_grad_(True)
  dom.QuantizeNamedTensor('act', x2).sum().backward()
  assert torch.allclose(x2.grad, torch.ones_like(x2))
  assert 'act' in dom.state_dict_ranges()


def test_projection_layer_qdomain():
  import torch
  from lingvo_amd.core import quant_utils
  from lingvo_amd.layers import layers as lingvo_layers
  p = lingvo_layers.ProjectionLayer.Params().Set(
      name='p', input_dim=8, output_dim=8, has_bias=True,
      random_seed=3,
      qdomain_tpl=quant_utils.QDomain.Params().Set(bits=8, decay=0.0))
  layer = p.Instantiate()
  layer.train()
  x = torch.randn(4, 8)
  out_q = layer.FProp(layer.theta, x)
  ref = lingvo_layers.ProjectionLayer.Params().Set(
      name='p', input_dim=8, output_dim=8, has_bias=True,
      random_seed=3).Instantiate()
  out_f = ref.FProp(ref.theta, x)
  # quantized output close but not identical to the fp path
  assert (out_q - out_f).abs().max() < 0.2
  assert (out_q - out_f).abs().max() > 0
  # straight-through grads flow
  (out_q.sum()).backward()
  assert layer.w.grad is not None


def test_starnet_train_and_decode():
  import torch
  from lingvo_amd.core import registry
  model_p = registry.GetParams('car.kitti.StarNet', 'Train')
  model_p.task.random_seed = 4
  model_p.task.Set(num_centers=16, num_neighbors=16, feat_dim=16)
  model_p.input.Set(batch_size=2, num_points=256)
  model = model_p.Instantiate()
  task = model.GetTask()
  batch = task.GetInputBatch()
  m = task.TrainStep(batch)
  assert torch.isfinite(m['loss'][0])
  out = task.Decode(batch)
  assert len(out.boxes) == 2
  for bx, sc in zip(out.boxes, out.scores):
    assert bx.shape[1] == 7 and bx.shape[0] == sc.shape[0]


def test_weight_quant_uses_own_scale():
  import torch
  from lingvo_amd.core import quant_utils
  dom = quant_utils.QDomain.Params().Set(
      name='d', bits=8, decay=0.0).Instantiate()
  dom.train()
  dom.SetStep(1)
  # calibrate activations at a LARGE range
  dom.QuantizeTensor(torch.linspace(-100, 100, 11))
  # a small weight must still quantize finely (own-max scale), not to
  # the ~0.8-wide bins the activation range would imply
  w = torch.linspace(-0.1, 0.1, 101)
  qw = dom.QuantizeWeight(w)
  assert (qw - w).abs().max() < 0.1 / 127 + 1e-6


def test_scheduled_clip_qdomain():
  from lingvo_amd.core import quant_utils
  p = quant_utils.SymmetricScheduledClipQDomain.Params().Set(
      name='qd', bits=8)
  p.cc_schedule.Set(clip_start_step=10, clip_end_step=20,
                    quant_start_step=30, start_cap=8.0, end_cap=1.0)
  qd = p.Instantiate()
  x = torch.linspace(-4, 4, 64)
  qd.SetStep(0)
  y0 = qd.QuantizeTensor(x)  # pre-clip ramp: cap=8, no quant
  assert torch.allclose(y0, x)
  qd.SetStep(20)
  y1 = qd.QuantizeTensor(x)  # cap ramped to 1.0
  assert float(y1.max()) <= 1.0 + 1e-6
  qd.SetStep(40)
  y2 = qd.QuantizeTensor(x)  # quantized on a 1.0 cap: 8-bit grid
  grid = torch.unique(y2)
  assert len(grid) <= 255
  assert float(y2.max()) <= 1.0 + 1e-6


def test_int8_weight_export_roundtrip():
  from lingvo_amd.core import quant_utils
  lin = torch.nn.Linear(16, 8)
  packed = quant_utils.MaterializeInt8Weights(lin)
  assert set(packed) == {'weight', 'bias'}
  q, scale = packed['weight']
  assert q.dtype == torch.int8
  back = quant_utils.DequantizeInt8(q, scale)
  rel = (back - lin.weight.detach()).abs().max() / \
      lin.weight.detach().abs().max()
  assert rel < 0.02


def test_point_to_grid():
  from lingvo_amd.models import car_ops
  pts = torch.tensor([
      [0.5, 0.5, 0.5, 1.0], [0.6, 0.4, 0.5, 2.0], [0.7, 0.5, 0.4, 3.0],
      [1.5, 0.5, 0.5, 4.0], [9.0, 9.0, 9.0, 5.0]])  # last out of range
  out, centers, counts = car_ops.PointToGrid(
      pts, num_points_per_cell=2, x_intervals=2, y_intervals=1,
      z_intervals=1, x_range=(0, 2), y_range=(0, 1), z_range=(0, 1))
  assert out.shape == (2, 1, 1, 2, 4)
  assert centers.shape == (2, 1, 1, 3)
  assert counts[0, 0, 0] == 2  # capacity-capped (3 candidates)
  assert counts[1, 0, 0] == 1
  # cell centers correct
  assert torch.allclose(centers[0, 0, 0], torch.tensor([0.5, 0.5, 0.5]))
  # padding slot of cell 1 has center xyz + zero feature
  assert float(out[1, 0, 0, 1, 3]) == 0.0
  assert torch.allclose(out[1, 0, 0, 1, :3],
                        torch.tensor([1.5, 0.5, 0.5]))


def test_ball_query():
  from lingvo_amd.models import car_ops
  pts = torch.tensor([[0., 0, 0], [0.1, 0, 0], [5, 5, 5], [0, 0.2, 0]])
  centers = torch.tensor([[0., 0, 0], [5, 5, 5]])
  idx = car_ops.BallQuery(pts, centers, radius=0.5, num_neighbors=3)
  assert set(idx[0].tolist()) <= {0, 1, 3}
  assert idx[1, 0] == 2 and idx[1, 1] == 2  # padding repeats first


def test_milan_score_functions_and_id_masking():
  from lingvo_amd.models import milan
  p = milan.DualEncoder.Params().Set(
      name='de', joint_dim=32, image_channels=[8], text_dim=32,
      text_layers=1, vocab_size=100, score_function='bilinear',
      label_smoothing=0.1, id_feature='ids', random_seed=3)
  p.input = milan.SyntheticImageTextInput.Params().Set(
      batch_size=4, image_size=16, text_len=6, vocab_size=100)
  task = p.Instantiate()
  batch = task.GetInputBatch()
  batch.ids = torch.tensor([0, 0, 1, 2])  # examples 0,1 share an id
  m = task.TrainStep(batch)
  assert torch.isfinite(m['loss'][0])
  assert hasattr(task, 'score_w')
  # id masking: duplicate pair contributes -inf logits, loss finite
  preds = task.ComputePredictions(task.theta, batch)
  mm, _ = task.ComputeLoss(task.theta, preds, batch)
  assert torch.isfinite(mm.loss[0])


def test_farthest_point_sampler():
  import torch
  from lingvo_amd.models import car_ops
  torch.manual_seed(0)
  pts = torch.randn(2, 32, 3)
  pad = torch.zeros(2, 32)
  pad[1, 16:] = 1.0  # second row: only first 16 points are real
  sampled, closest = car_ops.FarthestPointSampler(pts, pad, 8,
                                                  random_seed=3)
  assert sampled.shape == (2, 8) and closest.shape == (2, 32)
  # No duplicates among sampled points per row.
  for r in range(2):
    assert len(set(sampled[r].tolist())) == 8
  # Padded points are never sampled.
  assert (sampled[1] < 16).all()
  # closest_idx maps each point to the nearest sampled point.
  for r in range(2):
    d = (pts[r, :, None, :] - pts[r, sampled[r]][None]).pow(2).sum(-1)
    want = d.argmin(dim=1)
    real = pad[r] < 0.5
    assert (closest[r][real] == want[real]).all()
  # Seeded points come first, in order.
  sampled2, _ = car_ops.FarthestPointSampler(pts, pad, 8,
                                             num_seeded_points=3)
  assert sampled2[:, :3].tolist() == [[0, 1, 2], [0, 1, 2]]
