"""Tests for BaseLayer / py_utils foundations."""

import pytest
import torch

from lingvo_amd.core import py_utils
from lingvo_amd.core.base_layer import BaseLayer
from lingvo_amd.core.nested_map import NestedMap


class Linear(BaseLayer):

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('input_dim', 4, '')
    p.Define('output_dim', 4, '')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    self.CreateVariable('w', py_utils.WeightParams(
        [p.input_dim, p.output_dim], py_utils.WeightInit.Xavier(1.0),
        p.dtype))
    self.CreateVariable('b', py_utils.WeightParams(
        [p.output_dim], py_utils.WeightInit.Constant(0.0), p.dtype))

  def FProp(self, theta, x):
    return torch.matmul(x, theta.w) + theta.b


class TwoLayer(BaseLayer):

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('dims', 4, '')
    return p

  def __init__(self, params):
    super().__init__(params)
    d = self.p.dims
    self.CreateChild('l1', Linear.Params().Set(input_dim=d, output_dim=d))
    self.CreateChildren('reps', [
        Linear.Params().Set(input_dim=d, output_dim=d) for _ in range(2)])

  def FProp(self, theta, x):
    x = self.l1.FProp(theta.l1, x)
    for i, rep in enumerate(self.reps):
      x = rep.FProp(theta.reps[i], x)
    return x


def test_create_variable_and_theta():
  layer = Linear.Params().Set(name='lin', random_seed=7).Instantiate()
  theta = layer.theta
  assert isinstance(theta, NestedMap)
  assert theta.w.shape == (4, 4)
  assert torch.equal(theta.b, torch.zeros(4))
  # named_parameters integration for optimizers
  names = dict(layer.named_parameters())
  assert 'w' in names and 'b' in names


def test_deterministic_init():
  l1 = Linear.Params().Set(name='lin', random_seed=7).Instantiate()
  l2 = Linear.Params().Set(name='lin', random_seed=7).Instantiate()
  l3 = Linear.Params().Set(name='lin', random_seed=8).Instantiate()
  assert torch.equal(l1.theta.w, l2.theta.w)
  assert not torch.equal(l1.theta.w, l3.theta.w)


def test_children_and_nested_theta():
  m = TwoLayer.Params().Set(name='m', random_seed=1).Instantiate()
  theta = m.theta
  assert theta.l1.w.shape == (4, 4)
  assert len(theta.reps) == 2
  x = torch.randn(3, 4)
  y = m.FProp(theta, x)
  assert y.shape == (3, 4)
  # FPropDefaultTheta / forward agree
  assert torch.allclose(m(x), y)


def test_fprop_dtype_cast():
  p = TwoLayer.Params().Set(name='m', random_seed=1)
  p.fprop_dtype = torch.bfloat16
  m = p.Instantiate()
  theta = m.theta
  assert theta.l1.w.dtype == torch.bfloat16
  # master weights stay fp32
  assert m.l1.w.dtype == torch.float32
  # gradient flows to master
  x = torch.randn(3, 4, dtype=torch.bfloat16)
  m.FProp(theta, x).float().sum().backward()
  assert m.l1.w.grad is not None
  assert m.l1.w.grad.dtype == torch.float32


def test_weight_init_variants():
  for method in ('gaussian', 'uniform', 'xavier', 'geo_mean_xavier',
                 'truncated_gaussian', 'uniform_unit_scaling',
                 'gaussian_sqrt_dim', 'truncated_gaussian_sqrt_fanin'):
    spec = py_utils.WeightInit._Spec(method, 1.0)
    g = torch.Generator().manual_seed(3)
    w = py_utils.InitWeight([16, 8], spec, g)
    assert w.shape == (16, 8)
    assert torch.isfinite(w).all()


def test_step_seed_scope_determinism():
  x = torch.ones(1000)
  with py_utils.StepSeedScope(1, 5):
    a = py_utils.DeterministicDropout(x, 0.5)
    b = py_utils.DeterministicDropout(x, 0.5)
  with py_utils.StepSeedScope(1, 5):
    a2 = py_utils.DeterministicDropout(x, 0.5)
    b2 = py_utils.DeterministicDropout(x, 0.5)
  with py_utils.StepSeedScope(1, 6):
    c = py_utils.DeterministicDropout(x, 0.5)
  assert torch.equal(a, a2) and torch.equal(b, b2)
  assert not torch.equal(a, b)  # distinct ops within a step
  assert not torch.equal(a, c)  # distinct steps


def test_padding_utils():
  lengths = torch.tensor([3, 1])
  pad = py_utils.PaddingsFromLengths(lengths, 4)
  assert pad.tolist() == [[0, 0, 0, 1], [0, 1, 1, 1]]
  assert torch.equal(py_utils.LengthsFromPaddings(pad), lengths)
  x = torch.ones(2, 4, 2)
  masked = py_utils.ApplyPadding(pad, x)
  assert masked[1, 2].sum() == 0 and masked[0, 2].sum() == 2
  padded = py_utils.PadSequenceDimension(x, 6)
  assert padded.shape == (2, 6, 2)


def test_accumulator_registration():
  from lingvo_amd.core.base_layer import Accumulator
  layer = Linear.Params().Set(name='l').Instantiate()
  layer.RegisterAccumulator('acc', Accumulator('acc', torch.zeros(2)))
  acc = layer.GetAccumulator('acc')
  acc.SetValue(torch.ones(2))
  assert acc.GetValue().sum() == 2
  acc.Reset()
  assert acc.GetValue().sum() == 0


def test_graph_safe_uniform_statistics():
  """The device-path hash must decorrelate consecutive step seeds (a
  2-round mixer regressed to corr ~ -0.24; pinned here at < 0.02)."""
  import torch
  from lingvo_amd.core import py_utils

  def lshr(x, k):
    return (x >> k) & ((1 << (64 - k)) - 1)

  def device_formula(n, s1, buf):
    # mirror of the cuda branch in GraphSafeUniform
    idx = torch.arange(n, dtype=torch.int64)
    h = (idx * 0x9E3779B97F4A7C15) ^ (buf + s1)
    h = h ^ lshr(h, 33)
    h = h * -0xAE502812AA7333
    h = h ^ lshr(h, 33)
    h = h * -0x3B314601E57A13AD
    h = h ^ lshr(h, 33)
    return (h & 0x7FFFFFFF).float() / float(1 << 31)

  u = device_formula(200_000, 123456789, 987654321)
  u2 = device_formula(200_000, 123456790, 987654321)
  var = float(u.var())
  assert abs(float(u.mean()) - 0.5) < 0.01
  assert abs(var - 1 / 12) < 0.005
  corr = float(((u - 0.5) * (u2 - 0.5)).mean()) / var
  assert abs(corr) < 0.02, corr
  # CPU path: deterministic per (seed, step), distinct across steps
  with py_utils.StepSeedScope(5, 1):
    a = py_utils.GraphSafeUniform((1000,), 'cpu')
  with py_utils.StepSeedScope(5, 1):
    b = py_utils.GraphSafeUniform((1000,), 'cpu')
  with py_utils.StepSeedScope(5, 2):
    c = py_utils.GraphSafeUniform((1000,), 'cpu')
  assert torch.equal(a, b)
  assert not torch.equal(a, c)


def test_weight_init_distributions():
  import math
  import torch
  from lingvo_amd.core import py_utils as pu
  g = torch.Generator().manual_seed(5)
  # xavier: uniform in +-sqrt(6/(fan_in+fan_out))
  w = pu.InitWeight([256, 128], pu.WeightInit.Xavier(1.0), g)
  limit = math.sqrt(6.0 / (256 + 128))
  assert float(w.abs().max()) <= limit + 1e-6
  assert float(w.abs().max()) > 0.8 * limit  # actually fills the range
  # gaussian std
  w2 = pu.InitWeight([512, 512], pu.WeightInit.Gaussian(0.02), g)
  assert abs(float(w2.std()) - 0.02) < 0.002
  # truncated gaussian: bounded at 2 std
  w3 = pu.InitWeight([256, 256],
                     pu.WeightInit.TruncatedGaussian(0.1), g)
  assert float(w3.abs().max()) <= 0.2 + 1e-6
  # fan-in scaling
  w4 = pu.InitWeight([400, 100],
                     pu.WeightInit.TruncatedGaussianSqrtFanIn(1.0), g)
  assert abs(float(w4.std()) - 1 / math.sqrt(400)) < 0.01
  # constant
  w5 = pu.InitWeight([8], pu.WeightInit.Constant(3.0), g)
  assert torch.equal(w5, torch.full([8], 3.0))
