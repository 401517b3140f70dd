"""Per-step RNN-decoder attention family tests."""

import torch

from lingvo_amd.layers import attention_legacy as al


def _setup(cls, **kw):
  kwargs = dict(name='a', source_dim=12, query_dim=10, hidden_dim=8,
                random_seed=5)
  kwargs.update(kw)
  p = cls.Params().Set(**kwargs)
  layer = p.Instantiate()
  g = torch.Generator().manual_seed(3)
  src = torch.randn(2, 7, 12, generator=g)
  pad = torch.zeros(2, 7)
  pad[1, 5:] = 1.0
  q = torch.randn(2, 10, generator=g)
  packed = layer.InitForSourcePacked(layer.theta, src, None, pad)
  state = layer.ZeroAttentionState(7, 2)
  return layer, packed, q, state


def _check_probs(probs, pad):
  assert probs.shape == (2, 7)
  assert torch.allclose(probs.sum(-1), torch.ones(2), atol=1e-4)
  assert probs[1, 5:].abs().max() < 1e-6  # padded positions masked


def test_additive_attention():
  layer, packed, q, state = _setup(al.AdditiveAttention)
  ctx, probs, _ = layer.ComputeContextVector(layer.theta, packed, q, state)
  assert ctx.shape == (2, 12)
  _check_probs(probs, None)
  # manual energy for b=0, s=0
  th = layer.theta
  e = torch.tanh(packed.source_vecs[0, 0] @ th.source_var +
                 q[0] @ th.query_var) @ th.hidden_var
  full = torch.tanh(packed.projected[0] + (q[0] @ th.query_var)) \
      @ th.hidden_var
  assert torch.allclose(probs[0], torch.softmax(full, -1), atol=1e-5)
  assert abs(full[0].item() - e.item()) < 1e-5


def test_dot_product_attention():
  layer, packed, q, state = _setup(al.DotProductAttention,
                                   query_dim=12)
  ctx, probs, _ = layer.ComputeContextVector(
      layer.theta, packed, torch.randn(2, 12), state)
  assert ctx.shape == (2, 12)
  _check_probs(probs, None)


def test_location_sensitive_attention_moves():
  layer, packed, q, state = _setup(al.LocationSensitiveAttention)
  assert state.atten_probs[:, 0].min() == 1.0
  ctx, probs, state2 = layer.ComputeContextVector(
      layer.theta, packed, q, state)
  _check_probs(probs, None)
  assert torch.allclose(state2.atten_probs, probs.float(), atol=1e-5)
  # differentiable through the location conv
  loss = ctx.sum()
  loss.backward()
  assert layer.location_filter_var.grad is not None


def test_monotonic_attention_mass_conserved():
  layer, packed, q, state = _setup(al.MonotonicAttention)
  layer.eval()
  alpha_prev = state.emit_probs
  ctx, alpha, state2 = layer.ComputeContextVector(
      layer.theta, packed, q, state)
  assert ctx.shape == (2, 12)
  # expected-alignment mass can only be <= the incoming mass
  assert (alpha.sum(-1) <= alpha_prev.sum(-1) + 1e-4).all()
  assert (alpha >= -1e-6).all()
  # a second step keeps the invariant
  _, alpha3, _ = layer.ComputeContextVector(
      layer.theta, packed, q, state2)
  assert (alpha3.sum(-1) <= alpha.sum(-1) + 1e-4).all()


def test_monotonic_attention_saturated_is_hard():
  """With p_choose ~ 1 at frame 0 the alignment stays at frame 0."""
  layer, packed, q, state = _setup(al.MonotonicAttention)
  layer.eval()
  with torch.no_grad():
    layer.energy_bias_var.fill_(100.0)  # sigmoid -> 1 everywhere
  _, alpha, _ = layer.ComputeContextVector(layer.theta, packed, q, state)
  assert alpha[0, 0].item() > 0.999
  assert alpha[0, 1:].abs().max() < 1e-3


def test_gmm_monotonic_attention_advances():
  layer, packed, q, state = _setup(al.GmmMonotonicAttention)
  ctx, probs, state2 = layer.ComputeContextVector(
      layer.theta, packed, q, state)
  assert ctx.shape == (2, 12)
  _check_probs(probs, None)
  # means advance monotonically (softplus delta >= 0)
  assert (state2.position >= state.position - 1e-6).all()
  _, _, state3 = layer.ComputeContextVector(layer.theta, packed, q, state2)
  assert (state3.position >= state2.position - 1e-6).all()


def test_merger_layer_modes():
  g = torch.Generator().manual_seed(1)
  xs = [torch.randn(2, 4, generator=g) for _ in range(3)]
  for op in ['mean', 'sum', 'concat', 'weighted_sum', 'gated_avg']:
    m = al.MergerLayer.Params().Set(
        name='m', merger_op=op, num_sources=3, source_dim=4,
        random_seed=2).Instantiate()
    out = m.FProp(m.theta, xs)
    if op == 'concat':
      assert out.shape == (2, 12)
    else:
      assert out.shape == (2, 4)
  m = al.MergerLayer.Params().Set(name='m', merger_op='mean').Instantiate()
  assert torch.allclose(m.FProp(m.theta, xs),
                        (xs[0] + xs[1] + xs[2]) / 3, atol=1e-6)


def test_legacy_multiheaded_wrapper():
  from lingvo_amd.layers import attention_legacy as al
  p = al.MultiHeadedAttention.Params().Set(
      name='mha', source_dim=16, query_dim=12, hidden_dim=32,
      num_attention_heads=4, random_seed=1)
  layer = p.Instantiate()
  src = torch.randn(2, 7, 16)
  pad = torch.zeros(2, 7)
  pad[1, 5:] = 1.0
  packed = layer.InitForSourcePacked(layer.theta, src, None, pad)
  q = torch.randn(2, 12)
  ctx, probs, _ = layer.ComputeContextVector(
      layer.theta, packed, q, layer.ZeroAttentionState(7, 2))
  assert ctx.shape == (2, 32)
  assert probs.shape == (2, 7)
  # padded keys get ~zero prob; probs normalized
  assert float(probs[1, 5:].sum()) < 1e-6
  assert torch.allclose(probs.sum(-1), torch.ones(2), atol=1e-5)
  ctx.sum().backward()
  assert layer.source_proj.grad is not None


def test_multi_source_attention():
  from lingvo_amd.core.nested_map import NestedMap
  from lingvo_amd.layers import attention_legacy as al
  mk = lambda name: (name, al.DotProductAttention.Params().Set(
      source_dim=8, query_dim=8, hidden_dim=8, random_seed=1))
  p = al.MultiSourceAttention.Params().Set(
      name='ms', source_atten_tpls=[mk('a'), mk('b')],
      primary_source_key='b')
  layer = p.Instantiate()
  srcs = NestedMap(a=torch.randn(2, 5, 8), b=torch.randn(2, 3, 8))
  pads = NestedMap(a=torch.zeros(2, 5), b=torch.zeros(2, 3))
  packed = layer.InitForSourcePacked(layer.theta, srcs, None, pads)
  q = torch.randn(2, 8)
  ctx, probs, _ = layer.ComputeContextVector(layer.theta, packed, q,
                                             NestedMap())
  assert ctx.shape == (2, 8)
  assert probs.shape == (2, 3)  # primary source 'b' probs
