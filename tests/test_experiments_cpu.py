"""Shampoo / AdaGraft / GradDrop / entmax / revnet / pruning tests."""

import torch

from lingvo_amd.core import optimizer as optimizer_lib
from lingvo_amd.core import optimizer_experiments as oe
from lingvo_amd.core import pruning_utils
from lingvo_amd.layers import activations
from lingvo_amd.layers import layers as lingvo_layers
from lingvo_amd.layers import reversible_layers as rev


def _quadratic_losses(opt_factory, steps=60):
  torch.manual_seed(0)
  w = torch.nn.Parameter(torch.randn(16, 8))
  target = torch.randn(16, 8)
  opt = opt_factory([w])
  losses = []
  for _ in range(steps):
    opt.zero_grad()
    loss = ((w - target) ** 2).sum()
    loss.backward()
    opt.step()
    losses.append(loss.item())
  return losses


def test_shampoo_converges_on_quadratic():
  layer = oe.Shampoo.Params().Set(
      name='sh', statistics_compute_steps=5).Instantiate()
  losses = _quadratic_losses(
      lambda ps: layer.CreateTorchOptimizer(ps, lr=0.05))
  assert losses[-1] < 0.05 * losses[0], losses[-1]


def test_shampoo_diag_fallback_for_1d():
  layer = oe.Shampoo.Params().Set(name='sh').Instantiate()
  b = torch.nn.Parameter(torch.randn(32))
  opt = layer.CreateTorchOptimizer([b], lr=0.1)
  (b ** 2).sum().backward()
  opt.step()
  assert 'diag' in opt.state[b]


def test_adagraft_direction_is_sgd():
  """With identical lr, grafting Adam magnitude on SGD direction steps
  exactly along -grad."""
  layer = oe.AdaGraft.Params().Set(name='ag').Instantiate()
  w = torch.nn.Parameter(torch.tensor([3.0, -4.0]))
  opt = layer.CreateTorchOptimizer([w], lr=0.1)
  before = w.detach().clone()
  loss = (w * torch.tensor([1.0, 2.0])).sum()
  loss.backward()
  opt.step()
  step = w.detach() - before
  # direction parallel to -grad = -[1, 2]
  cos = torch.dot(step, torch.tensor([-1.0, -2.0])) / (
      step.norm() * torch.tensor([1.0, 2.0]).norm())
  assert cos.item() > 0.999


def test_graddrop_error_feedback():
  gd = oe.GradDropCompressor(keep_frac=0.25)
  g = torch.tensor([4.0, 0.1, 0.2, -0.3])
  out = gd.compress('w', g)
  # only the top-1 (25% of 4) survives
  assert out.tolist() == [4.0, 0.0, 0.0, 0.0]
  # dropped mass fed back: same grad again -> residual doubles smalls
  out2 = gd.compress('w', g)
  assert out2[0] == 4.0
  # after enough steps the residual forces smalls through
  for _ in range(10):
    out_n = gd.compress('w', g)
  assert (gd._residual['w'].abs().max() <= 0.3 * 12 + 1e-5)


def test_entmax15_properties():
  x = torch.randn(4, 9)
  p = activations.GetFn('ENTMAX15')(x)
  assert torch.allclose(p.sum(-1), torch.ones(4), atol=1e-5)
  assert (p >= 0).all()
  # sharper than softmax: more zeros with a spiky input
  spiky = torch.tensor([[5.0, 0.0, 0.0, 0.0]])
  ps = activations.Entmax15(spiky)
  assert ps[0, 1:].max() < 1e-6
  assert abs(ps[0, 0].item() - 1.0) < 1e-6


def test_revnet_matches_plain_and_reconstructs():
  def mk_fg(seed):
    return lingvo_layers.FCLayer.Params().Set(
        input_dim=8, output_dim=8, activation='TANH', random_seed=seed)

  p = rev.StackedRevNetLayer.Params().Set(
      name='rev', block_tpls=[
          rev.RevNetLayer.Params().Set(
              name=f'b{i}', f_tpl=mk_fg(10 + i), g_tpl=mk_fg(20 + i))
          for i in range(3)
      ])
  stack = p.Instantiate()
  x = torch.randn(4, 16, requires_grad=True)
  out = stack.FProp(stack.theta, x)
  loss = (out ** 2).sum()
  loss.backward()
  g_rev = x.grad.clone()
  w_grads = {n: q.grad.clone() for n, q in stack.named_parameters()}

  # plain (non-reversible) forward of the same math
  x2 = x.detach().clone().requires_grad_(True)
  x1p, x2p = x2.chunk(2, dim=-1)
  for i, block in enumerate(stack.blocks):
    th = stack.theta.blocks[i]
    x1p = x1p + block.f.FProp(th.f, x2p)
    x2p = x2p + block.g.FProp(th.g, x1p)
  out2 = torch.cat([x1p, x2p], dim=-1)
  assert torch.allclose(out, out2, atol=1e-5)
  for q in stack.parameters():
    q.grad = None
  ((out2 ** 2).sum()).backward()
  assert torch.allclose(g_rev, x2.grad, atol=1e-4)
  for n, q in stack.named_parameters():
    assert torch.allclose(w_grads[n], q.grad, atol=1e-4), n


def test_magnitude_pruner_schedule_and_masks():
  s0 = pruning_utils.PolynomialSparsity(0, 0.0, 0.9, 0, 100)
  s50 = pruning_utils.PolynomialSparsity(50, 0.0, 0.9, 0, 100)
  s100 = pruning_utils.PolynomialSparsity(100, 0.0, 0.9, 0, 100)
  assert s0 == 0.0 and s100 == 0.9 and 0.0 < s50 < 0.9

  model = torch.nn.Linear(32, 32)
  pruner = pruning_utils.MagnitudePruner(
      model, weight_regex='weight', final_sparsity=0.75, begin_step=0,
      end_step=10, frequency=1, min_numel=16)
  assert len(pruner._targets) == 1  # bias is 1-D, excluded
  pruner.Prune(10)
  sp = pruner.MeasuredSparsity()
  assert abs(sp - 0.75) < 0.02, sp
  # surviving weights keep their values; masked entries stay 0 after
  # a (fake) dense update followed by ApplyMasks
  with torch.no_grad():
    model.weight.add_(0.5)
  pruner.ApplyMasks()
  mask = pruner.masks['weight']
  assert (model.weight.detach()[~mask] == 0).all()


def test_egdd_optimizer_converges_and_clips():
  import torch
  from lingvo_amd.core import optimizer_experiments as oe
  torch.manual_seed(0)
  w = torch.nn.Parameter(torch.tensor([4.0, -3.0]))
  opt_p = oe.EGDD.Params().Set(name='egdd', momentum=0.9)
  opt = opt_p.Instantiate().CreateTorchOptimizer([w], lr=0.05)
  for _ in range(200):
    opt.zero_grad()
    loss = (w ** 2).sum()
    loss.backward()
    opt.step()
  assert float((w ** 2).sum()) < 1e-2
  st = opt.state[w]
  assert float(st['gain'].max()) <= 100.0
  assert 0.1 <= float(st['lr_scale']) <= 10.0
