"""MoE tests: gating semantics, EP all-to-all equivalence (gloo ws=2),
MoE LM train step."""

import os

import pytest
from conftest import dist_port
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from lingvo_amd.core import registry
from lingvo_amd.parallel import moe as moe_lib


def test_top2_gating_capacity_and_aux():
  torch.manual_seed(0)
  n, e = 64, 4
  logits = torch.randn(n, e)
  g = moe_lib.Top2Gating(logits, capacity=8)
  assert g.top1.shape == (n,)
  assert (g.top1 != g.top2).all()
  # positions within capacity where kept
  assert (g.pos1[g.keep1] < 8).all()
  # gates renormalized: g1+g2 == 1 where both kept
  both = g.keep1 & g.keep2
  assert torch.allclose((g.g1 + g.g2)[both], torch.ones(int(both.sum())),
                        atol=1e-5)
  assert g.aux_loss > 0


def _moe_layer(seed=3):
  p = moe_lib.MoEFeedForwardLayer.Params().Set(
      name='moe', input_dim=16, hidden_dim=32, num_experts=4,
      expert_capacity_factor=2.0, random_seed=seed)
  return p.Instantiate()


def test_moe_local_forward_backward():
  layer = _moe_layer()
  x = torch.randn(2, 12, 16, requires_grad=True)
  out = layer.FProp(layer.theta, x)
  assert out.shape == x.shape
  (out.sum() + layer.AuxLoss()).backward()
  assert layer.gate_w.grad is not None
  assert layer.wi.grad is not None


def _run_ep(rank, world, port, results):
  os.environ['MASTER_ADDR'] = '127.0.0.1'
  os.environ['MASTER_PORT'] = str(port)
  dist.init_process_group('gloo', rank=rank, world_size=world)
  layer = _moe_layer()
  g = torch.Generator().manual_seed(500 + rank)
  x = torch.randn(1, 16, 16, generator=g, requires_grad=True)
  out = layer.FProp(layer.theta, x)
  (out.sum() + layer.AuxLoss()).backward()
  results[f'out{rank}'] = out.detach()
  results[f'wi_shape{rank}'] = tuple(layer.wi.shape)
  results[f'wi_sharded{rank}'] = getattr(layer.wi, '_ep_sharded', False)
  results[f'wi_grad{rank}'] = layer.wi.grad.clone()
  dist.destroy_process_group()


def test_moe_ep2_matches_local():
  """EP=2 with E-dim sharded expert weights: each rank stores E/W
  experts (memory = total/W), outputs equal the local all-expert
  computation on that rank's tokens, and each rank's expert grads
  accumulate contributions from BOTH ranks' losses (via the
  all-to-all backward) with no DP all-reduce needed."""
  ctx = mp.get_context('spawn')
  with ctx.Manager() as mgr:
    results = mgr.dict()
    procs = [ctx.Process(target=_run_ep, args=(r, 2, dist_port(29534), results))
             for r in range(2)]
    for p in procs:
      p.start()
    for p in procs:
      p.join(120)
      assert p.exitcode == 0
    outs = {r: results[f'out{r}'] for r in range(2)}
    wigrads = {r: results[f'wi_grad{r}'] for r in range(2)}
    shapes = {r: results[f'wi_shape{r}'] for r in range(2)}
    sharded = {r: results[f'wi_sharded{r}'] for r in range(2)}

  # Per-rank expert memory = total / EP world (2 of 4 experts).
  assert shapes[0] == (2, 16, 32) and shapes[1] == (2, 16, 32)
  assert sharded[0] and sharded[1]

  ref_grads = []
  for rank in range(2):
    layer = _moe_layer()
    g = torch.Generator().manual_seed(500 + rank)
    x = torch.randn(1, 16, 16, generator=g, requires_grad=True)
    out = layer.FProp(layer.theta, x)
    (out.sum() + layer.AuxLoss()).backward()
    assert torch.allclose(outs[rank], out.detach(), atol=1e-5), rank
    ref_grads.append(layer.wi.grad.clone())

  # Sharded wi grad on rank r == the summed full-model grad rows of
  # rank r's experts (experts 0-1 on rank 0, 2-3 on rank 1).
  total = ref_grads[0] + ref_grads[1]
  assert torch.allclose(wigrads[0], total[:2], atol=1e-5)
  assert torch.allclose(wigrads[1], total[2:], atol=1e-5)


def test_moe_lm_train_step():
  p = registry.GetParams('lm.synthetic_packed_input.MoELm64E', 'Train')
  p.task.fprop_dtype = torch.float32
  p.task.lm.Set(model_dim=32, num_layers=2, num_heads=1, hidden_dim=64,
                vocab_size=64, num_experts=4, moe_every_n=2)
  p.input.Set(batch_size=2, seq_len=16, vocab_size=64)
  task = p.Instantiate().GetTask()
  m = task.TrainStep(task.GetInputBatch())
  assert torch.isfinite(m['loss'][0])
  # MoE layer exists and produced an aux loss
  from lingvo_amd.parallel.moe import MoEFeedForwardLayer
  moes = [mm for mm in task.modules()
          if isinstance(mm, MoEFeedForwardLayer)]
  assert len(moes) == 1
  assert moes[0].AuxLoss() is not None


def test_expert_choice_gating_balanced():
  import torch
  from lingvo_amd.parallel import moe
  g = torch.Generator().manual_seed(5)
  logits = torch.randn(32, 4, generator=g)
  out = moe.ExpertChoiceGating(logits, capacity=8)
  assert out.idx.shape == (4, 8) and out.gates.shape == (4, 8)
  # every expert processes exactly `capacity` tokens: balanced
  assert out.idx.max() < 32
  assert (out.gates >= 0).all() and (out.gates <= 1).all()


def test_moe_layer_expert_choice_path():
  import torch
  from lingvo_amd.parallel import moe
  p = moe.MoEFeedForwardLayer.Params().Set(
      name='moe', input_dim=16, hidden_dim=32, num_experts=4,
      gating='expert_choice', expert_capacity_factor=2.0, random_seed=3)
  layer = p.Instantiate()
  x = torch.randn(2, 12, 16, requires_grad=True)
  pad = torch.zeros(2, 12)
  pad[1, 10:] = 1.0
  out = layer.FProp(layer.theta, x, pad)
  assert out.shape == x.shape
  assert out[1, 10:].abs().max() < 1e-6
  out.sum().backward()
  assert layer.wi.grad is not None and x.grad is not None
  assert float(layer.AuxLoss()) == 0.0  # balanced by construction
