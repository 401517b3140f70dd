"""Optimizer unit tests (MasterAdamW bf16 path, Adafactor, schedules)."""

import math

import pytest
import torch

from lingvo_amd.core import optimizer as optimizer_lib
from lingvo_amd.core import schedule as schedule_lib


def test_master_adamw_matches_adamw_fp32():
  torch.manual_seed(0)
  w32 = torch.randn(32, 16)
  p_ref = w32.clone().requires_grad_(True)
  p_bf = w32.to(torch.bfloat16).requires_grad_(True)
  ref_opt = torch.optim.AdamW([p_ref], lr=0.01, betas=(0.9, 0.99),
                              eps=1e-8, weight_decay=0.01)
  bf_opt = optimizer_lib.MasterAdamW([p_bf], lr=0.01, betas=(0.9, 0.99),
                                     eps=1e-8, weight_decay=0.01)
  for i in range(10):
    g = torch.randn(32, 16)
    p_ref.grad = g.clone()
    p_bf.grad = g.to(torch.bfloat16)
    ref_opt.step()
    bf_opt.step()
  # master tracks the fp32 trajectory closely (bf16 grads only diff)
  master = bf_opt.state[p_bf]['master']
  assert (master - p_ref.detach()).abs().max() < 0.01
  # bf16 param is the rounded master
  assert torch.equal(p_bf.detach(), master.to(torch.bfloat16))


def test_master_adamw_state_dict_roundtrip():
  p_bf = torch.randn(8, 4).to(torch.bfloat16).requires_grad_(True)
  opt = optimizer_lib.MasterAdamW([p_bf], lr=0.01)
  p_bf.grad = torch.randn(8, 4).to(torch.bfloat16)
  opt.step()
  sd = opt.state_dict()
  p2 = p_bf.detach().clone().requires_grad_(True)
  opt2 = optimizer_lib.MasterAdamW([p2], lr=0.01)
  opt2.load_state_dict(sd)
  assert torch.equal(opt2.state[p2]['master'], opt.state[p_bf]['master'])


def test_adafactor_reduces_loss():
  torch.manual_seed(1)
  w = torch.nn.Parameter(torch.randn(256, 256))
  tgt = torch.randn(256, 256)
  p = optimizer_lib.Adafactor.Params().Set(name='af').Instantiate()
  opt = p.CreateTorchOptimizer([w], lr=0.1)
  losses = []
  for _ in range(20):
    loss = ((w - tgt) ** 2).mean()
    opt.zero_grad()
    loss.backward()
    opt.step()
    losses.append(float(loss))
  assert losses[-1] < losses[0] * 0.5
  # factored second moment (vr/vc not full v)
  st = opt.state[w]
  assert st['factored'] and 'vr' in st


def test_schedules():
  s = schedule_lib.TransformerSchedule.Params().Set(
      name='t', warmup_steps=100, model_dim=512).Instantiate()
  assert s.Value(50) < s.Value(100)
  assert s.Value(100) > s.Value(10000)
  c = schedule_lib.CosineSchedule.Params().Set(
      name='c', total_steps=100).Instantiate()
  assert abs(c.Value(0) - 1.0) < 1e-6
  assert abs(c.Value(100)) < 1e-6
  pw = schedule_lib.PiecewiseConstantSchedule.Params().Set(
      name='p', boundaries=[10, 20], values=[1.0, 0.5, 0.1]).Instantiate()
  assert pw.Value(5) == 1.0 and pw.Value(15) == 0.5 and pw.Value(25) == 0.1


def test_accumulator_matches_large_batch():
  """Accumulator(N) over N micro-batches == one step on the mean grad."""
  torch.manual_seed(7)
  data = [torch.randn(8, 16) for _ in range(4)]
  tgt = [torch.randn(8, 4) for _ in range(4)]

  def make(opt_params):
    torch.manual_seed(3)
    w = torch.nn.Parameter(torch.randn(16, 4))
    opt = opt_params.Instantiate().CreateTorchOptimizer([w], lr=0.1)
    return w, opt

  # Accumulated: 4 micro steps.
  wa, oa = make(optimizer_lib.Accumulator.Params().Set(
      name='acc', accum_steps=4,
      optimizer_tpl=optimizer_lib.SGD.Params()))
  for x, y in zip(data, tgt):
    oa.zero_grad()
    ((x @ wa - y) ** 2).mean().backward()
    oa.step()

  # Reference: one step on the concatenated batch (same mean grad).
  wb, ob = make(optimizer_lib.SGD.Params().Set(name='sgd'))
  ob.zero_grad()
  ((torch.cat(data) @ wb - torch.cat(tgt)) ** 2).mean().backward()
  ob.step()

  assert torch.allclose(wa.detach(), wb.detach(), atol=1e-6), \
      (wa - wb).abs().max()


def test_accumulator_no_update_between_applies():
  w = torch.nn.Parameter(torch.ones(4))
  p = optimizer_lib.Accumulator.Params().Set(
      name='acc', accum_steps=3,
      optimizer_tpl=optimizer_lib.SGD.Params())
  opt = p.Instantiate().CreateTorchOptimizer([w], lr=1.0)
  before = w.detach().clone()
  for i in range(2):
    opt.zero_grad()
    (w.sum()).backward()
    opt.step()
    assert torch.equal(w.detach(), before)  # accumulating only
  opt.zero_grad()
  (w.sum()).backward()
  opt.step()
  assert not torch.equal(w.detach(), before)  # applied on 3rd


def test_master_adamw_lazy_param_bias_correction():
  """A param whose grad appears late gets its own bias correction."""
  w1 = torch.nn.Parameter(torch.randn(8, dtype=torch.bfloat16))
  w2 = torch.nn.Parameter(torch.randn(8, dtype=torch.bfloat16))
  opt = optimizer_lib.MasterAdamW([w1, w2], lr=0.01)
  for i in range(5):
    opt.zero_grad()
    w1.grad = torch.ones_like(w1)
    if i >= 3:
      w2.grad = torch.ones_like(w2)
    opt.step()
  # w2 saw 2 steps; its state step count must be 2 (not 5).
  assert opt.state[w1]['step'] == 5
  assert opt.state[w2]['step'] == 2
  # Fresh optimizer stepping w2-like param twice gives same master.
