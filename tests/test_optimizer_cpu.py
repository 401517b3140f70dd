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
