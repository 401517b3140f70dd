"""GPipe pipeline tests: 2 stages over gloo == single-process reference
(gradients and loss must match exactly up to fp tolerance)."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp


def _make_stage(stage, seed=11):
  torch.manual_seed(seed + stage)
  return torch.nn.Sequential(torch.nn.Linear(16, 16), torch.nn.Tanh(),
                             torch.nn.Linear(16, 16), torch.nn.Tanh())


def _inputs(m):
  g = torch.Generator().manual_seed(100 + m)
  return torch.randn(4, 16, generator=g)


def _targets(m):
  g = torch.Generator().manual_seed(200 + m)
  return torch.randn(4, 16, generator=g)


def _run_stage(rank, world, port, num_micro, results):
  os.environ['MASTER_ADDR'] = '127.0.0.1'
  os.environ['MASTER_PORT'] = str(port)
  dist.init_process_group('gloo', rank=rank, world_size=world)
  from lingvo_amd.core.nested_map import NestedMap
  from lingvo_amd.parallel.pipeline import GPipeRunner

  stage = _make_stage(rank)
  runner = GPipeRunner(rank, world, num_micro)

  def fprop(nmap):
    return NestedMap(act=stage(nmap.act))

  def input_fn(m):
    return NestedMap(act=_inputs(m))

  def loss_fn(nmap, m):
    return ((nmap.act - _targets(m)) ** 2).mean()

  loss = runner.RunStep(fprop, input_fn=input_fn, loss_fn=loss_fn)
  results[f'loss{rank}'] = None if loss is None else float(loss)
  results[f'grads{rank}'] = torch.cat(
      [p.grad.reshape(-1) for p in stage.parameters()])
  dist.destroy_process_group()


def test_gpipe_two_stage_matches_reference():
  num_micro = 4
  ctx = mp.get_context('spawn')
  with ctx.Manager() as mgr:
    results = mgr.dict()
    procs = [ctx.Process(target=_run_stage,
                         args=(r, 2, 29533, num_micro, results))
             for r in range(2)]
    for p in procs:
      p.start()
    for p in procs:
      p.join(120)
      assert p.exitcode == 0
    loss_pipe = results['loss1']
    g0 = results['grads0']
    g1 = results['grads1']

  # Single-process reference: same stages applied sequentially, losses
  # averaged across the 4 microbatches.
  s0, s1 = _make_stage(0), _make_stage(1)
  losses = []
  for m in range(num_micro):
    out = s1(s0(_inputs(m)))
    losses.append(((out - _targets(m)) ** 2).mean())
  total = torch.stack(losses).mean()
  total.backward()
  ref0 = torch.cat([p.grad.reshape(-1) for p in s0.parameters()])
  ref1 = torch.cat([p.grad.reshape(-1) for p in s1.parameters()])

  assert abs(loss_pipe - float(total)) < 1e-5
  assert torch.allclose(g0, ref0, atol=1e-6), (g0 - ref0).abs().max()
  assert torch.allclose(g1, ref1, atol=1e-6)


def test_partition_sequential_layers():
  from lingvo_amd.parallel.pipeline import PartitionSequentialLayers
  parts = PartitionSequentialLayers(list(range(10)), 4)
  assert [len(p) for p in parts] == [3, 3, 2, 2]
  assert sum(parts, []) == list(range(10))
