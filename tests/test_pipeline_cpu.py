"""GPipe pipeline tests: 2 stages over gloo == single-process reference
(gradients and loss must match exactly up to fp tolerance)."""

import os

import pytest
from conftest import dist_port
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
                         args=(r, 2, dist_port(29533), num_micro, results))
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


def _run_lm_stage(rank, world, port, num_micro, results):
  os.environ['MASTER_ADDR'] = '127.0.0.1'
  os.environ['MASTER_PORT'] = str(port)
  dist.init_process_group('gloo', rank=rank, world_size=world)
  from lingvo_amd.parallel.gpipe_lm import (RunGPipeLmStep,
                                            TransformerLmStage)
  from lingvo_amd.parallel.pipeline import GPipeRunner
  from lingvo_amd.core.nested_map import NestedMap
  stage_p = TransformerLmStage.Params().Set(
      name=f'stage', vocab_size=64, model_dim=32, num_layers_total=4,
      num_heads=1, hidden_dim=64, stage_idx=rank, num_stages=world,
      random_seed=21)
  stage = stage_p.Instantiate()
  runner = GPipeRunner(rank, world, num_micro)
  batches = []
  for m in range(num_micro):
    g = torch.Generator().manual_seed(900 + m)
    ids = torch.randint(1, 64, (2, 8), generator=g)
    batches.append(NestedMap(ids=ids, labels=ids.roll(-1, 1),
                             paddings=torch.zeros(2, 8),
                             weights=torch.ones(2, 8)))
  loss = RunGPipeLmStep(stage, runner, batches)
  results[f'loss{rank}'] = None if loss is None else float(loss)
  results[f'gnorm{rank}'] = float(torch.cat(
      [p.grad.reshape(-1) for p in stage.parameters()
       if p.grad is not None]).norm())
  dist.destroy_process_group()


def test_gpipe_transformer_lm_two_stages():
  num_micro = 4
  ctx = mp.get_context('spawn')
  with ctx.Manager() as mgr:
    results = mgr.dict()
    procs = [ctx.Process(target=_run_lm_stage,
                         args=(r, 2, dist_port(29536), num_micro, results))
             for r in range(2)]
    for p in procs:
      p.start()
    for p in procs:
      p.join(240)
      assert p.exitcode == 0
    loss = results['loss1']
    g0, g1 = results['gnorm0'], results['gnorm1']
  assert loss is not None and loss == loss  # finite
  assert g0 > 0 and g1 > 0

  # Single-process reference: both stages chained, same microbatches.
  from lingvo_amd.parallel.gpipe_lm import TransformerLmStage
  from lingvo_amd.core.nested_map import NestedMap
  stages = []
  for r in range(2):
    sp = TransformerLmStage.Params().Set(
        name='stage', vocab_size=64, model_dim=32, num_layers_total=4,
        num_heads=1, hidden_dim=64, stage_idx=r, num_stages=2,
        random_seed=21)
    stages.append(sp.Instantiate())
  losses = []
  for m in range(num_micro):
    g = torch.Generator().manual_seed(900 + m)
    ids = torch.randint(1, 64, (2, 8), generator=g)
    nmap = NestedMap(ids=ids, paddings=torch.zeros(2, 8))
    out = stages[0].FProp(stages[0].theta, nmap)
    out = stages[1].FProp(stages[1].theta, out)
    xent = stages[1].XentLoss(stages[1].theta, out.act,
                              ids.roll(-1, 1), torch.ones(2, 8))
    losses.append(xent.avg_xent)
  ref_loss = float(torch.stack(losses).mean())
  assert abs(loss - ref_loss) < 1e-4, (loss, ref_loss)


def _run_stage_1f1b(rank, world, port, num_micro, results):
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

  loss = runner.RunStep(fprop, input_fn=input_fn, loss_fn=loss_fn,
                        schedule='1f1b')
  results[f'loss{rank}'] = None if loss is None else float(loss)
  results[f'grads{rank}'] = torch.cat(
      [p.grad.reshape(-1) for p in stage.parameters()])
  dist.destroy_process_group()


@pytest.mark.parametrize('world', [2, 3])
def test_1f1b_matches_fill_drain_reference(world):
  num_micro = 5
  ctx = mp.get_context('spawn')
  with ctx.Manager() as mgr:
    results = mgr.dict()
    procs = [ctx.Process(target=_run_stage_1f1b,
                         args=(r, world, dist_port(29537) + world, num_micro,
                               results))
             for r in range(world)]
    for p in procs:
      p.start()
    for p in procs:
      p.join(180)
      assert p.exitcode == 0
    loss_pipe = results[f'loss{world - 1}']
    grads = [results[f'grads{r}'] for r in range(world)]

  # Single-process reference.
  stages = [_make_stage(r) for r in range(world)]
  losses = []
  for m in range(num_micro):
    x = _inputs(m)
    for st in stages:
      x = st(x)
    losses.append(((x - _targets(m)) ** 2).mean())
  total = torch.stack(losses).mean()
  total.backward()
  assert abs(loss_pipe - float(total)) < 1e-5
  for r in range(world):
    ref = torch.cat([p.grad.reshape(-1)
                     for p in stages[r].parameters()])
    assert torch.allclose(grads[r], ref, atol=1e-6), r


def _mt_batches(num_micro):
  from lingvo_amd.core.nested_map import NestedMap
  out = []
  for m in range(num_micro):
    g = torch.Generator().manual_seed(700 + m)
    src = torch.randint(3, 48, (2, 6), generator=g)
    tgt = torch.randint(3, 48, (2, 5), generator=g)
    out.append(NestedMap(
        src=NestedMap(ids=src, paddings=torch.zeros(2, 6)),
        tgt=NestedMap(ids=tgt, paddings=torch.zeros(2, 5),
                      labels=tgt.roll(-1, 1), weights=torch.ones(2, 5))))
  return out


def _mt_stage_params(rank, world):
  from lingvo_amd.parallel.gpipe_mt import TransformerMtStage
  return TransformerMtStage.Params().Set(
      name='stage', vocab_size=48, model_dim=32,
      num_encoder_layers=2, num_decoder_layers=2, num_heads=1,
      hidden_dim=64, stage_idx=rank, num_stages=world, random_seed=31)


def _run_mt_stage(rank, world, port, num_micro, schedule, results):
  os.environ['MASTER_ADDR'] = '127.0.0.1'
  os.environ['MASTER_PORT'] = str(port)
  dist.init_process_group('gloo', rank=rank, world_size=world)
  from lingvo_amd.parallel.gpipe_mt import RunGPipeMtStep
  from lingvo_amd.parallel.pipeline import GPipeRunner
  stage = _mt_stage_params(rank, world).Instantiate()
  runner = GPipeRunner(rank, world, num_micro)
  loss = RunGPipeMtStep(stage, runner, _mt_batches(num_micro),
                        schedule=schedule)
  results[f'loss{rank}'] = None if loss is None else float(loss)
  results[f'gnorm{rank}'] = float(torch.cat(
      [p.grad.reshape(-1) for p in stage.parameters()
       if p.grad is not None]).norm())
  dist.destroy_process_group()


@pytest.mark.parametrize('schedule', ['fill_drain', '1f1b'])
def test_gpipe_mt_two_stages(schedule):
  num_micro = 4
  ctx = mp.get_context('spawn')
  with ctx.Manager() as mgr:
    results = mgr.dict()
    procs = [ctx.Process(target=_run_mt_stage,
                         args=(r, 2, dist_port(29541) + (schedule == '1f1b'),
                               num_micro, schedule, results))
             for r in range(2)]
    for p in procs:
      p.start()
    for p in procs:
      p.join(240)
      assert p.exitcode == 0
    loss = results['loss1']
    g0, g1 = results['gnorm0'], results['gnorm1']

  # Single-process reference: chain both stages.
  stages = [_mt_stage_params(r, 2).Instantiate() for r in range(2)]
  batches = _mt_batches(num_micro)
  from lingvo_amd.core.nested_map import NestedMap
  losses = []
  for m in range(num_micro):
    b = batches[m]
    nmap = NestedMap(src_ids=b.src.ids,
                     src_paddings=b.src.paddings.float(),
                     tgt_ids=b.tgt.ids,
                     tgt_paddings=b.tgt.paddings.float())
    out = stages[0].FProp(stages[0].theta, nmap)
    out = stages[1].FProp(stages[1].theta, out)
    xent = stages[1].XentLoss(stages[1].theta, out.tgt, b.tgt.labels,
                              b.tgt.weights)
    losses.append(xent.avg_xent)
  ref = torch.stack(losses).mean()
  ref.backward()
  assert abs(loss - float(ref)) < 1e-4
  # gradient norms match the reference per stage
  ref0 = float(torch.cat([p.grad.reshape(-1)
                          for p in stages[0].parameters()
                          if p.grad is not None]).norm())
  ref1 = float(torch.cat([p.grad.reshape(-1)
                          for p in stages[1].parameters()
                          if p.grad is not None]).norm())
  assert abs(g0 - ref0) < 1e-4 * max(1, ref0)
  assert abs(g1 - ref1) < 1e-4 * max(1, ref1)


def _lm_batches_for(dp_idx, num_micro):
  from lingvo_amd.core.nested_map import NestedMap
  batches = []
  for m in range(num_micro):
    g = torch.Generator().manual_seed(7000 + dp_idx * 131 + m)
    ids = torch.randint(1, 64, (2, 8), generator=g)
    batches.append(NestedMap(ids=ids, labels=ids.roll(-1, 1),
                             paddings=torch.zeros(2, 8),
                             weights=torch.ones(2, 8)))
  return batches


def _run_ppdp(rank, world, port, num_micro, results):
  os.environ['MASTER_ADDR'] = '127.0.0.1'
  os.environ['MASTER_PORT'] = str(port)
  dist.init_process_group('gloo', rank=rank, world_size=world)
  from lingvo_amd.parallel.gpipe_lm import (RunGPipeLmStep,
                                            TransformerLmStage)
  from lingvo_amd.parallel.topology import PpDpTopology
  topo = PpDpTopology(num_stages=2)
  stage = TransformerLmStage.Params().Set(
      name='stage', vocab_size=64, model_dim=32, num_layers_total=4,
      num_heads=1, hidden_dim=64, stage_idx=topo.stage_idx,
      num_stages=2, random_seed=21).Instantiate()
  runner = topo.MakeRunner(num_micro)
  loss = RunGPipeLmStep(stage, runner,
                        _lm_batches_for(topo.dp_idx, num_micro))
  topo.AllReduceStageGrads(stage)
  results[f'grads{rank}'] = {
      n: q.grad.clone() for n, q in stage.named_parameters()
      if q.grad is not None}
  results[f'coord{rank}'] = (topo.stage_idx, topo.dp_idx)
  results[f'loss{rank}'] = None if loss is None else float(loss)
  dist.destroy_process_group()


def test_ppdp_grid_2x2_matches_single_process():
  """2 pipeline stages x 2 DP replicas == serial run over all data."""
  num_micro = 3
  ctx = mp.get_context('spawn')
  with ctx.Manager() as mgr:
    results = mgr.dict()
    procs = [ctx.Process(target=_run_ppdp,
                         args=(r, 4, dist_port(29561), num_micro, results))
             for r in range(4)]
    for p in procs:
      p.start()
    for p in procs:
      p.join(300)
      assert p.exitcode == 0
    results = dict(results)

  coords = {results[f'coord{r}']: r for r in range(4)}
  assert set(coords) == {(0, 0), (0, 1), (1, 0), (1, 1)}
  # same-stage replicas agree after the DP all-reduce
  for s in range(2):
    ga = results[f'grads{coords[(s, 0)]}']
    gb = results[f'grads{coords[(s, 1)]}']
    for n in ga:
      assert torch.allclose(ga[n], gb[n], atol=1e-6), n

  # serial reference: both stages chained over BOTH replicas' data
  from lingvo_amd.parallel.gpipe_lm import TransformerLmStage
  from lingvo_amd.core.nested_map import NestedMap
  stages = [TransformerLmStage.Params().Set(
      name='stage', vocab_size=64, model_dim=32, num_layers_total=4,
      num_heads=1, hidden_dim=64, stage_idx=r, num_stages=2,
      random_seed=21).Instantiate() for r in range(2)]
  rep_losses = []
  for dp_idx in range(2):
    micro = []
    for b in _lm_batches_for(dp_idx, num_micro):
      nmap = NestedMap(ids=b.ids, paddings=b.paddings)
      out = stages[0].FProp(stages[0].theta, nmap)
      out = stages[1].FProp(stages[1].theta, out)
      xent = stages[1].XentLoss(stages[1].theta, out.act, b.labels,
                                b.weights)
      micro.append(xent.avg_xent)
    rep_losses.append(torch.stack(micro).mean())
  (torch.stack(rep_losses).mean()).backward()
  for s in range(2):
    got = results[f'grads{coords[(s, 0)]}']
    want = {n: q.grad for n, q in stages[s].named_parameters()
            if q.grad is not None}
    assert set(got) == set(want)
    for n in want:
      assert torch.allclose(got[n], want[n], atol=1e-4), \
          (s, n, (got[n] - want[n]).abs().max())


def _run_lm_1f1b(rank, world, port, num_micro, schedule, results):
  os.environ['MASTER_ADDR'] = '127.0.0.1'
  os.environ['MASTER_PORT'] = str(port)
  dist.init_process_group('gloo', rank=rank, world_size=world)
  from lingvo_amd.parallel.gpipe_lm import (RunGPipeLmStep,
                                            TransformerLmStage)
  from lingvo_amd.parallel.pipeline import GPipeRunner
  from lingvo_amd.core.nested_map import NestedMap
  stage = TransformerLmStage.Params().Set(
      name='stage', vocab_size=64, model_dim=32, num_layers_total=4,
      num_heads=1, hidden_dim=64, stage_idx=rank, num_stages=world,
      random_seed=21).Instantiate()
  runner = GPipeRunner(rank, world, num_micro)
  batches = []
  for m in range(num_micro):
    g = torch.Generator().manual_seed(900 + m)
    ids = torch.randint(1, 64, (2, 8), generator=g)
    batches.append(NestedMap(ids=ids, labels=ids.roll(-1, 1),
                             paddings=torch.zeros(2, 8),
                             weights=torch.ones(2, 8)))
  loss = RunGPipeLmStep(stage, runner, batches, schedule=schedule)
  results[f'{schedule}_loss{rank}'] = \
      None if loss is None else float(loss)
  results[f'{schedule}_g{rank}'] = torch.cat(
      [p.grad.reshape(-1) for p in stage.parameters()
       if p.grad is not None])
  dist.destroy_process_group()


def test_gpipe_lm_1f1b_matches_fill_drain():
  ctx = mp.get_context('spawn')
  with ctx.Manager() as mgr:
    results = mgr.dict()
    for schedule in ('fill_drain', '1f1b'):
      procs = [ctx.Process(target=_run_lm_1f1b,
                           args=(r, 2, dist_port(29602), 4, schedule,
                                 results))
               for r in range(2)]
      for p in procs:
        p.start()
      for p in procs:
        p.join(240)
        assert p.exitcode == 0
    results = dict(results)
  assert abs(results['fill_drain_loss1'] - results['1f1b_loss1']) < 1e-6
  for r in range(2):
    assert torch.allclose(results[f'fill_drain_g{r}'],
                          results[f'1f1b_g{r}'], atol=1e-6), r


def test_partition_by_cost_minmax():
  from lingvo_amd.parallel.pipeline import PartitionByCost
  # Uniform costs with no extras reduces to the balanced split.
  parts = PartitionByCost([1.0] * 8, 4)
  assert [len(x) for x in parts] == [2, 2, 2, 2]
  # A heavy last-stage extra shifts layers off the last stage.
  parts = PartitionByCost([1.0] * 8, 4, extra_last=2.0)
  assert len(parts[-1]) < 2
  assert sum(len(x) for x in parts) == 8
  # contiguity
  flat = [i for x in parts for i in x]
  assert flat == list(range(8))
  # Heterogeneous costs: one expensive unit gets its own stage.
  parts = PartitionByCost([1, 1, 10, 1, 1, 1], 3)
  assert any(x == [2] for x in parts)


def test_partition_by_cost_balances_within_10pct():
  """32-layer GPipe LM config over 4 stages: max stage cost within 10%
  of the ideal (VERDICT item 8 acceptance)."""
  from lingvo_amd.parallel.pipeline import (PartitionByCost,
                                            SoftmaxFlops,
                                            TransformerLayerFlops)
  d, ff, v = 2048, 8192, 32000
  lc = TransformerLayerFlops(d, ff)
  sc = SoftmaxFlops(d, v)
  parts = PartitionByCost([lc] * 32, 4, extra_last=sc)
  loads = [len(x) * lc for x in parts]
  loads[-1] += sc
  ideal = (32 * lc + sc) / 4
  assert max(loads) <= ideal * 1.1, (loads, ideal)
  # last stage takes fewer transformer layers than the others
  assert len(parts[-1]) < len(parts[0])


def test_gpipe_lm_stage_uses_cost_partition():
  from lingvo_amd.parallel import gpipe_lm
  counts = []
  for s in range(4):
    p = gpipe_lm.TransformerLmStage.Params().Set(
        name=f's{s}', vocab_size=32000, model_dim=64, hidden_dim=256,
        num_layers_total=8, num_heads=2, stage_idx=s, num_stages=4)
    stage = p.Instantiate()
    counts.append(len(stage.layers))
  assert sum(counts) == 8
  # softmax-carrying stage takes the fewest layers (vocab >> dim)
  assert counts[-1] == min(counts)
