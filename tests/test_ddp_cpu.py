"""Multi-process DP tests on gloo (world_size=2, CPU) — the distributed
path the driver's 8-GPU scaling bench exercises with RCCL."""

import os

import pytest
from conftest import dist_port
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from lingvo_amd.core.nested_map import NestedMap


def _run_gradsync(rank, world, port, results):
  os.environ['MASTER_ADDR'] = '127.0.0.1'
  os.environ['MASTER_PORT'] = str(port)
  dist.init_process_group('gloo', rank=rank, world_size=world)
  torch.manual_seed(123)  # same init on both ranks
  model = torch.nn.Sequential(
      torch.nn.Linear(8, 32), torch.nn.ReLU(), torch.nn.Linear(32, 4))
  from lingvo_amd.parallel.ddp import GradSync
  sync = GradSync(model, bucket_cap_mb=0.0001)  # force multiple buckets
  torch.manual_seed(rank)  # different data per rank
  x = torch.randn(4, 8)
  y = model(x).sum()
  y.backward()
  sync.Finalize()
  grads = torch.cat([p.grad.reshape(-1) for p in model.parameters()])
  results[rank] = grads
  dist.destroy_process_group()


def test_gradsync_averages_across_ranks(tmp_path):
  port = dist_port(29531)
  ctx = mp.get_context('spawn')
  with ctx.Manager() as mgr:
    results = mgr.dict()
    procs = [ctx.Process(target=_run_gradsync, args=(r, 2, port, results))
             for r in range(2)]
    for p in procs:
      p.start()
    for p in procs:
      p.join(120)
      assert p.exitcode == 0
    g0, g1 = results[0], results[1]
  # Both ranks end with identical (averaged) grads.
  assert torch.allclose(g0, g1, atol=1e-6)

  # And they equal the mean of per-rank local grads computed standalone.
  local = []
  for rank in range(2):
    torch.manual_seed(123)
    model = torch.nn.Sequential(
        torch.nn.Linear(8, 32), torch.nn.ReLU(), torch.nn.Linear(32, 4))
    torch.manual_seed(rank)
    x = torch.randn(4, 8)
    model(x).sum().backward()
    local.append(torch.cat([p.grad.reshape(-1)
                            for p in model.parameters()]))
  want = (local[0] + local[1]) / 2
  assert torch.allclose(g0, want, atol=1e-5)


def _run_bench_dp(rank, world, port, results):
  os.environ['MASTER_ADDR'] = '127.0.0.1'
  os.environ['MASTER_PORT'] = str(port)
  os.environ['WORLD_SIZE'] = str(world)
  os.environ['RANK'] = str(rank)
  dist.init_process_group('gloo', rank=rank, world_size=world)
  from lingvo_amd.core import registry
  from lingvo_amd.parallel.ddp import GradSync
  model_p = registry.GetParams('image.mnist.LeNet5', 'Train')
  model_p.task.random_seed = 42
  model = model_p.Instantiate()
  task = model.GetTask()
  sync = GradSync(task)
  for _ in range(2):
    batch = task.GetInputBatch()
    metrics = task.TrainStep(batch, grad_sync_finalize=sync.Finalize)
  # weights identical across ranks after synced steps
  flat = torch.cat([p.detach().reshape(-1) for p in task.parameters()])
  results[rank] = flat
  dist.destroy_process_group()


def test_full_task_dp_training_keeps_replicas_in_sync():
  port = dist_port(29532)
  ctx = mp.get_context('spawn')
  with ctx.Manager() as mgr:
    results = mgr.dict()
    procs = [ctx.Process(target=_run_bench_dp, args=(r, 2, port, results))
             for r in range(2)]
    for p in procs:
      p.start()
    for p in procs:
      p.join(240)
      assert p.exitcode == 0
    assert torch.allclose(results[0], results[1], atol=1e-6)


def _run_sharded_ckpt(rank, world, port, tmpdir, results):
  os.environ['MASTER_ADDR'] = '127.0.0.1'
  os.environ['MASTER_PORT'] = str(port)
  dist.init_process_group('gloo', rank=rank, world_size=world)
  from lingvo_amd.core import checkpointer as ckpt_lib

  class Shard(torch.nn.Module):
    def __init__(self):
      super().__init__()
      # per-rank distinct parameter (like a TP/PP shard)
      self.w = torch.nn.Parameter(
          torch.full((4,), float(rank + 1)))
      self.global_step = 7

  model = Shard()
  opt = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9)
  (model.w ** 2).sum().backward()
  opt.step()
  ck = ckpt_lib.ShardedCheckpointer(
      ckpt_lib.Checkpointer.Params().Set(keep_latest_n=2),
      tmpdir, model, [opt])
  ck.Save(step=7)
  saved_w = model.w.detach().clone()
  with torch.no_grad():
    model.w.fill_(-99.0)
  opt2 = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9)
  ck2 = ckpt_lib.ShardedCheckpointer(
      ckpt_lib.Checkpointer.Params(), tmpdir, model, [opt2])
  step = ck2.Restore()
  results[f'ok{rank}'] = bool(
      step == 7 and torch.allclose(model.w.detach(), saved_w) and
      'momentum_buffer' in list(opt2.state.values())[0])
  dist.destroy_process_group()


def test_sharded_checkpoint_roundtrip(tmp_path):
  ctx = mp.get_context('spawn')
  port = 29563 + os.getpid() % 997  # avoid TIME_WAIT collisions
  with ctx.Manager() as mgr:
    results = mgr.dict()
    procs = [ctx.Process(target=_run_sharded_ckpt,
                         args=(r, 2, port, str(tmp_path), results))
             for r in range(2)]
    for p in procs:
      p.start()
    for p in procs:
      p.join(120)
      assert p.exitcode == 0
    assert results['ok0'] and results['ok1']
  import glob as globlib
  shards = globlib.glob(str(tmp_path / 'ckpt-00000007.shard-*.pt'))
  assert len(shards) == 2
  with open(tmp_path / 'checkpoint') as f:
    txt = f.read()
  assert 'ckpt-00000007' in txt and 'num_shards: 2' in txt


def _run_graddrop_sync(rank, world, port, results):
  os.environ['MASTER_ADDR'] = '127.0.0.1'
  os.environ['MASTER_PORT'] = str(port)
  dist.init_process_group('gloo', rank=rank, world_size=world)
  from lingvo_amd.core.optimizer_experiments import GradDropCompressor
  from lingvo_amd.parallel.ddp import GradSync
  lin = torch.nn.Linear(4, 1, bias=False)
  with torch.no_grad():
    lin.weight.fill_(0.0)
  comp = GradDropCompressor(keep_frac=0.25)
  sync = GradSync(lin, compressor=comp)
  # per-rank distinct grads: rank r has one large element at index r.
  # Set .grad directly (no backward) to exercise the Finalize pull
  # path, which also runs the compressor.
  g = torch.full((1, 4), 0.1)
  g[0, rank] = 4.0
  lin.weight.grad = g.clone()
  sync.Finalize()
  results[f'grad{rank}'] = lin.weight.grad.clone()
  results[f'res{rank}'] = comp._residual['weight'].clone()
  dist.destroy_process_group()


def test_gradsync_with_graddrop_compressor():
  ctx = mp.get_context('spawn')
  port = 29570 + os.getpid() % 997
  with ctx.Manager() as mgr:
    results = mgr.dict()
    procs = [ctx.Process(target=_run_graddrop_sync,
                         args=(r, 2, port, results))
             for r in range(2)]
    for p in procs:
      p.start()
    for p in procs:
      p.join(120)
      assert p.exitcode == 0
    g0, g1 = results['grad0'], results['grad1']
    res0 = results['res0']
  # both ranks see the same averaged compressed grad: rank r kept only
  # its 4.0 element -> average [2.0, 2.0, 0, 0]
  assert torch.allclose(g0, g1)
  assert torch.allclose(g0, torch.tensor([[2.0, 2.0, 0.0, 0.0]]),
                        atol=1e-5)
  # dropped 0.1 elements live in the residual for the next step
  assert abs(float(res0[0, 2]) - 0.1) < 1e-6


def _run_two_shot(rank, world, port, results):
  os.environ['MASTER_ADDR'] = '127.0.0.1'
  os.environ['MASTER_PORT'] = str(port)
  dist.init_process_group('gloo', rank=rank, world_size=world)
  from lingvo_amd.parallel.ddp import TwoShotAllReduce
  g = torch.Generator().manual_seed(100 + rank)
  t = torch.randn(37, generator=g)  # odd size: exercises padding
  ref = t.clone()
  dist.all_reduce(ref)
  out = TwoShotAllReduce(t)
  results[f'ok{rank}'] = bool(torch.allclose(out, ref, atol=1e-6))
  dist.destroy_process_group()


def test_two_shot_all_reduce_matches_ring():
  ctx = mp.get_context('spawn')
  port = dist_port(29597)
  with ctx.Manager() as mgr:
    results = mgr.dict()
    procs = [ctx.Process(target=_run_two_shot, args=(r, 2, port, results))
             for r in range(2)]
    for p in procs:
      p.start()
    for p in procs:
      p.join(120)
      assert p.exitcode == 0
    assert results['ok0'] and results['ok1']


def _run_sync_bn(rank, world, port, results):
  os.environ['MASTER_ADDR'] = '127.0.0.1'
  os.environ['MASTER_PORT'] = str(port)
  dist.init_process_group('gloo', rank=rank, world_size=world)
  from lingvo_amd.layers import bn_layers
  bn = bn_layers.BatchNormLayer.Params().Set(
      name='bn', dim=4, enable_cross_replica_sum_on_tpu=True,
      random_seed=3).Instantiate()
  bn.train()
  # replicas see shards with DIFFERENT means
  g = torch.Generator().manual_seed(500 + rank)
  x = torch.randn(3, 6, 4, generator=g) + 3.0 * rank
  out = bn.FProp(bn.theta, x, torch.zeros(3, 6))
  results[f'out{rank}'] = out.detach()
  results[f'x{rank}'] = x
  dist.destroy_process_group()


def test_sync_batch_norm_matches_global_moments():
  ctx = mp.get_context('spawn')
  port = dist_port(29599)
  with ctx.Manager() as mgr:
    results = mgr.dict()
    procs = [ctx.Process(target=_run_sync_bn, args=(r, 2, port, results))
             for r in range(2)]
    for p in procs:
      p.start()
    for p in procs:
      p.join(120)
      assert p.exitcode == 0
    results = dict(results)
  # single-process reference over the CONCATENATED batch
  from lingvo_amd.layers import bn_layers
  bn = bn_layers.BatchNormLayer.Params().Set(
      name='bn', dim=4, random_seed=3).Instantiate()
  bn.train()
  x_all = torch.cat([results['x0'], results['x1']], dim=0)
  ref = bn.FProp(bn.theta, x_all, torch.zeros(6, 6))
  assert torch.allclose(results['out0'], ref[:3], atol=1e-4)
  assert torch.allclose(results['out1'], ref[3:], atol=1e-4)


def _run_two_shot_sizes(rank, world, port, results):
  os.environ['MASTER_ADDR'] = '127.0.0.1'
  os.environ['MASTER_PORT'] = str(port)
  dist.init_process_group('gloo', rank=rank, world_size=world)
  from lingvo_amd.parallel.ddp import TwoShotAllReduce
  ok = True
  for n in (1, 3, 64, 257):
    g = torch.Generator().manual_seed(10 * n + rank)
    t = torch.randn(n, generator=g)
    ref = t.clone()
    dist.all_reduce(ref)
    out = TwoShotAllReduce(t)
    ok = ok and torch.allclose(out, ref, atol=1e-6)
  results[f'ok{rank}'] = ok
  dist.destroy_process_group()


def test_two_shot_all_reduce_size_sweep():
  ctx = mp.get_context('spawn')
  port = dist_port(29606)
  with ctx.Manager() as mgr:
    results = mgr.dict()
    procs = [ctx.Process(target=_run_two_shot_sizes,
                         args=(r, 2, port, results)) for r in range(2)]
    for p in procs:
      p.start()
    for p in procs:
      p.join(120)
      assert p.exitcode == 0
    assert results['ok0'] and results['ok1']


def _run_graddrop_hooks(rank, world, port, results):
  os.environ['MASTER_ADDR'] = '127.0.0.1'
  os.environ['MASTER_PORT'] = str(port)
  dist.init_process_group('gloo', rank=rank, world_size=world)
  from lingvo_amd.core.optimizer_experiments import GradDropCompressor
  from lingvo_amd.parallel.ddp import GradSync
  torch.manual_seed(7)  # identical init
  lin = torch.nn.Linear(8, 4)
  comp = GradDropCompressor(keep_frac=0.5)
  sync = GradSync(lin, compressor=comp)
  torch.manual_seed(100 + rank)  # distinct data
  x = torch.randn(4, 8)
  lin(x).sum().backward()  # hooks fire -> compressed into buckets
  sync.Finalize()
  results[f'g{rank}'] = torch.cat(
      [p.grad.reshape(-1) for p in lin.parameters()])
  results[f'res{rank}'] = torch.cat(
      [v.reshape(-1) for v in comp._residual.values()])
  dist.destroy_process_group()


def test_gradsync_compressor_hook_path():
  """Backward-triggered hook path also runs the compressor: ranks end
  identical and residuals hold the dropped mass."""
  ctx = mp.get_context('spawn')
  port = dist_port(29608)
  with ctx.Manager() as mgr:
    results = mgr.dict()
    procs = [ctx.Process(target=_run_graddrop_hooks,
                         args=(r, 2, port, results)) for r in range(2)]
    for p in procs:
      p.start()
    for p in procs:
      p.join(120)
      assert p.exitcode == 0
    results = dict(results)
  assert torch.allclose(results['g0'], results['g1'], atol=1e-6)
  assert results['res0'].abs().sum() > 0  # something was dropped


def _run_syncbn(rank, world, port, results):
  os.environ['MASTER_ADDR'] = '127.0.0.1'
  os.environ['MASTER_PORT'] = str(port)
  dist.init_process_group('gloo', rank=rank, world_size=world)
  from lingvo_amd.layers import bn_layers
  p = bn_layers.BatchNormLayer.Params().Set(
      name='bn', dim=8, enable_cross_replica_sum_on_tpu=True,
      random_seed=1)
  bn = p.Instantiate()
  torch.manual_seed(100 + rank)  # different data per rank
  x = torch.randn(3, 5, 8) * (rank + 1) + rank  # different mean/var
  pad = torch.zeros(3, 5)
  pad[0, 3:] = 1.0
  out = bn.FProp(bn.theta, x, pad)
  results[f'out{rank}'] = out.detach()
  results[f'x{rank}'] = x
  results[f'pad{rank}'] = pad
  dist.destroy_process_group()


def test_sync_batch_norm_uses_global_moments():
  """Cross-replica BN: each rank normalizes with GLOBAL moments
  (sufficient-statistics all-reduce, reference bn_layers.py:139
  cross-replica option)."""
  port = dist_port(29561)
  ctx = mp.get_context('spawn')
  with ctx.Manager() as mgr:
    results = mgr.dict()
    procs = [ctx.Process(target=_run_syncbn, args=(r, 2, port, results))
             for r in range(2)]
    for p in procs:
      p.start()
    for p in procs:
      p.join(120)
      assert p.exitcode == 0
    outs = {r: results[f'out{r}'] for r in range(2)}
    xs = {r: results[f'x{r}'] for r in range(2)}
    pads = {r: results[f'pad{r}'] for r in range(2)}

  # Global moments over BOTH ranks' valid frames.
  mask = torch.cat([(1 - pads[0]).reshape(-1), (1 - pads[1]).reshape(-1)])
  allx = torch.cat([xs[0].reshape(-1, 8), xs[1].reshape(-1, 8)])
  m = mask.unsqueeze(1)
  count = mask.sum()
  mean = (allx * m).sum(0) / count
  var = ((allx - mean) ** 2 * m).sum(0) / count
  want0 = (xs[0] - mean) * torch.rsqrt(var + 1e-3)
  want0 = want0 * (1 - pads[0]).unsqueeze(-1)
  got = outs[0]  # gamma=0 init => scale (1+0)=1, beta 0
  assert torch.allclose(got, want0.to(got.dtype), atol=1e-3), \
      (got - want0).abs().max()


def _run_sharded_emb(rank, world, port, results):
  os.environ['MASTER_ADDR'] = '127.0.0.1'
  os.environ['MASTER_PORT'] = str(port)
  dist.init_process_group('gloo', rank=rank, world_size=world)
  from lingvo_amd.layers import layers as lingvo_layers
  p = lingvo_layers.ShardedEmbeddingLayer.Params().Set(
      name='emb', vocab_size=32, embedding_dim=8, random_seed=11)
  emb = p.Instantiate()
  ids = torch.tensor([[0, 5, 17, 31], [2, 2, 30, 9]])
  out = emb.EmbLookup(emb.theta, ids)
  out.square().sum().backward()
  results[f'out{rank}'] = out.detach()
  results[f'shard{rank}'] = tuple(emb.wm.shape)
  results[f'grad{rank}'] = emb.wm.grad.clone()
  dist.destroy_process_group()


def test_sharded_embedding_matches_unsharded():
  """Vocab-sharded table: same outputs as the unsharded layer with the
  same seed; per-rank table memory = total/W; grads land only on the
  owning shard."""
  port = dist_port(29571)
  ctx = mp.get_context('spawn')
  with ctx.Manager() as mgr:
    results = mgr.dict()
    procs = [ctx.Process(target=_run_sharded_emb,
                         args=(r, 2, port, results)) for r in range(2)]
    for p in procs:
      p.start()
    for p in procs:
      p.join(120)
      assert p.exitcode == 0
    outs = {r: results[f'out{r}'] for r in range(2)}
    shards = {r: results[f'shard{r}'] for r in range(2)}
    grads = {r: results[f'grad{r}'] for r in range(2)}

  assert shards[0] == (16, 8) and shards[1] == (16, 8)
  # Unsharded reference with the same seed.
  from lingvo_amd.layers import layers as lingvo_layers
  ref = lingvo_layers.ShardedEmbeddingLayer.Params().Set(
      name='emb', vocab_size=32, embedding_dim=8,
      random_seed=11).Instantiate()
  ids = torch.tensor([[0, 5, 17, 31], [2, 2, 30, 9]])
  want = ref.EmbLookup(ref.theta, ids)
  assert torch.allclose(outs[0], want.detach(), atol=1e-5)
  assert torch.allclose(outs[0], outs[1], atol=1e-6)
  # Grad for id 17 (row 1 of rank-1 shard) is on rank 1 only.
  assert grads[1][1].abs().sum() > 0
  assert grads[0].shape == (16, 8)
