"""Ring-attention CP tests: CP=2 over gloo == single-process attention."""

import os

from conftest import dist_port
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from lingvo_amd.parallel import context_parallel as cp
from lingvo_amd.ops import flash_attn


def _make_inputs(causal):
  g = torch.Generator().manual_seed(101 if causal else 102)
  B, S, N, H = 2, 16, 2, 8
  q = torch.randn(B, S, N, H, generator=g)
  k = torch.randn(B, S, N, H, generator=g)
  v = torch.randn(B, S, N, H, generator=g)
  klen = torch.tensor([16, 11])
  return q, k, v, klen


def _run_ring(rank, world, port, causal, results):
  os.environ['MASTER_ADDR'] = '127.0.0.1'
  os.environ['MASTER_PORT'] = str(port)
  dist.init_process_group('gloo', rank=rank, world_size=world)
  q, k, v, klen = _make_inputs(causal)
  ql = cp.ShardSequence(q, rank, world).detach().requires_grad_(True)
  kl = cp.ShardSequence(k, rank, world).detach().requires_grad_(True)
  vl = cp.ShardSequence(v, rank, world).detach().requires_grad_(True)
  out = cp.RingAttention(ql, kl, vl, klen=klen, causal=causal)
  out.square().sum().backward()
  results[f'out{rank}'] = out.detach()
  results[f'dq{rank}'] = ql.grad.clone()
  results[f'dk{rank}'] = kl.grad.clone()
  results[f'dv{rank}'] = vl.grad.clone()
  dist.destroy_process_group()


def _check_ring(causal, port):
  ctx = mp.get_context('spawn')
  with ctx.Manager() as mgr:
    results = mgr.dict()
    procs = [ctx.Process(target=_run_ring,
                         args=(r, 2, port, causal, results))
             for r in range(2)]
    for p in procs:
      p.start()
    for p in procs:
      p.join(120)
      assert p.exitcode == 0
    results = dict(results)

  # single-process reference over the full sequence
  q, k, v, klen = _make_inputs(causal)
  q, k, v = (t.requires_grad_(True) for t in (q, k, v))
  ref = flash_attn.flash_attention(q, k, v, klen.to(torch.int32), None,
                                   -1, 0 if causal else -1)
  ref.square().sum().backward()
  L = q.shape[1] // 2
  for r in range(2):
    sl = slice(r * L, (r + 1) * L)
    assert torch.allclose(results[f'out{r}'], ref[:, sl].detach(),
                          atol=1e-4), (causal, r)
    assert torch.allclose(results[f'dq{r}'], q.grad[:, sl], atol=1e-4)
    # dK/dV arrive on the owning rank via the reverse ring
    assert torch.allclose(results[f'dk{r}'], k.grad[:, sl], atol=1e-4)
    assert torch.allclose(results[f'dv{r}'], v.grad[:, sl], atol=1e-4)


def test_ring_attention_full():
  _check_ring(causal=False, port=dist_port(29555))


def test_ring_attention_causal():
  _check_ring(causal=True, port=dist_port(29556))


def _run_cp_layer(rank, world, port, results):
  os.environ['MASTER_ADDR'] = '127.0.0.1'
  os.environ['MASTER_PORT'] = str(port)
  dist.init_process_group('gloo', rank=rank, world_size=world)
  layer = cp.CpMultiHeadedAttention.Params().Set(
      name='cpa', input_dim=32, hidden_dim=32, num_heads=2, causal=True,
      random_seed=7).Instantiate()
  g = torch.Generator().manual_seed(33)
  x = torch.randn(2, 12, 32, generator=g)
  pad = torch.zeros(2, 12)
  pad[1, 9:] = 1.0
  xl = cp.ShardSequence(x, rank, world).detach().requires_grad_(True)
  out = layer.FProp(layer.theta, xl, cp.ShardSequence(pad, rank, world))
  out.square().sum().backward()
  results[f'out{rank}'] = out.detach()
  results[f'dx{rank}'] = xl.grad.clone()
  results[f'dw{rank}'] = layer.qkv_w.grad.clone()
  dist.destroy_process_group()


def test_cp_layer_matches_full():
  ctx = mp.get_context('spawn')
  with ctx.Manager() as mgr:
    results = mgr.dict()
    procs = [ctx.Process(target=_run_cp_layer, args=(r, 2, dist_port(29557), results))
             for r in range(2)]
    for p in procs:
      p.start()
    for p in procs:
      p.join(120)
      assert p.exitcode == 0
    results = dict(results)

  from lingvo_amd.layers import attention as attention_lib
  layer = attention_lib.MultiHeadedAttention.Params().Set(
      name='cpa', input_dim=32, hidden_dim=32, num_heads=2, causal=True,
      random_seed=7).Instantiate()
  g = torch.Generator().manual_seed(33)
  x = torch.randn(2, 12, 32, generator=g, requires_grad=True)
  pad = torch.zeros(2, 12)
  pad[1, 9:] = 1.0
  ref = layer.FProp(layer.theta, x, pad)
  ref.square().sum().backward()
  for r in range(2):
    sl = slice(r * 6, (r + 1) * 6)
    assert torch.allclose(results[f'out{r}'], ref[:, sl].detach(),
                          atol=1e-4), r
    assert torch.allclose(results[f'dx{r}'], x.grad[:, sl], atol=1e-4)
  # replicated weights: SUM of per-rank grads == full grad (finished by
  # the DP all-reduce in training).
  assert torch.allclose(results['dw0'] + results['dw1'],
                        layer.qkv_w.grad, atol=1e-4)


def _run_ulysses(rank, world, port, results):
  os.environ['MASTER_ADDR'] = '127.0.0.1'
  os.environ['MASTER_PORT'] = str(port)
  dist.init_process_group('gloo', rank=rank, world_size=world)
  layer = cp.UlyssesMultiHeadedAttention.Params().Set(
      name='u', input_dim=32, hidden_dim=32, num_heads=4, causal=True,
      rel_pos_bias=True, random_seed=19).Instantiate()
  gb = torch.Generator().manual_seed(77)
  with torch.no_grad():
    layer.rel_bias.copy_(
        torch.randn(layer.rel_bias.shape, generator=gb) * 0.1)
  g = torch.Generator().manual_seed(44)
  x = torch.randn(2, 12, 32, generator=g)
  pad = torch.zeros(2, 12)
  pad[1, 9:] = 1.0
  xl = cp.ShardSequence(x, rank, world).detach().requires_grad_(True)
  out = layer.FProp(layer.theta, xl, cp.ShardSequence(pad, rank, world))
  out.square().sum().backward()
  results[f'out{rank}'] = out.detach()
  results[f'dx{rank}'] = xl.grad.clone()
  results[f'dw{rank}'] = layer.qkv_w.grad.clone()
  results[f'bias{rank}'] = layer.rel_bias.detach().clone()
  dist.destroy_process_group()


def test_ulysses_attention_matches_full():
  ctx = mp.get_context('spawn')
  with ctx.Manager() as mgr:
    results = mgr.dict()
    procs = [ctx.Process(target=_run_ulysses, args=(r, 2, dist_port(29575), results))
             for r in range(2)]
    for p in procs:
      p.start()
    for p in procs:
      p.join(120)
      assert p.exitcode == 0
    results = dict(results)

  from lingvo_amd.layers import attention as attention_lib
  ref = attention_lib.MultiHeadedAttention.Params().Set(
      name='u', input_dim=32, hidden_dim=32, num_heads=4, causal=True,
      rel_pos_bias=True, random_seed=19).Instantiate()
  with torch.no_grad():
    ref.rel_bias.copy_(results['bias0'])
  g = torch.Generator().manual_seed(44)
  x = torch.randn(2, 12, 32, generator=g, requires_grad=True)
  pad = torch.zeros(2, 12)
  pad[1, 9:] = 1.0
  full = ref.FProp(ref.theta, x, pad)
  full.square().sum().backward()
  for r in range(2):
    sl = slice(r * 6, (r + 1) * 6)
    assert torch.allclose(results[f'out{r}'], full[:, sl].detach(),
                          atol=1e-4), r
    assert torch.allclose(results[f'dx{r}'], x.grad[:, sl], atol=1e-4)
  # replicated qkv_w: per-rank grads SUM to the full grad
  assert torch.allclose(results['dw0'] + results['dw1'],
                        ref.qkv_w.grad, atol=1e-4)


def _run_zigzag(rank, world, port, results):
  os.environ['MASTER_ADDR'] = '127.0.0.1'
  os.environ['MASTER_PORT'] = str(port)
  dist.init_process_group('gloo', rank=rank, world_size=world)
  q, k, v, klen = _make_inputs(causal=True)
  ql = cp.ZigzagShard(q, rank, world).detach().requires_grad_(True)
  kl = cp.ZigzagShard(k, rank, world).detach().requires_grad_(True)
  vl = cp.ZigzagShard(v, rank, world).detach().requires_grad_(True)
  out = cp.RingAttentionZigzag(ql, kl, vl, klen=klen, causal=True)
  out.square().sum().backward()
  results[f'out{rank}'] = out.detach()
  results[f'dq{rank}'] = ql.grad.clone()
  results[f'dk{rank}'] = kl.grad.clone()
  dist.destroy_process_group()


def test_zigzag_ring_attention_exact():
  ctx = mp.get_context('spawn')
  with ctx.Manager() as mgr:
    results = mgr.dict()
    procs = [ctx.Process(target=_run_zigzag, args=(r, 2, dist_port(29581), results))
             for r in range(2)]
    for p in procs:
      p.start()
    for p in procs:
      p.join(120)
      assert p.exitcode == 0
    results = dict(results)

  q, k, v, klen = _make_inputs(causal=True)
  q, k, v = (t.requires_grad_(True) for t in (q, k, v))
  ref = flash_attn.flash_attention(q, k, v, klen.to(torch.int32), None,
                                   -1, 0)
  ref.square().sum().backward()
  for r in range(2):
    pos = cp.ZigzagPositions(r, 2, q.shape[1])
    assert torch.allclose(results[f'out{r}'],
                          ref[:, pos].detach(), atol=1e-4), r
    assert torch.allclose(results[f'dq{r}'], q.grad[:, pos], atol=1e-4)
    assert torch.allclose(results[f'dk{r}'], k.grad[:, pos], atol=1e-4)


def _run_ring_gqa(rank, world, port, results):
  os.environ['MASTER_ADDR'] = '127.0.0.1'
  os.environ['MASTER_PORT'] = str(port)
  dist.init_process_group('gloo', rank=rank, world_size=world)
  g = torch.Generator().manual_seed(321)
  B, S, N, NKV, H = 2, 12, 4, 2, 8
  q = torch.randn(B, S, N, H, generator=g)
  k = torch.randn(B, S, NKV, H, generator=g)
  v = torch.randn(B, S, NKV, H, generator=g)
  ql = cp.ShardSequence(q, rank, world).detach().requires_grad_(True)
  kl = cp.ShardSequence(k, rank, world).detach().requires_grad_(True)
  vl = cp.ShardSequence(v, rank, world).detach().requires_grad_(True)
  out = cp.RingAttention(ql, kl, vl, causal=True)
  out.square().sum().backward()
  results[f'out{rank}'] = out.detach()
  results[f'dk{rank}'] = kl.grad.clone()
  dist.destroy_process_group()


def test_ring_attention_gqa():
  """Ring attention with grouped KV heads (NKV < N) stays exact."""
  ctx = mp.get_context('spawn')
  with ctx.Manager() as mgr:
    results = mgr.dict()
    procs = [ctx.Process(target=_run_ring_gqa,
                         args=(r, 2, dist_port(29604), results))
             for r in range(2)]
    for p in procs:
      p.start()
    for p in procs:
      p.join(120)
      assert p.exitcode == 0
    results = dict(results)
  g = torch.Generator().manual_seed(321)
  B, S, N, NKV, H = 2, 12, 4, 2, 8
  q = torch.randn(B, S, N, H, generator=g).requires_grad_(True)
  k = torch.randn(B, S, NKV, H, generator=g).requires_grad_(True)
  v = torch.randn(B, S, NKV, H, generator=g).requires_grad_(True)
  ref = flash_attn.flash_attention(q, k, v, None, None, -1, 0)
  ref.square().sum().backward()
  for r in range(2):
    sl = slice(r * 6, (r + 1) * 6)
    assert torch.allclose(results[f'out{r}'], ref[:, sl].detach(),
                          atol=1e-4)
    assert torch.allclose(results[f'dk{r}'], k.grad[:, sl], atol=1e-4)
