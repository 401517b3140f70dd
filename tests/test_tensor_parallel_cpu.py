"""Tensor-parallel tests: TP=2 over gloo == single-process full layer."""

import os

import pytest
from conftest import dist_port
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from lingvo_amd.parallel import tensor_parallel as tp


def _ffn_params(seed=9, tp_group=None):
  return tp.TpFeedForwardLayer.Params().Set(
      name='ffn', input_dim=16, hidden_dim=32, activation='RELU',
      random_seed=seed, tp_group=tp_group)


def _run_tp(rank, world, port, results):
  os.environ['MASTER_ADDR'] = '127.0.0.1'
  os.environ['MASTER_PORT'] = str(port)
  dist.init_process_group('gloo', rank=rank, world_size=world)
  layer = _ffn_params().Instantiate()
  g = torch.Generator().manual_seed(77)
  x = torch.randn(2, 6, 16, generator=g, requires_grad=True)
  out = layer.FProp(layer.theta, x)
  out.sum().backward()
  results[f'out{rank}'] = out.detach()
  results[f'wi_grad{rank}'] = layer.wi.w.grad.clone()
  results[f'dx{rank}'] = x.grad.clone()
  dist.destroy_process_group()


def test_tp2_matches_single_process():
  ctx = mp.get_context('spawn')
  with ctx.Manager() as mgr:
    results = mgr.dict()
    procs = [ctx.Process(target=_run_tp, args=(r, 2, dist_port(29535), results))
             for r in range(2)]
    for p in procs:
      p.start()
    for p in procs:
      p.join(120)
      assert p.exitcode == 0
    out0, out1 = results['out0'], results['out1']
    dx0, dx1 = results['dx0'], results['dx1']
    wi_grad0, wi_grad1 = results['wi_grad0'], results['wi_grad1']

  # TP=1 reference (no dist initialized).
  layer = _ffn_params().Instantiate()
  g = torch.Generator().manual_seed(77)
  x = torch.randn(2, 6, 16, generator=g, requires_grad=True)
  ref = layer.FProp(layer.theta, x)
  ref.sum().backward()

  assert torch.allclose(out0, out1, atol=1e-5)
  assert torch.allclose(out0, ref.detach(), atol=1e-5), \
      (out0 - ref.detach()).abs().max()
  assert torch.allclose(dx0, x.grad, atol=1e-5)
  # rank 0's wi shard grad == first half of full wi grad
  assert torch.allclose(wi_grad0, layer.wi.w.grad[:, :16], atol=1e-5)
  assert torch.allclose(wi_grad1, layer.wi.w.grad[:, 16:], atol=1e-5)


def test_shard_planner_rewrites_ffn():
  from lingvo_amd.layers import transformer as transformer_lib
  sp = transformer_lib.StackedTransformerLayers.Params().Set(
      name='s', model_dim=16, num_layers=1, num_heads=1, hidden_dim=32)
  sp.transformer_tpl.input_dim = 16
  sp.transformer_tpl.tr_fflayer_tpl.input_dim = 16
  sp.transformer_tpl.tr_fflayer_tpl.hidden_dim = 32
  tp.ShardTransformerStackForTp(sp)
  assert sp.transformer_tpl.tr_fflayer_tpl.cls is tp.TpFeedForwardLayer
  # instantiates and runs (TP world 1)
  sp.random_seed = 3
  stack = sp.Instantiate()
  x = torch.randn(2, 4, 16)
  out = stack.FProp(stack.theta, x, torch.zeros(2, 4))
  assert out.shape == x.shape


def _tp_attn_params(seed=13):
  return tp.TpMultiHeadedAttention.Params().Set(
      name='tpa', input_dim=128, hidden_dim=128, num_heads=2, causal=True,
      rel_pos_bias=True, random_seed=seed)


def _run_tp_attn(rank, world, port, results):
  os.environ['MASTER_ADDR'] = '127.0.0.1'
  os.environ['MASTER_PORT'] = str(port)
  dist.init_process_group('gloo', rank=rank, world_size=world)
  layer = _tp_attn_params().Instantiate()
  g = torch.Generator().manual_seed(55)
  x = torch.randn(2, 8, 128, generator=g, requires_grad=True)
  out = layer.FProp(layer.theta, x, torch.zeros(2, 8))
  out.sum().backward()
  results[f'out{rank}'] = out.detach()
  results[f'dx{rank}'] = x.grad.clone()
  results[f'qkv_shape{rank}'] = tuple(layer.qkv_w.shape)
  dist.destroy_process_group()


def test_tp_attention_matches_single_process():
  ctx = mp.get_context('spawn')
  with ctx.Manager() as mgr:
    results = mgr.dict()
    procs = [ctx.Process(target=_run_tp_attn, args=(r, 2, dist_port(29545), results))
             for r in range(2)]
    for p in procs:
      p.start()
    for p in procs:
      p.join(120)
      assert p.exitcode == 0
    out0, out1 = results['out0'], results['out1']
    dx0 = results['dx0']
    shard_shape = results['qkv_shape0']

  # each rank holds 1 of 2 heads: qkv shard [(1+2)*64] cols
  assert shard_shape == (128, 3 * 64)
  # TP=1 reference
  layer = _tp_attn_params().Instantiate()
  g = torch.Generator().manual_seed(55)
  x = torch.randn(2, 8, 128, generator=g, requires_grad=True)
  ref = layer.FProp(layer.theta, x, torch.zeros(2, 8))
  ref.sum().backward()
  assert torch.allclose(out0, out1, atol=1e-5)
  assert torch.allclose(out0, ref.detach(), atol=1e-4), \
      (out0 - ref.detach()).abs().max()
  assert torch.allclose(dx0, x.grad, atol=1e-4)


def test_shard_attention_planner():
  from lingvo_amd.layers import transformer as transformer_lib
  sp = transformer_lib.StackedTransformerLayers.Params().Set(
      name='s', model_dim=128, num_layers=1, num_heads=2, hidden_dim=256,
      random_seed=2)
  tp.ShardAttentionForTp(sp)
  tp.ShardTransformerStackForTp(sp)
  sp.transformer_tpl.tr_fflayer_tpl.input_dim = 128
  sp.transformer_tpl.tr_fflayer_tpl.hidden_dim = 256
  stack = sp.Instantiate()
  x = torch.randn(2, 6, 128)
  out = stack.FProp(stack.theta, x, torch.zeros(2, 6))
  assert out.shape == x.shape


def test_lower_sharding_annotations():
  """Annotated stacks rewrite to TP classes; unannotated stay."""
  from lingvo_amd.layers import transformer as transformer_lib
  sp = transformer_lib.StackedTransformerLayers.Params().Set(
      name='s', model_dim=16, num_layers=1, num_heads=1, hidden_dim=32,
      random_seed=2)
  sp.transformer_tpl.tr_fflayer_tpl.Set(
      input_dim=16, hidden_dim=32, weight_split_dims_mapping=[-1, 0])
  outer = tp.TpFeedForwardLayer.Params()  # host tree containing the stack
  from lingvo_amd.core.hyperparams import Params
  host = Params()
  host.Define('stack', sp, 'nested')
  tp.LowerShardingAnnotations(host)
  assert sp.transformer_tpl.tr_fflayer_tpl.cls is tp.TpFeedForwardLayer
  assert sp.transformer_tpl.tr_atten_tpl.atten_tpl.cls is \
      tp.TpMultiHeadedAttention

  sp2 = transformer_lib.StackedTransformerLayers.Params().Set(
      name='s2', model_dim=16, num_layers=1, num_heads=1, hidden_dim=32)
  tp.LowerShardingAnnotations(sp2)
  assert sp2.transformer_tpl.tr_fflayer_tpl.cls is not \
      tp.TpFeedForwardLayer


def _run_tpdp(rank, world, port, results):
  os.environ['MASTER_ADDR'] = '127.0.0.1'
  os.environ['MASTER_PORT'] = str(port)
  dist.init_process_group('gloo', rank=rank, world_size=world)
  from lingvo_amd.parallel.topology import TpDpTopology
  topo = TpDpTopology(tp_degree=2)
  layer = _ffn_params(tp_group=topo.tp_group).Instantiate()
  sync = topo.MakeGradSync(layer)
  g = torch.Generator().manual_seed(600 + topo.dp_idx)  # per-replica data
  x = torch.randn(2, 6, 16, generator=g)
  out = layer.FProp(layer.theta, x)
  out.sum().backward()
  sync.Finalize()
  results[f'coord{rank}'] = (topo.tp_idx, topo.dp_idx)
  results[f'wi{rank}'] = layer.wi.w.grad.clone()
  results[f'out{rank}'] = out.detach()
  dist.destroy_process_group()


def test_tpdp_grid_2x2():
  """TP=2 x DP=2: shard grads DP-average to the single-process ref."""
  ctx = mp.get_context('spawn')
  with ctx.Manager() as mgr:
    results = mgr.dict()
    procs = [ctx.Process(target=_run_tpdp,
                         args=(r, 4, dist_port(29591), results))
             for r in range(4)]
    for p in procs:
      p.start()
    for p in procs:
      p.join(180)
      assert p.exitcode == 0
    results = dict(results)
  coords = {results[f'coord{r}']: r for r in range(4)}
  assert set(coords) == {(0, 0), (0, 1), (1, 0), (1, 1)}

  # single-process reference over BOTH replicas' data, averaged
  layer = _ffn_params().Instantiate()
  for dp_idx in range(2):
    g = torch.Generator().manual_seed(600 + dp_idx)
    x = torch.randn(2, 6, 16, generator=g)
    out = layer.FProp(layer.theta, x)
    (out.sum() / 2).backward()
    # replicas agree on their outputs with the full layer
    for t in range(2):
      r = coords[(t, dp_idx)]
      assert torch.allclose(results[f'out{r}'], out.detach(), atol=1e-5)
  # each tp shard's DP-averaged grad == matching slice of full grad
  for t in range(2):
    sl = slice(t * 16, (t + 1) * 16)
    for dp_idx in range(2):
      r = coords[(t, dp_idx)]
      assert torch.allclose(results[f'wi{r}'],
                            layer.wi.w.grad[:, sl], atol=1e-5), (t, dp_idx)


def _run_tp_lm_train(rank, world, port, results):
  os.environ['MASTER_ADDR'] = '127.0.0.1'
  os.environ['MASTER_PORT'] = str(port)
  dist.init_process_group('gloo', rank=rank, world_size=world)
  from lingvo_amd.core.base_model import SingleTaskModel
  from lingvo_amd.models import lm as lm_model
  task_p = lm_model.LanguageModel.Params().Set(name='lm', random_seed=7)
  task_p.lm = lm_model.TransformerLm.Params().Set(
      vocab_size=32, model_dim=16, num_layers=1, num_heads=2,
      hidden_dim=32, dropout_prob=0.0,
      weight_split_dims_mapping=[-1, 0])  # annotate for TP
  input_p = lm_model.SyntheticLmInput.Params().Set(
      name='in', batch_size=4, seq_len=8, vocab_size=32)
  model_p = SingleTaskModel.Params().Set(name='m', task=task_p,
                                         input=input_p)
  tp.LowerShardingAnnotations(model_p)
  model = model_p.Instantiate()
  task = model.GetTask()
  losses = []
  for _ in range(2):
    m = task.TrainStep(task.GetInputBatch())
    losses.append(float(m['loss'][0]))
  results[f'loss{rank}'] = losses
  # TP-sharded FFN weight differs per rank; replicated emb identical
  results[f'emb{rank}'] = task.lm.softmax.linear_w.detach().clone()
  dist.destroy_process_group()


def test_tp_lm_end_to_end_training():
  """Annotated LM lowers to TP layers and trains at TP=2: identical
  losses on both ranks (replicated math + sharded collectives)."""
  ctx = mp.get_context('spawn')
  with ctx.Manager() as mgr:
    results = mgr.dict()
    procs = [ctx.Process(target=_run_tp_lm_train,
                         args=(r, 2, dist_port(29593), results))
             for r in range(2)]
    for p in procs:
      p.start()
    for p in procs:
      p.join(180)
      assert p.exitcode == 0
    results = dict(results)
  assert results['loss0'] == results['loss1']
  assert all(l == l for l in results['loss0'])
  # replicated params stay bit-identical across ranks after training
  assert torch.equal(results['emb0'], results['emb1'])
