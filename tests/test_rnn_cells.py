"""RNN cell variants (reference lingvo/core/rnn_cell.py): zoneout,
CIFG, grouped/shuffled, double-projection, conv LSTM."""
import torch

from lingvo_amd.core.nested_map import NestedMap
from lingvo_amd.layers import rnn_cell


def _step(cell, batch, d_in, padding=None):
  s0 = cell.InitState(batch, 'cpu', torch.float32)
  if padding is None:
    padding = torch.zeros(batch, 1)
  ins = NestedMap(act=torch.randn(batch, d_in), padding=padding)
  return s0, cell.FProp(cell.theta, s0, ins)


def test_zoneout_eval_expectation():
  torch.manual_seed(0)
  p = rnn_cell.LSTMCellSimple.Params().Set(
      name='c', num_input_nodes=6, num_output_nodes=4, zo_prob=0.25)
  cell = p.Instantiate().eval()
  p0 = p.Copy().Set(name='c0', zo_prob=0.0)
  cell0 = p0.Instantiate()
  for (a, _), (b, _) in zip(cell.named_parameters(),
                            cell0.named_parameters()):
    assert a == b
  cell0.load_state_dict(cell.state_dict())
  cell0.eval()
  s0 = cell.InitState(3, 'cpu', torch.float32)
  ins = NestedMap(act=torch.randn(3, 6), padding=torch.zeros(3, 1))
  s1 = cell.FProp(cell.theta, s0, ins)
  s1_ref = cell0.FProp(cell0.theta, s0, ins)
  # Zero initial state: eval zoneout mixes toward 0 by zo_prob exactly.
  assert torch.allclose(s1.c, 0.75 * s1_ref.c, atol=1e-6)
  assert torch.allclose(s1.m, 0.75 * s1_ref.m, atol=1e-6)


def test_zoneout_train_keeps_padded_state():
  torch.manual_seed(0)
  p = rnn_cell.LSTMCellSimple.Params().Set(
      name='c', num_input_nodes=6, num_output_nodes=4, zo_prob=0.5)
  cell = p.Instantiate()
  s0 = cell.InitState(2, 'cpu', torch.float32)
  s0.c = torch.randn_like(s0.c)
  s0.m = torch.randn_like(s0.m)
  pad = torch.tensor([[0.0], [1.0]])
  ins = NestedMap(act=torch.randn(2, 6), padding=pad)
  s1 = cell.FProp(cell.theta, s0, ins)
  assert torch.equal(s1.c[1], s0.c[1])
  assert torch.equal(s1.m[1], s0.m[1])
  # Train-mode zoneout keeps each element either previous or current.
  p0 = p.Copy().Set(name='c0', zo_prob=0.0)
  cell0 = p0.Instantiate()
  cell0.load_state_dict(cell.state_dict())
  s1_ref = cell0.FProp(cell0.theta, s0, ins)
  keep = torch.isclose(s1.c[0], s0.c[0], atol=1e-6)
  cur = torch.isclose(s1.c[0], s1_ref.c[0], atol=1e-6)
  assert torch.all(keep | cur)


def test_no_lstm_bias():
  p = rnn_cell.LSTMCellSimple.Params().Set(
      name='c', num_input_nodes=6, num_output_nodes=4,
      enable_lstm_bias=False)
  cell = p.Instantiate()
  assert not any(n == 'b' for n, _ in cell.named_parameters())
  _step(cell, 3, 6)


def test_weight_normalized_lstm():
  torch.manual_seed(1)
  p = rnn_cell.WeightNormalizedLSTMCellSimple.Params().Set(
      name='c', num_input_nodes=6, num_output_nodes=4)
  cell = p.Instantiate()
  s0, s1 = _step(cell, 3, 6)
  assert s1.m.shape == (3, 4)
  # With g == ||v||, matches the unnormalized cell exactly.
  with torch.no_grad():
    cell.vars.wm_g.copy_(cell.vars.wm.norm(dim=0))
  p0 = rnn_cell.LSTMCellSimple.Params().Set(
      name='c0', num_input_nodes=6, num_output_nodes=4)
  cell0 = p0.Instantiate()
  with torch.no_grad():
    cell0.vars.wm.copy_(cell.vars.wm)
    cell0.vars.b.copy_(cell.vars.b)
  ins = NestedMap(act=torch.randn(3, 6), padding=torch.zeros(3, 1))
  s0 = cell.InitState(3, 'cpu', torch.float32)
  out_a = cell.FProp(cell.theta, s0, ins)
  out_b = cell0.FProp(cell0.theta, s0, ins)
  assert torch.allclose(out_a.m, out_b.m, atol=1e-5)


def test_grouped_lstm_shapes_and_shuffle():
  torch.manual_seed(0)
  p = rnn_cell.LSTMCellGrouped.Params().Set(
      name="c", num_input_nodes=9, num_output_nodes=12,
      num_hidden_nodes=12, num_groups=3, num_shuffle_shards=2)
  cell = p.Instantiate()
  s0, s1 = _step(cell, 4, 9)
  assert len(s1.groups) == 3
  out = cell.GetOutput(s1)
  assert out.shape == (4, 12)
  # Shuffle is a permutation: every shard appears exactly once.
  shards = list(range(6))
  shuffled = cell._ShuffleShards(shards)
  assert sorted(shuffled) == shards and shuffled != shards
  # Reference example (3 groups x 2 shards): g0 gets [0_0, 1_1].
  assert shuffled[0] == 0 and shuffled[1] == 3


def test_double_projection_lstm():
  torch.manual_seed(0)
  p = rnn_cell.DoubleProjectionLSTMCell.Params().Set(
      name='c', num_input_nodes=6, num_output_nodes=4,
      num_input_hidden_nodes=5, num_hidden_nodes=7)
  cell = p.Instantiate()
  s0, s1 = _step(cell, 3, 6)
  assert s1.c.shape == (3, 7) and s1.m.shape == (3, 4)
  # Padded rows carry state.
  pad = torch.ones(3, 1)
  s0b = cell.InitState(3, 'cpu', torch.float32)
  s0b.c = torch.randn_like(s0b.c)
  s0b.m = torch.randn_like(s0b.m)
  ins = NestedMap(act=torch.randn(3, 6), padding=pad)
  s1b = cell.FProp(cell.theta, s0b, ins)
  assert torch.equal(s1b.c, s0b.c) and torch.equal(s1b.m, s0b.m)


def test_conv_lstm_cell():
  torch.manual_seed(0)
  p = rnn_cell.ConvLSTMCell.Params().Set(
      name='c', inputs_shape=[None, 5, 6, 3],
      cell_shape=[None, 5, 6, 4], filter_shape=[3, 3])
  cell = p.Instantiate()
  s0 = cell.InitState(2, 'cpu', torch.float32)
  s0.c = torch.randn_like(s0.c)
  s0.m = torch.randn_like(s0.m)
  pad = torch.tensor([[0.0], [1.0]])
  ins = NestedMap(act=torch.randn(2, 5, 6, 3), padding=pad)
  s1 = cell.FProp(cell.theta, s0, ins)
  assert s1.m.shape == (2, 5, 6, 4)
  assert torch.equal(s1.c[1], s0.c[1]) and torch.equal(s1.m[1], s0.m[1])
  assert not torch.equal(s1.c[0], s0.c[0])


def test_quantized_lstm_cell_clipping_schedule():
  torch.manual_seed(0)
  p = rnn_cell.QuantizedLSTMCell.Params().Set(
      name='q', num_input_nodes=6, num_output_nodes=4)
  p.cc_schedule.Set(start_step=0, end_step=100, start_cap=8.0,
                    end_cap=1.0)
  cell = p.Instantiate()
  assert cell.cc_schedule.Value() == 8.0
  cell.PostTrainingStepUpdate(50)
  assert abs(cell.cc_schedule.Value() - 4.5) < 1e-6
  cell.PostTrainingStepUpdate(1000)
  assert cell.cc_schedule.Value() == 1.0
  s0 = cell.InitState(3, 'cpu', torch.float32)
  s0.c = torch.full((3, 4), 100.0)
  ins = NestedMap(act=torch.randn(3, 6), padding=torch.zeros(3, 1))
  s1 = cell.FProp(cell.theta, s0, ins)
  assert s1.c.abs().max() <= 1.0 + 1e-6
  assert s1.m.abs().max() <= 1.0 + 1e-6
