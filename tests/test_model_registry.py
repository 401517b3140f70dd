"""Registry-wide smoke tests (reference models_test.py /
model_import_test.py capability): every registered model's params
resolve, serialize, and (for small configs) instantiate and step."""

import pytest
import torch

from lingvo_amd.core import registry

ALL_KEYS = sorted(registry.GetAllRegisteredClasses())

# Configs small enough to instantiate + train-step on CPU.
SMALL = ['image.mnist.LeNet5', 'punctuator.codelab.RNMTModel',
         'milan.cxc.ImageTextDualEncoder']


def test_expected_registry_contents():
  for key in ['image.mnist.LeNet5',
              'asr.librispeech.Librispeech960WpmConformerL',
              'lm.one_billion_wds.OneBWdsTransformerLm',
              'lm.synthetic_packed_input.MoELm64E',
              'mt.wmt14_en_de.WmtEnDeTransformerBig',
              'punctuator.codelab.RNMTModel',
              'milan.cxc.ImageTextDualEncoder']:
    assert key in ALL_KEYS, key


@pytest.mark.parametrize('key', ALL_KEYS)
def test_params_resolve_and_serialize(key):
  for dataset in ('Train', 'Dev', 'Test'):
    model_p = registry.GetParams(key, dataset)
    assert model_p.task is not None
    text = model_p.ToText()
    assert 'task.name' in text


@pytest.mark.parametrize('key', SMALL)
def test_small_models_train_step(key):
  model_p = registry.GetParams(key, 'Train')
  model_p.task.fprop_dtype = torch.float32
  model_p.task.train.bf16_weights = False
  model_p.task.random_seed = 11
  if 'input' in model_p and model_p.input is not None:
    model_p.input.batch_size = 4
  model = model_p.Instantiate()
  task = model.GetTask()
  m = task.TrainStep(task.GetInputBatch())
  assert torch.isfinite(m['loss'][0])


def test_las_encoder_small_train_step():
  """LAS (conv + biLSTM) ASR variant instantiates and steps on CPU."""
  model_p = registry.GetParams('asr.librispeech.Librispeech960Base',
                               'Train')
  model_p.task.fprop_dtype = torch.float32
  model_p.task.train.bf16_weights = False
  model_p.task.random_seed = 5
  model_p.task.encoder.Set(model_dim=64, num_lstm_layers=2,
                           subsample_channels=8)
  model_p.task.decoder.Set(rnn_cell_dim=32, source_dim=64, emb_dim=16,
                           vocab_size=32)
  model_p.input.Set(batch_size=2, frame_len=32, target_len=6,
                    vocab_size=32)
  model = model_p.Instantiate()
  task = model.GetTask()
  m = task.TrainStep(task.GetInputBatch())
  assert torch.isfinite(m['loss'][0])


def test_dense_lm_family_and_waymo_registered():
  keys = registry.GetAllRegisteredClasses()
  for k in ['lm.synthetic_packed_input.DenseLm8B',
            'lm.synthetic_packed_input.DenseLm128B8x8',
            'mt.wmt14_en_de.WmtEnDeTransformerSmall',
            'asr.librispeech.Librispeech960Base',
            'car.waymo.WaymoPillars']:
    assert k in keys, k
  # the 128B config carries TP sharding annotations
  p = registry.GetParams('lm.synthetic_packed_input.DenseLm128B8x8',
                         'Train')
  assert p.task.lm.weight_split_dims_mapping == [-1, 0]


def test_waymo_pillars_small_train_step():
  model_p = registry.GetParams('car.waymo.WaymoPillars', 'Train')
  model_p.task.random_seed = 2
  model_p.task.Set(grid_size=16, backbone_channels=[8, 16],
                   point_feat_dim=8)
  model_p.input.Set(batch_size=2, num_points=128)
  model = model_p.Instantiate()
  task = model.GetTask()
  m = task.TrainStep(task.GetInputBatch())
  assert torch.isfinite(m['loss'][0])


def test_base_model_params_contract():
  from lingvo_amd.core import base_model_params as bmp
  from lingvo_amd.models import mnist as mnist_model

  class MyParams(bmp.SingleTaskModelParams):

    def Train(self):
      return mnist_model.FakeMnistData.Params().Set(batch_size=4,
                                                    name='train')

    def Task(self):
      p = mnist_model.ModelV1.Params().Set(
          name='m', hidden_dim=8, filter_shapes=[(3, 3, 1, 2)])
      p.softmax.num_classes = 10
      return p

  mp = MyParams()
  assert mp.GetDatasetParams('Train').batch_size == 4
  import pytest
  with pytest.raises(bmp.DatasetError):
    mp.GetDatasetParams('Nope')
  model_p = mp.Model()
  model = model_p.Instantiate()
  assert model.GetTask().p.name == 'm'
