"""Registered MNIST model params (reference lingvo/tasks/image/params/mnist.py)."""

from __future__ import annotations

from lingvo_amd.core import learner as learner_lib
from lingvo_amd.core import optimizer as optimizer_lib
from lingvo_amd.core import registry
from lingvo_amd.core import schedule as schedule_lib
from lingvo_amd.core.base_model_params import SingleTaskModelParams
from lingvo_amd.models import mnist as mnist_model


@registry.RegisterSingleTaskModel
class LeNet5(SingleTaskModelParams):
  """LeNet-5-ish conv net on (fake) MNIST — BASELINE config 1."""

  BATCH_SIZE = 50

  def Train(self):
    return mnist_model.FakeMnistData.Params().Set(
        name='train', batch_size=self.BATCH_SIZE, num_samples=60000)

  def Dev(self):
    return self.Test()

  def Test(self):
    return mnist_model.FakeMnistData.Params().Set(
        name='test', batch_size=self.BATCH_SIZE, num_samples=10000)

  def Task(self):
    p = mnist_model.ModelV1.Params().Set(
        name='mnist',
        filter_shapes=[(5, 5, 1, 20), (5, 5, 20, 50)],
        window_shape=(2, 2),
        hidden_dim=300)
    p.softmax.num_classes = 10
    p.train.learner = learner_lib.Learner.Params().Set(
        learning_rate=0.001,
        optimizer=optimizer_lib.Adam.Params(),
        lr_schedule=schedule_lib.Constant.Params(),
        clip_gradient_norm_to_value=5.0)
    return p
