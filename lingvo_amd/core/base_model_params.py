"""Model-params base classes selected via the registry.

Reference: lingvo/core/base_model_params.py:45-149. Subclasses override
`Train()/Dev()/Test()` (input params) and `Task()`; `Model()` wraps the
task into a SingleTaskModel.
"""

from __future__ import annotations

from typing import Optional

from lingvo_amd.core.hyperparams import InstantiableParams


class DatasetError(Exception):
  pass


class _BaseModelParams:

  def GetDatasetParams(self, dataset: str):
    fn = getattr(self, dataset, None)
    if fn is None or not callable(fn):
      raise DatasetError(
          f'Dataset {dataset!r} not defined on {type(self).__name__}')
    return fn()


class SingleTaskModelParams(_BaseModelParams):
  """Defines Train/Dev/Test inputs + a single Task."""

  def Train(self):
    return None

  def Dev(self):
    return None

  def Test(self):
    return None

  def Task(self) -> InstantiableParams:
    raise NotImplementedError('Subclass must implement Task()')

  def ProgramSchedule(self):
    return None

  def Model(self) -> InstantiableParams:
    from lingvo_amd.core.base_model import SingleTaskModel
    task_p = self.Task()
    return SingleTaskModel.Params(task_p)
