"""Dataset introspection (reference lingvo/datasets.py): lists the
dataset methods a registered model-params class defines."""

from __future__ import annotations

import inspect
from typing import List

from lingvo_amd.core import registry
from lingvo_amd.core.base_model_params import _BaseModelParams


def GetDatasets(cls_or_key) -> List[str]:
  """Returns dataset method names (Train/Dev/Test/...) on the params
  class, excluding the non-dataset API methods."""
  cls = (registry.GetClass(cls_or_key)
         if isinstance(cls_or_key, str) else cls_or_key)
  exclude = {'Task', 'Model', 'ProgramSchedule', 'GetDatasetParams'}
  out = []
  for name, member in inspect.getmembers(cls, inspect.isfunction):
    if name.startswith('_') or name in exclude:
      continue
    out.append(name)
  return sorted(out)
