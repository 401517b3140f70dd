"""Model registry: class-key -> model-params class.

Capability of the reference's model_registry (lingvo/model_registry.py:70-388):
`@RegisterSingleTaskModel`, `GetClass(key)`, `GetParams(key, dataset)`.
Keys are `<pkg>.<module>.<ClassName>` with the leading
`lingvo_amd.models.params.` elided, e.g. `image.mnist.LeNet5`.
"""

from __future__ import annotations

import importlib
from typing import Dict, List, Optional, Type

_MODEL_REGISTRY: Dict[str, type] = {}

# Modules auto-imported by ImportAllParams so the registry is populated on
# demand (reference lingvo/model_imports.py).
_PARAM_MODULES = [
    'lingvo_amd.models.params.image.mnist',
    'lingvo_amd.models.params.lm.one_billion_wds',
    'lingvo_amd.models.params.lm.synthetic_packed_input',
    'lingvo_amd.models.params.asr.librispeech',
    'lingvo_amd.models.params.mt.wmt14_en_de',
    'lingvo_amd.models.params.mt.wmtm16_en_de',
    'lingvo_amd.models.params.punctuator.codelab',
    'lingvo_amd.models.params.milan.cxc',
    'lingvo_amd.models.params.car.kitti',
    'lingvo_amd.models.params.car.waymo',
]


def _KeyFromClass(cls: type) -> str:
  module = cls.__module__
  for prefix in ('lingvo_amd.models.params.', 'lingvo_amd.models.'):
    if module.startswith(prefix):
      module = module[len(prefix):]
      break
  return f'{module}.{cls.__name__}'


def RegisterSingleTaskModel(cls: type) -> type:
  """Class decorator registering a SingleTaskModelParams subclass."""
  key = _KeyFromClass(cls)
  if key in _MODEL_REGISTRY and _MODEL_REGISTRY[key] is not cls:
    raise ValueError(f'Duplicate model registration: {key}')
  _MODEL_REGISTRY[key] = cls
  cls._registry_key = key
  return cls


# Multi-task params share the same registration machinery.
RegisterMultiTaskModel = RegisterSingleTaskModel


def ImportAllParams() -> None:
  for mod in _PARAM_MODULES:
    try:
      importlib.import_module(mod)
    except ImportError:
      pass


def GetAllRegisteredClasses() -> Dict[str, type]:
  ImportAllParams()
  return dict(_MODEL_REGISTRY)


def GetClass(key: str) -> type:
  if key not in _MODEL_REGISTRY:
    ImportAllParams()
  if key not in _MODEL_REGISTRY:
    raise LookupError(
        f'Model {key!r} not registered. Known: {sorted(_MODEL_REGISTRY)}')
  return _MODEL_REGISTRY[key]


def GetParams(key: str, dataset: str = 'Train'):
  """Returns fully-resolved model Params for (registry key, dataset)."""
  cls = GetClass(key)
  inst = cls()
  model_p = inst.Model()
  input_p = inst.GetDatasetParams(dataset)
  if input_p is not None:
    model_p.input = input_p
  model_p.model_key = key
  return model_p
