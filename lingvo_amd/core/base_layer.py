"""BaseLayer: root of every layer and model.

Re-creates the reference's BaseLayer contract (lingvo/core/base_layer.py:204)
on top of torch.nn.Module: `Params()` classmethod, `CreateVariable`,
`CreateChild(ren)`, the `theta` NestedMap view of parameters (cast to
fprop_dtype), and the pure-function `FProp(theta, *args)` forward contract.
SPMD-style sharding annotations (`device_mesh`,
`weight_split_dims_mapping`, `activation_split_dims_mapping`,
base_layer.py:262-280 in the reference) are kept as params so model code is
annotated identically; the parallel/ planner lowers them to explicit RCCL
collectives.
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional, Sequence, Union

import torch
from torch import nn

from lingvo_amd.core import py_utils
from lingvo_amd.core.hyperparams import InstantiableParams, Params
from lingvo_amd.core.nested_map import NestedMap


class Accumulator:
  """A non-trainable per-step accumulator (reference base_layer.py:40)."""

  def __init__(self, name: str, default_value: torch.Tensor):
    self.name = name
    self._default = default_value.clone()
    self._value = default_value.clone()

  def GetValue(self) -> torch.Tensor:
    return self._value

  def SetValue(self, value: torch.Tensor) -> None:
    self._value = value

  def Reset(self) -> None:
    self._value = self._default.clone()


class BaseLayer(nn.Module):
  """Base class for all layers and models.

  Subclasses override `Params()` (adding their hyperparameters) and
  `FProp(theta, ...)`. Construction happens via
  `MyLayer.Params().Set(...).Instantiate()`.
  """

  @classmethod
  def Params(cls) -> InstantiableParams:
    p = InstantiableParams(cls)
    p.Define('name', '', 'Layer name.')
    p.Define('dtype', torch.float32, 'Variable dtype (master weights).')
    p.Define('fprop_dtype', None,
             'Activation/compute dtype; None means dtype. Set bf16 for '
             'mixed-precision: master weights stay fp32, theta is cast.')
    p.Define('params_init', py_utils.WeightInit.Xavier(1.0),
             'Default weight initialization spec.')
    p.Define('random_seed', None,
             'Seed for deterministic variable init; None derives one from '
             'the layer name hash.')
    p.Define('skip_lp_regularization', None,
             'If True, exclude vars from L1/L2 regularization.')
    # Sharding annotation surface (lowered by lingvo_amd.parallel planner).
    p.Define('device_mesh', None, 'Optional device mesh np.ndarray.')
    p.Define('weight_split_dims_mapping', None,
             'Mesh-dim mapping per weight dim (TP annotation).')
    p.Define('activation_split_dims_mapping', None,
             'Mesh-dim mapping per activation dim.')
    return p

  def __init__(self, params: InstantiableParams):
    super().__init__()
    assert params.cls is type(self), (
        f'Params.cls {params.cls} != {type(self)}')
    self._params = params.Copy()
    self._params.Freeze()
    self._children_names: List[str] = []
    self._var_specs: Dict[str, Params] = {}
    self._accumulators: Dict[str, Accumulator] = {}
    self._created_variables = False

  # ---- identity ---------------------------------------------------------
  @property
  def params(self) -> InstantiableParams:
    return self._params

  @property
  def p(self) -> InstantiableParams:
    return self._params

  @property
  def do_eval(self) -> bool:
    return not self.training

  @property
  def layer_name(self) -> str:
    return self._params.name or type(self).__name__

  def _InitGenerator(self, var_name: str) -> torch.Generator:
    import zlib
    seed = self._params.random_seed
    if seed is None:
      seed = 1234
    # Stable cross-process hash (Python's hash() is salted per process,
    # which would give DP ranks different initial weights).
    name_hash = zlib.crc32(f'{self.layer_name}/{var_name}'.encode())
    mixed = (name_hash ^ (seed * 2654435761)) & 0x7FFFFFFFFFFFFFFF
    g = torch.Generator()
    g.manual_seed(mixed)
    return g

  # ---- variable / child creation ---------------------------------------
  def CreateVariable(self, name: str, var_params: Params,
                     trainable: bool = True) -> None:
    """Creates an nn.Parameter described by WeightParams and registers it.

    The tensor is initialized deterministically from
    (params.random_seed, layer name, var name).
    """
    if hasattr(self, name):
      raise ValueError(f'Variable {name!r} already exists on '
                       f'{self.layer_name}')
    g = self._InitGenerator(name)
    value = py_utils.InitWeight(var_params.shape, var_params.init, g,
                                var_params.dtype)
    if trainable:
      param = nn.Parameter(value)
    else:
      param = nn.Parameter(value, requires_grad=False)
    if self._params.skip_lp_regularization:
      param._skip_lp_regularization = True  # consumed by Learner
    self.register_parameter(name, param)
    self._var_specs[name] = var_params

  def CreateChild(self, name: str, child_params: InstantiableParams) -> None:
    """Instantiates a sub-layer and registers it as a child module."""
    if hasattr(self, name) and not isinstance(getattr(self, name), nn.Module):
      raise ValueError(f'Child {name!r} collides on {self.layer_name}')
    cp = child_params.Copy()
    if not cp.name:
      cp.name = name
    self._PropagateDtypes(cp)
    child = cp.Instantiate()
    self.add_module(name, child)
    self._children_names.append(name)

  def AddChild(self, name: str, child: 'BaseLayer',
               replace: bool = False) -> None:
    """Registers an ALREADY-INSTANTIATED layer as a child (module
    sharing across tasks — reference multitask_model.py:42 AddChild).
    Gradients accumulate across all users; torch's parameter iterators
    dedupe the shared weights. replace=True swaps out an existing
    child of the same name (its own weights are discarded)."""
    if name in self._children_names:
      if not replace:
        raise ValueError(f'Child {name!r} already exists on '
                         f'{self.layer_name}')
      del self._modules[name]
      self._children_names.remove(name)
    self.add_module(name, child)
    self._children_names.append(name)

  def CreateChildren(self, name: str,
                     children_params: Sequence[InstantiableParams]) -> None:
    """Instantiates a list of sub-layers as an nn.ModuleList."""
    mods = []
    for i, cp0 in enumerate(children_params):
      cp = cp0.Copy()
      if not cp.name:
        cp.name = f'{name}_{i}'
      self._PropagateDtypes(cp)
      mods.append(cp.Instantiate())
    self.add_module(name, nn.ModuleList(mods))
    self._children_names.append(name)

  def _PropagateDtypes(self, cp: InstantiableParams) -> None:
    """Child inherits dtype/fprop_dtype/random_seed unless overridden."""
    p = self._params
    for f in ('dtype', 'fprop_dtype'):
      if f in cp and getattr(cp, f) == BaseLayer.Params().Get(f):
        try:
          setattr(cp, f, p.Get(f))
        except (AttributeError, TypeError):
          pass
    if 'random_seed' in cp and cp.random_seed is None:
      cp.random_seed = p.random_seed

  def RegisterAccumulator(self, name: str, acc: Accumulator) -> None:
    self._accumulators[name] = acc

  def GetAccumulator(self, name: str) -> Accumulator:
    return self._accumulators[name]

  # ---- theta ------------------------------------------------------------
  @property
  def fprop_dtype(self) -> torch.dtype:
    return self._params.fprop_dtype or self._params.dtype

  @property
  def theta(self) -> NestedMap:
    """NestedMap of this layer's (and children's) parameters.

    Floating-point parameters are cast to fprop_dtype, recorded by
    autograd, so gradients flow to the fp32 masters (mixed precision).
    """
    ret = NestedMap()
    cast_to = self.fprop_dtype
    for name, param in self.named_parameters(recurse=False):
      t = param
      if t.is_floating_point() and t.dtype != cast_to:
        t = t.to(cast_to)
      ret[name] = t
    for name in self._children_names:
      child = getattr(self, name)
      if isinstance(child, nn.ModuleList):
        ret[name] = [c.theta for c in child]
      else:
        ret[name] = child.theta
    return ret

  @property
  def vars(self) -> NestedMap:
    """NestedMap of raw (uncast) nn.Parameters."""
    ret = NestedMap()
    for name, param in self.named_parameters(recurse=False):
      ret[name] = param
    for name in self._children_names:
      child = getattr(self, name)
      if isinstance(child, nn.ModuleList):
        ret[name] = [c.vars for c in child]
      else:
        ret[name] = child.vars
    return ret

  # ---- forward ----------------------------------------------------------
  def FProp(self, theta: NestedMap, *args, **kwargs):
    raise NotImplementedError(
        f'{type(self).__name__} must implement FProp')

  def FPropDefaultTheta(self, *args, **kwargs):
    return self.FProp(self.theta, *args, **kwargs)

  def forward(self, *args, **kwargs):  # torch entry point
    return self.FPropDefaultTheta(*args, **kwargs)

  def PostTrainingStepUpdate(self, global_step: int) -> None:
    """Hook called after each optimizer step (reference base_layer.py:1129).
    Default: recurse into children."""
    for name in self._children_names:
      child = getattr(self, name)
      if isinstance(child, nn.ModuleList):
        for c in child:
          c.PostTrainingStepUpdate(global_step)
      else:
        child.PostTrainingStepUpdate(global_step)

  def extra_repr(self) -> str:
    return f'name={self.layer_name}'
