"""Builder/combinator layers (reference lingvo/core/builder.py:38 and
builder_layers.py: SequentialLayer, ParallelLayer:1031, RepeatLayer:117,
LinearLayer:1131, BiasLayer:1203, RematerializationLayer:1370)."""

from __future__ import annotations

from typing import List, Optional

import torch

from lingvo_amd.core import py_utils
from lingvo_amd.core.base_layer import BaseLayer
from lingvo_amd.core.hyperparams import InstantiableParams
from lingvo_amd.core.nested_map import NestedMap


class SequentialLayer(BaseLayer):
  """Applies sub-layers in order (builder _Seq)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('sub', [], 'List of sub-layer params.')
    p.Define('repeat', 1, 'Repeat the sequence this many times '
             '(distinct weights).')
    return p

  def __init__(self, params):
    super().__init__(params)
    subs = []
    for r in range(self.p.repeat):
      for i, sp in enumerate(self.p.sub):
        subs.append(sp.Copy().Set(name=f'sub_{r}_{i}'))
    self.CreateChildren('seq', subs)

  def FProp(self, theta: NestedMap, *args):
    out = args
    for i, layer in enumerate(self.seq):
      result = layer.FProp(theta.seq[i], *out)
      out = result if isinstance(result, tuple) else (result,)
    return out[0] if len(out) == 1 else out


class ParallelLayer(BaseLayer):
  """Runs sub-layers on the same input and merges (builder _Par)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('sub', [], 'Sub-layer params.')
    p.Define('merge', 'sum', "'sum' | 'concat' | 'tuple'.")
    return p

  def __init__(self, params):
    super().__init__(params)
    self.CreateChildren('par', [sp.Copy() for sp in self.p.sub])

  def FProp(self, theta: NestedMap, *args):
    outs = [l.FProp(theta.par[i], *args) for i, l in enumerate(self.par)]
    if self.p.merge == 'sum':
      out = outs[0]
      for o in outs[1:]:
        out = out + o
      return out
    if self.p.merge == 'concat':
      return torch.cat(outs, dim=-1)
    return tuple(outs)


class LinearLayer(BaseLayer):
  """y = x @ w (builder_layers.py:1131)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('input_dims', 0, 'In.')
    p.Define('output_dims', 0, 'Out.')
    return p

  def __init__(self, params):
    super().__init__(params)
    self.CreateVariable('w', py_utils.WeightParams(
        [self.p.input_dims, self.p.output_dims], self.p.params_init,
        self.p.dtype))

  def FProp(self, theta, x):
    return torch.matmul(x, theta.w)


class BiasLayer(BaseLayer):
  """y = x + b (builder_layers.py:1203)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('dims', 0, 'Dim.')
    return p

  def __init__(self, params):
    super().__init__(params)
    self.CreateVariable('b', py_utils.WeightParams(
        [self.p.dims], py_utils.WeightInit.Constant(0.0), self.p.dtype))

  def FProp(self, theta, x):
    return x + theta.b


class MapLayer(BaseLayer):
  """Applies a python fn elementwise (builder _Fn)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('fn', None, 'Callable applied to the input.')
    return p

  def FProp(self, theta, *args):
    return self.p.fn(*args)


class RematerializationLayer(BaseLayer):
  """Gradient-checkpoints its body (builder_layers.py:1370;
  py_utils.RematerializeFn:5005)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('body', None, 'Body layer params.')
    return p

  def __init__(self, params):
    super().__init__(params)
    self.CreateChild('body', self.p.body)

  def FProp(self, theta, *args):
    if self.training and torch.is_grad_enabled():
      return torch.utils.checkpoint.checkpoint(
          lambda *a: self.body.FProp(theta.body, *a), *args,
          use_reentrant=False)
    return self.body.FProp(theta.body, *args)


class RepeatLayer(BaseLayer):
  """N iterations of one body with per-iteration weights stored as a
  leading-dim stack (reference repeat_layer.py GenericRepeatLayer /
  builder_layers.py:117): the body is scanned over its own weights, so
  the graph contains ONE body instance regardless of depth."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('body', None, 'Body layer params.')
    p.Define('repeat', 1, 'Iterations.')
    p.Define('per_layer_vars', True, 'Distinct weights per iteration.')
    p.Define('remat', False, 'Checkpoint each iteration.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    if p.per_layer_vars:
      bodies = [p.body.Copy().Set(name=f'body_{i}')
                for i in range(p.repeat)]
      self.CreateChildren('bodies', bodies)
    else:
      self.CreateChild('shared_body', p.body)

  def FProp(self, theta: NestedMap, *args):
    p = self.p
    out = args
    for i in range(p.repeat):
      if p.per_layer_vars:
        layer, th = self.bodies[i], theta.bodies[i]
      else:
        layer, th = self.shared_body, theta.shared_body
      if p.remat and self.training and torch.is_grad_enabled():
        result = torch.utils.checkpoint.checkpoint(
            lambda *a, _l=layer, _t=th: _l.FProp(_t, *a), *out,
            use_reentrant=False)
      else:
        result = layer.FProp(th, *out)
      out = result if isinstance(result, tuple) else (result,)
    return out[0] if len(out) == 1 else out


class GraphLayer(BaseLayer):
  """Executes a DAG of sub-layers over named tensors (reference
  builder_layers.py:886 GraphLayer / GraphTensors).

  Each entry of p.sub is ('in1,in2->out1,out2', layer_params). FProp
  binds positional args to p.input_endpoints, runs entries in order
  (each may read any previously-defined name) and returns
  p.output_endpoints.
  """

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('sub', [], "List of (signature, layer params).")
    p.Define('input_endpoints', [], 'Names bound to FProp args.')
    p.Define('output_endpoints', [], 'Names returned from FProp.')
    return p

  def __init__(self, params):
    super().__init__(params)
    self._sigs = []
    subs = []
    for i, (sig, sp) in enumerate(self.p.sub):
      ins, outs = sig.split('->')
      self._sigs.append(([s.strip() for s in ins.split(',') if s.strip()],
                         [s.strip() for s in outs.split(',') if s.strip()]))
      subs.append(sp.Copy().Set(name=f'g{i}_{sp.name or "sub"}'))
    self.CreateChildren('nodes', subs)

  def FProp(self, theta: NestedMap, *args):
    p = self.p
    assert len(args) == len(p.input_endpoints), (
        len(args), p.input_endpoints)
    env = dict(zip(p.input_endpoints, args))
    for i, layer in enumerate(self.nodes):
      ins, outs = self._sigs[i]
      result = layer.FProp(theta.nodes[i], *[env[n] for n in ins])
      result = result if isinstance(result, tuple) else (result,)
      assert len(result) == len(outs), (outs, len(result))
      env.update(zip(outs, result))
    ret = tuple(env[n] for n in p.output_endpoints)
    return ret[0] if len(ret) == 1 else ret


class FnLayer(BaseLayer):
  """Wraps a stateless function as a layer (builder _Fn)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('fn', None, 'Callable applied to FProp args.')
    return p

  def FProp(self, theta: NestedMap, *args):
    return self.p.fn(*args)


class Builder:
  """Pattern-based model-stack builder (reference builder.py:38): each
  method returns a Params TREE; composition happens on params, and a
  single Instantiate() materializes the network. Subclass and add
  domain-specific patterns (the reference's DenseBuilder / LmBuilder
  idiom)."""

  def __init__(self, dtype=torch.float32):
    self.dtype = dtype

  def _Seq(self, name, *subs):
    return SequentialLayer.Params().Set(name=name, sub=list(subs))

  def _Rep(self, name, repeat, *subs):
    return SequentialLayer.Params().Set(name=name, sub=list(subs),
                                        repeat=repeat)

  def _Par(self, name, merge, *subs):
    return ParallelLayer.Params().Set(name=name, merge=merge,
                                      sub=list(subs))

  def _Graph(self, name, input_endpoints, output_endpoints, *entries):
    return GraphLayer.Params().Set(
        name=name, input_endpoints=list(input_endpoints),
        output_endpoints=list(output_endpoints), sub=list(entries))

  def _Linear(self, name, input_dim, output_dim):
    return LinearLayer.Params().Set(name=name, input_dims=input_dim,
                                    output_dims=output_dim)

  def _Bias(self, name, dim):
    return BiasLayer.Params().Set(name=name, dims=dim)

  def _Fn(self, name, fn):
    return FnLayer.Params().Set(name=name, fn=fn)

  def _LN(self, name, dim):
    from lingvo_amd.layers import layers as lingvo_layers
    return lingvo_layers.LayerNorm.Params().Set(name=name, input_dim=dim)

  def _Dropout(self, name, keep_prob):
    from lingvo_amd.layers import layers as lingvo_layers
    return lingvo_layers.DropoutLayer.Params().Set(
        name=name, keep_prob=keep_prob)

  def _Remat(self, name, body):
    return RematerializationLayer.Params().Set(name=name, body=body)
