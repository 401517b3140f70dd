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


class FirstNLayer(BaseLayer):
  """Returns the first n positional args (reference builder_layers.py:29)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('n', 1, 'Number of args to return.')
    return p

  def FProp(self, theta: NestedMap, *args):
    n = self.p.n
    assert len(args) >= n
    return args[0] if n == 1 else tuple(args[:n])


class ArgIndexLayer(BaseLayer):
  """Selects args by index (reference builder_layers.py:60)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('idx', [], 'Indices of args to return.')
    return p

  def FProp(self, theta: NestedMap, *args):
    out = tuple(args[i] for i in self.p.idx)
    return out[0] if len(out) == 1 else out


class CreateNestedMapLayer(BaseLayer):
  """Packs positional args into a NestedMap by key
  (reference builder_layers.py:100)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('keys', [], 'Dotted keys, one per positional arg.')
    return p

  def FProp(self, theta: NestedMap, *args):
    out = NestedMap()
    for key, value in zip(self.p.keys, args):
      out.Set(key, value)
    return out


class UnarySequentialLayer(BaseLayer):
  """Sequential over exactly one tensor (reference builder_layers.py:554)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('sub', [], 'Sub-layer params.')
    return p

  def __init__(self, params):
    super().__init__(params)
    self.CreateChildren(
        'seq', [sp.Copy().Set(name=f'sub_{i}')
                for i, sp in enumerate(self.p.sub)])

  def FProp(self, theta: NestedMap, x):
    for i, layer in enumerate(self.seq):
      x = layer.FProp(theta.seq[i], x)
    return x


class BranchLayer(BaseLayer):
  """Runs a GraphLayer body and appends named intermediate tensors to
  its outputs (reference builder_layers.py:1256 BranchLayer; fetches
  here are graph tensor names rather than TF activation fetch points)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('body', None, 'GraphLayer params.')
    p.Define('fetches', [], 'Graph tensor names to append to outputs.')
    return p

  def __init__(self, params):
    super().__init__(params)
    body = self.p.body.Copy()
    body.output_endpoints = list(body.output_endpoints) +         [f for f in self.p.fetches if f not in body.output_endpoints]
    self.CreateChild('body', body)

  def FProp(self, theta: NestedMap, *args):
    out = self.body.FProp(theta.body, *args)
    return out


class PrintShapeLayer(BaseLayer):
  """Identity that logs arg shapes (reference builder_layers.py:1402)."""

  def FProp(self, theta: NestedMap, *args):
    for i, a in enumerate(args):
      if isinstance(a, torch.Tensor):
        print(f'{self.p.name} arg{i}: shape={tuple(a.shape)} '
              f'dtype={a.dtype}')
      else:
        print(f'{self.p.name} arg{i}: {a!r}')
    return args[0] if len(args) == 1 else args


class ReshapeLayer(BaseLayer):
  """Reshapes input (reference builder_layers.py:1428); -1 allowed."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('shape', [], 'Target shape.')
    return p

  def FProp(self, theta: NestedMap, x: torch.Tensor) -> torch.Tensor:
    return x.reshape(*self.p.shape)


class ConcatLayer(BaseLayer):
  """Concatenates args along an axis (reference builder_layers.py:1451)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('axis', -1, 'Concat axis.')
    return p

  def FProp(self, theta: NestedMap, *args):
    return torch.cat(list(args), dim=self.p.axis)


class SliceLayer(BaseLayer):
  """Slices the last dim (reference builder_layers.py:1474)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('begin', 0, 'Start index (last dim).')
    p.Define('size', -1, 'Length (-1 = to end).')
    return p

  def FProp(self, theta: NestedMap, x: torch.Tensor) -> torch.Tensor:
    b = self.p.begin
    return x[..., b:] if self.p.size < 0 else         x[..., b:b + self.p.size]


class SoftCondLayer(BaseLayer):
  """Soft conditional computation (reference builder_layers.py:274;
  arXiv:1904.04971): body runs with a per-batch sigmoid-weighted
  average over num_experts copies of its theta. Weights come from the
  mean input embedding (the reference also collapses the batch)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('body', None, 'Params of the wrapped layer.')
    p.Define('num_experts', 0, 'Expert (theta copy) count.')
    p.Define('cond_dim', 0, 'Input dim for the gating projection.')
    p.Define('nonzeros_mean', False,
             'Mean over nonzero rows only (packed inputs).')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    assert p.num_experts and p.cond_dim
    self.CreateChildren(
        'experts',
        [p.body.Copy().Set(name=f'expert_{i}')
         for i in range(p.num_experts)])
    self.CreateVariable('w', py_utils.WeightParams(
        [p.cond_dim, p.num_experts], p.params_init, p.dtype))

  def _GetExpertDist(self, theta: NestedMap,
                     inputs: torch.Tensor) -> torch.Tensor:
    flat = inputs.reshape(-1, self.p.cond_dim).float()
    if self.p.nonzeros_mean:
      nonzero = (flat.abs().sum(-1, keepdim=True) > 0).float()
      emb = (flat * nonzero).sum(0) / nonzero.sum().clamp_min(1e-10)
    else:
      emb = flat.mean(0)
    return torch.sigmoid(emb @ theta.w.float())

  def FProp(self, theta: NestedMap, inputs: torch.Tensor, *args):
    dist = self._GetExpertDist(theta, inputs)
    flat_thetas = [theta.experts[i].Flatten()
                   for i in range(self.p.num_experts)]
    mixed = []
    for vals in zip(*flat_thetas):
      if isinstance(vals[0], torch.Tensor):
        stacked = torch.stack([v.float() for v in vals])
        mix = torch.einsum('e,e...->...', dist, stacked)
        mixed.append(mix.to(vals[0].dtype))
      else:
        mixed.append(vals[0])
    weighted = theta.experts[0].Pack(mixed)
    return self.experts[0].FProp(weighted, inputs, *args)


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
