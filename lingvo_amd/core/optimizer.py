"""Optimizers as Params-instantiable wrappers (reference lingvo/core/optimizer.py).

Each class exposes `Params()` and `CreateTorchOptimizer(params_iter, lr)`.
On a GPU, Adam runs through the fused multi-tensor HIP kernel (K16 in
SURVEY.md §2.9) via lingvo_amd.ops; pure-torch everywhere else.
"""

from __future__ import annotations

import math
from typing import Iterable, List, Optional

import torch

from lingvo_amd.core.base_layer import BaseLayer
from lingvo_amd.core.hyperparams import InstantiableParams


class Base(BaseLayer):

  @classmethod
  def Params(cls):
    p = super().Params()
    return p

  def CreateTorchOptimizer(self, params, lr: float) -> torch.optim.Optimizer:
    raise NotImplementedError


class SGD(Base):

  def CreateTorchOptimizer(self, params, lr):
    return torch.optim.SGD(params, lr=lr)


class Momentum(Base):

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('momentum', 0.9, 'Momentum coefficient.')
    p.Define('use_nesterov', False, 'Nesterov momentum.')
    return p

  def CreateTorchOptimizer(self, params, lr):
    return torch.optim.SGD(params, lr=lr, momentum=self.p.momentum,
                           nesterov=self.p.use_nesterov)


class RMSProp(Base):

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('decay', 0.9, 'Decay rate.')
    p.Define('momentum', 0.0, 'Momentum.')
    p.Define('epsilon', 1e-10, 'Epsilon.')
    return p

  def CreateTorchOptimizer(self, params, lr):
    return torch.optim.RMSprop(params, lr=lr, alpha=self.p.decay,
                               momentum=self.p.momentum, eps=self.p.epsilon)


class Adagrad(Base):

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('initial_accumulator_value', 0.1, 'Initial accumulator.')
    return p

  def CreateTorchOptimizer(self, params, lr):
    return torch.optim.Adagrad(
        params, lr=lr,
        initial_accumulator_value=self.p.initial_accumulator_value)


class Adam(Base):

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('beta1', 0.9, 'Beta1.')
    p.Define('beta2', 0.999, 'Beta2.')
    p.Define('epsilon', 1e-6, 'Epsilon.')
    p.Define('weight_decay', 0.0, 'Decoupled weight decay (AdamW).')
    return p

  @classmethod
  def ParamsA(cls):
    """Common ASR preset (reference optimizer.py Adam.ParamsA)."""
    return cls.Params().Set(beta1=0.9, beta2=0.999, epsilon=1e-8)

  @classmethod
  def ParamsB(cls):
    """Common NMT preset (reference optimizer.py Adam.ParamsB)."""
    return cls.Params().Set(beta1=0.9, beta2=0.98, epsilon=1e-9)

  def CreateTorchOptimizer(self, params, lr):
    p = self.p
    params = list(params)
    if params and any(t.dtype == torch.bfloat16 for t in params):
      # bf16-weight training: fp32 masters live in optimizer state,
      # updated with ~10 multi-tensor foreach launches.
      return MasterAdamW(params, lr=lr, betas=(p.beta1, p.beta2),
                         eps=p.epsilon, weight_decay=p.weight_decay)
    if p.weight_decay:
      return torch.optim.AdamW(params, lr=lr, betas=(p.beta1, p.beta2),
                               eps=p.epsilon, weight_decay=p.weight_decay)
    return torch.optim.Adam(params, lr=lr, betas=(p.beta1, p.beta2),
                            eps=p.epsilon)


class MasterAdamW(torch.optim.Optimizer):
  """AdamW for bf16 model weights with fp32 master copies.

  The whole update is a fixed number of _foreach launches independent of
  parameter count: grads cast fp32 -> moments update -> master update ->
  bf16 copy-back. Masters persist in state_dict for exact resume.
  """

  def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
               weight_decay=0.0):
    super().__init__(params, dict(lr=lr, betas=betas, eps=eps,
                                  weight_decay=weight_decay))

  def load_state_dict(self, state_dict):
    # torch's default load casts fp32 state to the (bf16) param dtype,
    # which would destroy the masters; restore exact tensors instead.
    params = [p for g in self.param_groups for p in g['params']]
    saved_ids = [i for g in state_dict['param_groups']
                 for i in g['params']]
    for pid, prm in zip(saved_ids, params):
      if pid in state_dict['state']:
        self.state[prm] = {
            k: (v.clone().to(prm.device) if torch.is_tensor(v) else v)
            for k, v in state_dict['state'][pid].items()
        }
    for g, sg in zip(self.param_groups, state_dict['param_groups']):
      for k, v in sg.items():
        if k != 'params':
          g[k] = v

  @torch.no_grad()
  def step(self, closure=None):
    for group in self.param_groups:
      lr = group['lr']
      beta1, beta2 = group['betas']
      eps = group['eps']
      wd = group['weight_decay']
      params = [prm for prm in group['params'] if prm.grad is not None]
      if not params:
        continue
      # State is created lazily on first grad, so params whose grads
      # appear later can have a lower step count: bucket by per-param
      # step so bias correction is exact (normally one bucket).
      by_step = {}
      for prm in params:
        st = self.state[prm]
        if not st:
          st['step'] = 0
          st['master'] = prm.detach().float().clone()
          st['m'] = torch.zeros_like(st['master'])
          st['v'] = torch.zeros_like(st['master'])
          st['g32'] = torch.zeros_like(st['master'])
        st['step'] += 1
        by_step.setdefault(st['step'], []).append(prm)
      for t, bucket in by_step.items():
        grads, masters, ms, vs = [], [], [], []
        for prm in bucket:
          st = self.state[prm]
          grads.append(st['g32'])
          masters.append(st['master'])
          ms.append(st['m'])
          vs.append(st['v'])
        # One batched cast of all bf16 grads to the fp32 scratch buffers.
        torch._foreach_copy_(grads, [prm.grad for prm in bucket])
        bc1 = 1.0 - beta1 ** t
        bc2 = 1.0 - beta2 ** t
        torch._foreach_mul_(ms, beta1)
        torch._foreach_add_(ms, grads, alpha=1.0 - beta1)
        torch._foreach_mul_(vs, beta2)
        torch._foreach_addcmul_(vs, grads, grads, value=1.0 - beta2)
        denom = torch._foreach_sqrt(vs)
        torch._foreach_div_(denom, bc2 ** 0.5)
        torch._foreach_add_(denom, eps)
        if wd:
          torch._foreach_mul_(masters, 1.0 - lr * wd)
        torch._foreach_addcdiv_(masters, ms, denom, value=-(lr / bc1))
        torch._foreach_copy_(bucket, masters)  # bf16 cast back
    return None


class Adafactor(Base):
  """Factored second-moment optimizer (reference optimizer.py:905
  XLAShardingAdafactor, sans sharding — memory already fits in 288 GB HBM)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('beta1', 0.0, 'First-moment decay; 0 disables momentum.')
    p.Define('decay_exponent', 0.8, 'Second-moment decay exponent.')
    p.Define('epsilon1', 1e-30, 'Regularization epsilon.')
    p.Define('epsilon2', 1e-3, 'RMS-clip epsilon.')
    p.Define('min_dim_size_to_factor', 128,
             'Only factor dims both >= this.')
    return p

  def CreateTorchOptimizer(self, params, lr):
    return _AdafactorImpl(params, self.p, lr)


class _AdafactorImpl(torch.optim.Optimizer):

  def __init__(self, params, p, lr):
    defaults = dict(lr=lr)
    self._cfg = p
    super().__init__(params, defaults)

  @torch.no_grad()
  def step(self, closure=None):
    cfg = self._cfg
    for group in self.param_groups:
      lr = group['lr']
      for param in group['params']:
        if param.grad is None:
          continue
        g = param.grad.float()
        state = self.state[param]
        if not state:
          state['step'] = 0
          shape = g.shape
          factored = (g.dim() == 2 and
                      shape[0] >= cfg.min_dim_size_to_factor and
                      shape[1] >= cfg.min_dim_size_to_factor)
          state['factored'] = factored
          if factored:
            state['vr'] = torch.zeros(shape[0], device=g.device)
            state['vc'] = torch.zeros(shape[1], device=g.device)
          else:
            state['v'] = torch.zeros_like(g)
          if cfg.beta1:
            state['m'] = torch.zeros_like(g)
        state['step'] += 1
        t = state['step']
        beta2 = 1.0 - t ** (-cfg.decay_exponent)
        g2 = g * g + cfg.epsilon1
        if state['factored']:
          state['vr'].mul_(beta2).add_(g2.mean(dim=1), alpha=1 - beta2)
          state['vc'].mul_(beta2).add_(g2.mean(dim=0), alpha=1 - beta2)
          r = state['vr'] / state['vr'].mean().clamp_min(cfg.epsilon1)
          u = g / (r.sqrt()[:, None] * state['vc'].sqrt()[None, :] +
                   cfg.epsilon1)
        else:
          state['v'].mul_(beta2).add_(g2, alpha=1 - beta2)
          u = g / state['v'].sqrt().clamp_min(cfg.epsilon1)
        # RMS clip (Adafactor update clipping, d=1).
        rms = u.pow(2).mean().sqrt().clamp_min(1.0)
        u = u / rms
        if cfg.beta1:
          state['m'].mul_(cfg.beta1).add_(u, alpha=1 - cfg.beta1)
          u = state['m']
        param.add_(u.to(param.dtype), alpha=-lr)
    return None


class Accumulator(Base):
  """Gradient accumulation wrapper: applies the inner optimizer every
  accum_steps steps (reference optimizer.py:507)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('optimizer_tpl', Adam.Params(), 'Inner optimizer params.')
    p.Define('accum_steps', 1, 'Steps between applies.')
    return p

  def __init__(self, params):
    super().__init__(params)
    self.CreateChild('inner', self.p.optimizer_tpl)

  def CreateTorchOptimizer(self, params, lr):
    params = list(params)
    inner = self.inner.CreateTorchOptimizer(params, lr)
    if self.p.accum_steps <= 1:
      return inner
    return _AccumulatorImpl(inner, params, self.p.accum_steps)


class _AccumulatorImpl:
  """Duck-typed optimizer: sums grads into side buffers each step() call
  and applies the inner optimizer with the averaged grads every
  accum_steps calls (reference optimizer.py:507 Accumulator semantics:
  intermediate steps are pure accumulation, no parameter update)."""

  def __init__(self, inner, params, accum_steps):
    self._inner = inner
    self._params = params
    self._n = int(accum_steps)
    self._count = 0
    self._acc = {}  # id(param) -> fp32 accumulation buffer

  @property
  def param_groups(self):
    return self._inner.param_groups

  @property
  def state(self):
    return self._inner.state

  def zero_grad(self, set_to_none=True):
    self._inner.zero_grad(set_to_none=set_to_none)

  @torch.no_grad()
  def step(self, closure=None):
    for prm in self._params:
      if prm.grad is None:
        continue
      buf = self._acc.get(id(prm))
      if buf is None:
        buf = torch.zeros_like(prm, dtype=torch.float32)
        self._acc[id(prm)] = buf
      buf.add_(prm.grad.float())
    self._count += 1
    if self._count < self._n:
      return None
    self._count = 0
    inv = 1.0 / self._n
    for prm in self._params:
      buf = self._acc.get(id(prm))
      if buf is None:
        continue
      if prm.grad is None:
        prm.grad = torch.zeros_like(prm)
      prm.grad.copy_((buf * inv).to(prm.grad.dtype))
      buf.zero_()
    return self._inner.step()

  def state_dict(self):
    sd = self._inner.state_dict()
    return {'inner': sd, 'accum_count': self._count,
            'accum_bufs': [self._acc.get(id(prm))
                           for prm in self._params]}

  def load_state_dict(self, state_dict):
    if 'inner' not in state_dict:  # plain inner-optimizer checkpoint
      self._inner.load_state_dict(state_dict)
      return
    self._inner.load_state_dict(state_dict['inner'])
    self._count = state_dict.get('accum_count', 0)
    for prm, buf in zip(self._params, state_dict.get('accum_bufs', [])):
      if buf is not None:
        self._acc[id(prm)] = buf.to(prm.device)


class AdaDelta(Base):

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('decay', 0.95, 'Rho.')
    p.Define('epsilon', 1e-6, 'Epsilon.')
    return p

  def CreateTorchOptimizer(self, params, lr):
    return torch.optim.Adadelta(params, lr=lr, rho=self.p.decay,
                                eps=self.p.epsilon)


class CompositeOptimizer(Base):
  """Regex -> optimizer routing (reference optimizer.py:199): parameters
  whose names match a pattern use that sub-optimizer."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('optimizer_map', [],
             'List of (name_regex, optimizer_params, lr_mult).')
    p.Define('default_optimizer', Adam.Params(), 'Fallback optimizer.')
    return p

  def __init__(self, params):
    super().__init__(params)
    subs = [op.Copy().Set(name=f'sub_{i}')
            for i, (_, op, _) in enumerate(self.p.optimizer_map)]
    subs.append(self.p.default_optimizer.Copy().Set(name='sub_default'))
    self.CreateChildren('subs', subs)

  def CreateTorchOptimizer(self, params, lr):
    import re as _re
    # params must be (name, param) pairs for routing; Learner passes raw
    # params, so CompositeOptimizer users call CreateRoutedOptimizers.
    return self.subs[-1].CreateTorchOptimizer(params, lr)

  def CreateRoutedOptimizers(self, named_params, lr):
    import re as _re
    buckets = [[] for _ in self.p.optimizer_map]
    default = []
    for name, prm in named_params:
      for i, (pat, _, _) in enumerate(self.p.optimizer_map):
        if _re.search(pat, name):
          buckets[i].append(prm)
          break
      else:
        default.append(prm)
    opts = []
    for i, (_, _, mult) in enumerate(self.p.optimizer_map):
      if buckets[i]:
        opts.append(self.subs[i].CreateTorchOptimizer(buckets[i],
                                                      lr * mult))
    if default:
      opts.append(self.subs[-1].CreateTorchOptimizer(default, lr))
    return opts
