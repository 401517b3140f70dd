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
    use_fused = bool(params) and all(
        t.is_cuda for t in params if isinstance(t, torch.Tensor))
    if use_fused:
      try:
        from lingvo_amd.ops.fused_adam import FusedAdamW
        return FusedAdamW(params, lr=lr, betas=(p.beta1, p.beta2),
                          eps=p.epsilon, weight_decay=p.weight_decay)
      except Exception:
        pass
    if p.weight_decay:
      return torch.optim.AdamW(params, lr=lr, betas=(p.beta1, p.beta2),
                               eps=p.epsilon, weight_decay=p.weight_decay)
    return torch.optim.Adam(params, lr=lr, betas=(p.beta1, p.beta2),
                            eps=p.epsilon)


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
    return self.inner.CreateTorchOptimizer(params, lr)
