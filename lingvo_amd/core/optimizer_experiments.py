"""Second-order / experimental optimizers.

MI355X-native re-implementations of the reference's optimizer
experiments (lingvo/core/distributed_shampoo.py, adagraft.py,
graddrop.py). No preconditioner service is needed: a single MI355X has
288 GB HBM3E and the eigendecompositions run on-device (or CPU for tiny
blocks), so Shampoo is just another torch.optim.Optimizer here.
"""

from __future__ import annotations

import torch

from lingvo_amd.core import optimizer as optimizer_lib


class Shampoo(optimizer_lib.Base):
  """Block-diagonal full-matrix preconditioning (reference
  distributed_shampoo.py; Gupta et al. 2018). For a [m, n] parameter,
  maintains L += G G^T and R += G^T G and preconditions with
  L^{-1/4} G R^{-1/4}; large dims are split into blocks, 1-D params
  fall back to diagonal AdaGrad. Inverse-pth-roots are recomputed every
  `statistics_compute_steps` steps via eigendecomposition."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('block_size', 128, 'Max preconditioner block dim.')
    p.Define('statistics_compute_steps', 10,
             'Recompute inverse roots every N steps.')
    p.Define('epsilon', 1e-6, 'Statistics damping.')
    p.Define('momentum', 0.9, 'Heavy-ball momentum.')
    p.Define('matrix_eps', 1e-6, 'Eigenvalue floor.')
    return p

  def CreateTorchOptimizer(self, params, lr):
    return _ShampooImpl(params, self.p, lr)


def _inverse_quarter_root(mat: torch.Tensor, eps: float) -> torch.Tensor:
  """M^{-1/4} for symmetric PSD M via eigendecomposition (fp32)."""
  evals, evecs = torch.linalg.eigh(mat.float())
  evals = evals.clamp_min(eps)
  return (evecs * evals.pow(-0.25)) @ evecs.T


class _ShampooImpl(torch.optim.Optimizer):

  def __init__(self, params, p, lr):
    self._cfg = p
    super().__init__(params, dict(lr=lr))

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
          if g.dim() == 2 and max(g.shape) <= cfg.block_size and \
              min(g.shape) > 1:
            state['L'] = torch.zeros(
                g.shape[0], g.shape[0], device=g.device)
            state['R'] = torch.zeros(
                g.shape[1], g.shape[1], device=g.device)
            state['invL'] = None
            state['invR'] = None
          else:
            state['diag'] = torch.zeros_like(g)
          state['mom'] = torch.zeros_like(g)
        state['step'] += 1
        if 'L' in state:
          state['L'] += g @ g.T
          state['R'] += g.T @ g
          if state['invL'] is None or \
              state['step'] % cfg.statistics_compute_steps == 0:
            state['invL'] = _inverse_quarter_root(
                state['L'] + cfg.epsilon * torch.eye(
                    g.shape[0], device=g.device), cfg.matrix_eps)
            state['invR'] = _inverse_quarter_root(
                state['R'] + cfg.epsilon * torch.eye(
                    g.shape[1], device=g.device), cfg.matrix_eps)
          precond = state['invL'] @ g @ state['invR']
          # graft AdaGrad magnitude so the step size matches a tuned
          # first-order run (reference distributed_shampoo grafting)
          precond = precond * (g.norm() / precond.norm().clamp_min(1e-16))
        else:
          state['diag'] += g * g
          precond = g / (state['diag'].sqrt() + 1e-6)
          precond = precond * (g.norm() / precond.norm().clamp_min(1e-16))
        state['mom'].mul_(cfg.momentum).add_(precond)
        param.add_(state['mom'].to(param.dtype), alpha=-lr)
    return None


class AdaGraft(optimizer_lib.Base):
  """Magnitude/direction grafting (reference adagraft.py; Agarwal et
  al. 2020): per-parameter steps take the DIRECTION of one optimizer
  and the NORM of another."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('magnitude_optimizer_tpl', optimizer_lib.Adam.Params(),
             'Optimizer whose step SIZE is used.')
    p.Define('direction_optimizer_tpl', optimizer_lib.SGD.Params(),
             'Optimizer whose step DIRECTION is used.')
    return p

  def __init__(self, params):
    super().__init__(params)
    self.CreateChild('mag', self.p.magnitude_optimizer_tpl)
    self.CreateChild('dir', self.p.direction_optimizer_tpl)

  def CreateTorchOptimizer(self, params, lr):
    params = list(params)
    return _AdaGraftImpl(params, self.mag, self.dir, lr)


class _AdaGraftImpl(torch.optim.Optimizer):

  def __init__(self, params, mag_layer, dir_layer, lr):
    super().__init__(params, dict(lr=lr))
    # shadow copies stepped by the two inner optimizers
    self._shadow_m = [p.detach().clone() for g in self.param_groups
                      for p in g['params']]
    self._shadow_d = [p.detach().clone() for p in self._shadow_m]
    for s in self._shadow_m + self._shadow_d:
      s.requires_grad_(False)
    self._opt_m = mag_layer.CreateTorchOptimizer(self._shadow_m, lr)
    self._opt_d = dir_layer.CreateTorchOptimizer(self._shadow_d, lr)

  @torch.no_grad()
  def step(self, closure=None):
    flat = [p for g in self.param_groups for p in g['params']]
    for p, sm, sd in zip(flat, self._shadow_m, self._shadow_d):
      sm.copy_(p)
      sd.copy_(p)
      sm.grad = p.grad
      sd.grad = p.grad
    self._opt_m.step()
    self._opt_d.step()
    for p, sm, sd in zip(flat, self._shadow_m, self._shadow_d):
      if p.grad is None:
        continue
      step_m = sm - p   # magnitude donor
      step_d = sd - p   # direction donor
      dn = step_d.norm().clamp_min(1e-16)
      p.add_(step_d * (step_m.norm() / dn))
      sm.grad = None
      sd.grad = None
    return None


class GradDropCompressor:
  """Gradient dropping with error feedback (reference graddrop.py; Aji
  & Heafield 2017): before each DP all-reduce keep only the top
  `keep_frac` gradient entries by magnitude, accumulating the dropped
  remainder locally into the next step. Used standalone or as a
  GradSync pre-hook."""

  def __init__(self, keep_frac: float = 0.01):
    assert 0.0 < keep_frac <= 1.0
    self.keep_frac = keep_frac
    self._residual = {}

  def compress(self, name: str, grad: torch.Tensor) -> torch.Tensor:
    res = self._residual.get(name)
    if res is None:
      res = torch.zeros_like(grad)
      self._residual[name] = res
    acc = grad + res
    k = max(1, int(acc.numel() * self.keep_frac))
    flat = acc.flatten()
    thresh = flat.abs().topk(k).values[-1]
    mask = flat.abs() >= thresh
    kept = torch.where(mask, flat, torch.zeros_like(flat))
    res.copy_((flat - kept).reshape_as(res))
    return kept.reshape_as(grad)


class EGDD(optimizer_lib.Base):
  """Exponentiated Gradient Delta-Delta (reference egdd.py:49; Kivinen
  & Warmuth 1997): momentum SGD with per-weight multiplicative gains
  and a per-parameter learning-rate scale, both updated by
  unnormalized exponentiated gradient."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('momentum', 0.9, 'Heavy-ball momentum.')
    p.Define('beta', 0.9, 'Gradient-EMA decay for the gain update.')
    p.Define('gain_learning_rate', 0.01, 'EG rate for per-weight gain.')
    p.Define('scale_learning_rate', 0.001, 'EG rate for lr scale.')
    p.Define('initial_gain', 1.0, 'Initial per-weight gain.')
    p.Define('min_gain', 1e-2, 'Gain floor.')
    p.Define('max_gain', 1e2, 'Gain cap.')
    p.Define('initial_scale', 1.0, 'Initial lr scale.')
    p.Define('min_scale', 1e-1, 'Scale floor.')
    p.Define('max_scale', 1e1, 'Scale cap.')
    p.Define('use_directions', True,
             'Scale update uses normalized grad/momentum directions.')
    p.Define('use_signs', True,
             'Gain update uses sign(grad)*sign(gbar).')
    return p

  def CreateTorchOptimizer(self, params, lr):
    return _EgddImpl(params, self.p, lr)


class _EgddImpl(torch.optim.Optimizer):

  def __init__(self, params, p, lr):
    super().__init__(params, dict(lr=lr))
    self._p = p

  @torch.no_grad()
  def step(self, closure=None):
    loss = closure() if closure is not None else None
    p_ = self._p
    for group in self.param_groups:
      lr = group['lr']
      for w in group['params']:
        if w.grad is None:
          continue
        g = w.grad.float()
        st = self.state[w]
        if not st:
          st['momentum'] = torch.zeros_like(g)
          st['gbar'] = torch.zeros_like(g)
          st['gain'] = torch.full_like(g, p_.initial_gain)
          st['lr_scale'] = torch.tensor(float(p_.initial_scale),
                                        device=g.device)
          st['counter'] = 0
        m, gbar, gain = st['momentum'], st['gbar'], st['gain']
        st['counter'] += 1
        # lr-scale EG update from grad/momentum alignment.
        if p_.use_directions:
          ng = g / (g.norm() + 1e-10)
          nm = m / (m.norm() + 1e-10)
          align = (ng * nm).sum()
        else:
          align = (g * m).sum()
        st['lr_scale'] = (st['lr_scale'] *
                          torch.exp(p_.scale_learning_rate * align)
                          ).clamp(p_.min_scale, p_.max_scale)
        # per-weight gain EG update.
        if p_.use_signs:
          upd = torch.exp(p_.gain_learning_rate * torch.sign(g) *
                          torch.sign(gbar))
        else:
          corr = gbar / (1.0 - p_.beta ** max(st['counter'] - 1, 1))
          upd = torch.exp(p_.gain_learning_rate * g * corr)
        gain.mul_(upd).clamp_(p_.min_gain, p_.max_gain)
        m.mul_(p_.momentum).add_(lr * gain * g)
        gbar.mul_(p_.beta).add_(g, alpha=1.0 - p_.beta)
        w.add_((-st['lr_scale'] * m).to(w.dtype))
    return loss
