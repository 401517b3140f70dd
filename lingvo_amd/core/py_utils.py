"""Foundation utilities: weight init specs, padding ops, deterministic RNG.

Covers the capability surface of the reference's py_utils
(lingvo/core/py_utils.py): WeightInit/WeightParams (py_utils.py:1085,1250),
padded-sequence ops (4282-4539), deterministic per-step seeds (3933), and
numeric checks (208). All tensor math is torch; sequence layout is
batch-major [B, T, ...] with a float `paddings` tensor [B, T] where 1.0
marks padded positions (the reference's convention).
"""

from __future__ import annotations

import contextlib
import math
import threading
from typing import Any, List, Optional, Sequence, Tuple, Union

import torch

from lingvo_amd.core.hyperparams import Params
from lingvo_amd.core.nested_map import NestedMap


# --------------------------------------------------------------------------
# Weight initialization
# --------------------------------------------------------------------------
class WeightInit:
  """Factory for weight-initialization specs (method + scale)."""

  @staticmethod
  def _Spec(method: str, scale: float) -> Params:
    p = Params()
    p.Define('method', method, 'Initialization method.')
    p.Define('scale', scale, 'Initialization scale.')
    return p

  @staticmethod
  def Gaussian(scale: float = 1.0) -> Params:
    return WeightInit._Spec('gaussian', scale)

  @staticmethod
  def Uniform(scale: float = 1.0) -> Params:
    return WeightInit._Spec('uniform', scale)

  @staticmethod
  def Constant(scale: float = 0.0) -> Params:
    return WeightInit._Spec('constant', scale)

  @staticmethod
  def Xavier(scale: float = 1.0) -> Params:
    return WeightInit._Spec('xavier', scale)

  @staticmethod
  def GeoMeanXavier(scale: float = 1.0) -> Params:
    return WeightInit._Spec('geo_mean_xavier', scale)

  @staticmethod
  def TruncatedGaussian(scale: float = 1.0) -> Params:
    return WeightInit._Spec('truncated_gaussian', scale)

  @staticmethod
  def GaussianSqrtDim(scale: float = 1.0) -> Params:
    return WeightInit._Spec('gaussian_sqrt_dim', scale)

  @staticmethod
  def UniformSqrtDim(scale: float = 1.0) -> Params:
    return WeightInit._Spec('uniform_sqrt_dim', scale)

  @staticmethod
  def UniformUnitScaling(scale: float = 1.0) -> Params:
    return WeightInit._Spec('uniform_unit_scaling', scale)

  @staticmethod
  def TruncatedGaussianSqrtFanIn(scale: float = 1.0) -> Params:
    return WeightInit._Spec('truncated_gaussian_sqrt_fanin', scale)

  @staticmethod
  def TruncatedGaussianSqrtFanOut(scale: float = 1.0) -> Params:
    return WeightInit._Spec('truncated_gaussian_sqrt_fanout', scale)


def WeightParams(shape: Sequence[int],
                 init: Optional[Params] = None,
                 dtype: torch.dtype = torch.float32,
                 collections: Optional[List[str]] = None) -> Params:
  """Describes a trainable weight (reference py_utils.py:1250)."""
  p = Params()
  p.Define('shape', list(shape), 'Weight shape.')
  p.Define('init', init or WeightInit.Xavier(1.0), 'Initialization spec.')
  p.Define('dtype', dtype, 'Weight dtype.')
  p.Define('collections', collections or [], 'Weight collections.')
  return p


def _FanInFanOut(shape: Sequence[int]) -> Tuple[int, int]:
  if not shape:
    return 1, 1
  if len(shape) == 1:
    return shape[0], shape[0]
  # Conv kernels: [..spatial.., in, out]; matmul: [in, out].
  receptive = 1
  for d in shape[:-2]:
    receptive *= d
  return shape[-2] * receptive, shape[-1] * receptive


def InitWeight(shape: Sequence[int], spec: Params,
               generator: Optional[torch.Generator] = None,
               dtype: torch.dtype = torch.float32) -> torch.Tensor:
  """Materializes an initialized weight tensor from a WeightInit spec."""
  method, scale = spec.method, spec.scale
  shape = list(shape)
  dim0 = shape[0] if shape else 1
  fan_in, fan_out = _FanInFanOut(shape)

  def randn():
    return torch.randn(shape, generator=generator, dtype=torch.float32)

  def rand():
    return torch.rand(shape, generator=generator, dtype=torch.float32) * 2 - 1

  if method == 'constant':
    w = torch.full(shape, float(scale), dtype=torch.float32)
  elif method == 'gaussian':
    w = randn() * scale
  elif method == 'uniform':
    w = rand() * scale
  elif method == 'gaussian_sqrt_dim':
    w = randn() * (scale / math.sqrt(max(dim0, 1)))
  elif method == 'uniform_sqrt_dim':
    w = rand() * (scale / math.sqrt(max(dim0, 1)))
  elif method == 'xavier':
    limit = scale * math.sqrt(6.0 / (fan_in + fan_out))
    w = rand() * limit
  elif method == 'geo_mean_xavier':
    limit = scale * math.sqrt(3.0 / math.sqrt(fan_in * fan_out))
    w = rand() * limit
  elif method == 'uniform_unit_scaling':
    w = rand() * (scale * math.sqrt(3.0 / max(fan_in, 1)))
  elif method in ('truncated_gaussian', 'truncated_gaussian_sqrt_fanin',
                  'truncated_gaussian_sqrt_fanout'):
    std = scale
    if method.endswith('fanin'):
      std = scale / math.sqrt(max(fan_in, 1))
    elif method.endswith('fanout'):
      std = scale / math.sqrt(max(fan_out, 1))
    w = torch.empty(shape, dtype=torch.float32)
    torch.nn.init.trunc_normal_(w, std=std, a=-2 * std, b=2 * std,
                                generator=generator)
  else:
    raise ValueError(f'Unknown init method {method!r}')
  return w.to(dtype)


# --------------------------------------------------------------------------
# Deterministic step-seeded RNG (reference py_utils.py:3933)
# --------------------------------------------------------------------------
_RNG_STATE = threading.local()


def _GetRngStack() -> List[NestedMap]:
  if not hasattr(_RNG_STATE, 'stack'):
    _RNG_STATE.stack = [NestedMap(global_seed=1234, step=0, op_counter=[0])]
  return _RNG_STATE.stack


@contextlib.contextmanager
def StepSeedScope(global_seed: int, step: int):
  """Makes GenerateStepSeedPair deterministic in (global_seed, step)."""
  stack = _GetRngStack()
  stack.append(NestedMap(global_seed=int(global_seed), step=int(step),
                         op_counter=[0]))
  if torch.cuda.is_available():
    from lingvo_amd.ops import dropout as dropout_ops
    dropout_ops.SetStepSeed(int(step), int(global_seed))
  try:
    yield
  finally:
    stack.pop()


def GenerateStepSeedPair(op_seed: Optional[int] = None) -> Tuple[int, int]:
  """Returns a deterministic (seed1, seed2) for the current step scope.

  Each call without op_seed increments a per-scope op counter, so multiple
  stochastic ops in one step get distinct but reproducible seeds.
  """
  top = _GetRngStack()[-1]
  if op_seed is None:
    op_seed = top.op_counter[0]
    top.op_counter[0] += 1
  mixed = (top.global_seed * 1000003 + op_seed) & 0x7FFFFFFF
  return mixed, top.step


def MakeStepGenerator(device: Union[str, torch.device],
                      op_seed: Optional[int] = None) -> torch.Generator:
  s1, s2 = GenerateStepSeedPair(op_seed)
  g = torch.Generator(device=device)
  g.manual_seed((s1 * 2654435761 + s2) & 0x7FFFFFFFFFFFFFFF)
  return g


def DeterministicDropout(x: torch.Tensor, keep_prob: float,
                         op_seed: Optional[int] = None,
                         act: str = 'NONE') -> torch.Tensor:
  """Dropout reproducible under StepSeedScope (reference py_utils.py:3978).

  GPU: one fused HIP kernel (mask recomputed from the seed in backward);
  CPU: torch.Generator reference path. act in {'NONE','SWISH','RELU'}
  applies the activation INSIDE the same kernel (dropout(act(x))) —
  the FFN hidden tensor is read/written once instead of twice.
  """
  if keep_prob >= 1.0:
    if act == 'SWISH':
      return torch.nn.functional.silu(x)
    if act == 'RELU':
      return torch.relu(x)
    return x
  if x.is_cuda and x.numel() % 8 == 0:
    from lingvo_amd.ops import dropout as dropout_ops
    s1, _ = GenerateStepSeedPair(op_seed)
    # step-dependence comes from the device step-seed buffer (graph-safe)
    return dropout_ops.dropout(x, keep_prob, s1, act=act)
  if act == 'SWISH':
    x = torch.nn.functional.silu(x)
  elif act == 'RELU':
    x = torch.relu(x)
  g = MakeStepGenerator(x.device, op_seed)
  mask = (torch.rand(x.shape, generator=g, device=x.device,
                     dtype=torch.float32) < keep_prob)
  return x * mask.to(x.dtype) / keep_prob


def DeterministicDropoutAdd(x: torch.Tensor, keep_prob: float,
                            residual: torch.Tensor,
                            op_seed: Optional[int] = None,
                            scale: float = 1.0,
                            paddings: Optional[torch.Tensor] = None
                            ) -> torch.Tensor:
  """residual + dropout(x * scale * (1 - paddings)): one HIP kernel on
  GPU — the residual weight and ApplyPadding mask fuse into the same
  pass instead of separate elementwise kernels."""
  if keep_prob >= 1.0:
    if scale != 1.0:
      x = x * scale
    if paddings is not None:
      x = ApplyPadding(paddings, x)
    return residual + x
  if x.is_cuda and x.numel() % 8 == 0 and \
      (paddings is None or x.shape[-1] % 8 == 0):
    from lingvo_amd.ops import dropout as dropout_ops
    s1, _ = GenerateStepSeedPair(op_seed)
    return dropout_ops.dropout(x, keep_prob, s1, residual=residual,
                               scale=scale, paddings=paddings)
  if scale != 1.0:
    x = x * scale
  if paddings is not None:
    x = ApplyPadding(paddings, x)
  return residual + DeterministicDropout(x, keep_prob, op_seed)


# --------------------------------------------------------------------------
# Padded-sequence utilities ([B, T] paddings, 1.0 == padded)
# --------------------------------------------------------------------------
def PaddingsFromLengths(lengths: torch.Tensor, maxlen: int) -> torch.Tensor:
  """[B] lengths -> [B, maxlen] float paddings."""
  pos = torch.arange(maxlen, device=lengths.device)[None, :]
  return (pos >= lengths[:, None]).float()


def LengthsFromPaddings(paddings: torch.Tensor) -> torch.Tensor:
  """[B, T] paddings -> [B] int lengths (reference py_utils.py:4429)."""
  return (1.0 - paddings).sum(dim=1).round().long()


def ApplyPadding(padding: torch.Tensor, x: torch.Tensor,
                 padded_value: float = 0.0) -> torch.Tensor:
  """Zeroes (or sets) x where padding==1; padding broadcast against x."""
  while padding.dim() < x.dim():
    padding = padding.unsqueeze(-1)
  padding = padding.to(x.dtype)
  if padded_value == 0.0:
    return x * (1.0 - padding)
  return x * (1.0 - padding) + padded_value * padding


def PadSequenceDimension(x: torch.Tensor, length: int, pad_val: float = 0.0,
                         axis: int = 1) -> torch.Tensor:
  """Pads axis `axis` of x up to `length` (reference py_utils.py:4282)."""
  cur = x.shape[axis]
  if cur == length:
    return x
  if cur > length:
    raise ValueError(f'Cannot pad dim {axis} from {cur} down to {length}')
  pad_shape = list(x.shape)
  pad_shape[axis] = length - cur
  pad = torch.full(pad_shape, pad_val, dtype=x.dtype, device=x.device)
  return torch.cat([x, pad], dim=axis)


def ConcatenatePaddedSequences(x: torch.Tensor, y: torch.Tensor,
                               px: torch.Tensor, py: torch.Tensor
                               ) -> Tuple[torch.Tensor, torch.Tensor]:
  """Concats per-example valid segments of two padded [B,T,...] batches."""
  b = x.shape[0]
  lx = LengthsFromPaddings(px)
  ly = LengthsFromPaddings(py)
  tot = int((lx + ly).max().item())
  feat = x.shape[2:]
  out = torch.zeros((b, tot) + tuple(feat), dtype=x.dtype, device=x.device)
  pout = torch.ones((b, tot), dtype=px.dtype, device=px.device)
  for i in range(b):
    n1, n2 = int(lx[i]), int(ly[i])
    out[i, :n1] = x[i, :n1]
    out[i, n1:n1 + n2] = y[i, :n2]
    pout[i, :n1 + n2] = 0.0
  return out, pout


# --------------------------------------------------------------------------
# Numeric checks & misc
# --------------------------------------------------------------------------
def TrimTrailingPaddings(x: torch.Tensor, paddings: torch.Tensor):
  """Trims time steps that are padded in EVERY batch row
  (reference py_utils.py TrimTrailingPaddings). x [B, T, ...],
  paddings [B, T]; returns (trimmed_x, trimmed_paddings)."""
  lengths = LengthsFromPaddings(paddings)
  max_len = int(lengths.max().item()) if lengths.numel() else 0
  max_len = max(max_len, 1)
  return x[:, :max_len], paddings[:, :max_len]


def ReversePaddedSequence(x: torch.Tensor,
                          paddings: torch.Tensor) -> torch.Tensor:
  """Reverses the VALID prefix of each row, keeping padding in place
  (reference py_utils.py ReversePaddedSequence). x [B, T, ...]."""
  b, t = paddings.shape
  lengths = LengthsFromPaddings(paddings).to(torch.long)  # [B]
  pos = torch.arange(t, device=x.device).unsqueeze(0).expand(b, t)
  rev = (lengths.unsqueeze(1) - 1 - pos).clamp_min(0)
  idx = torch.where(pos < lengths.unsqueeze(1), rev, pos)
  shaped = idx.reshape(b, t, *([1] * (x.dim() - 2))).expand_as(x)
  return torch.gather(x, 1, shaped)


def ShiftLeft(x: torch.Tensor, shift: int,
              pad_val: float = 0.0) -> torch.Tensor:
  """Shifts [B, T, ...] left along time, padding the tail."""
  if shift <= 0:
    return x
  pad = x.new_full((x.shape[0], shift, *x.shape[2:]), pad_val)
  return torch.cat([x[:, shift:], pad], dim=1)


def MixByWeight(fns, weights, seed: Optional[int] = None):
  """Calls one of fns sampled by normalized weights; returns
  (result, index) (reference py_utils.py MixByWeight)."""
  import random as _random
  rng = _random.Random(seed)
  total = float(sum(weights))
  r = rng.uniform(0.0, total)
  acc = 0.0
  for i, (fn, w) in enumerate(zip(fns, weights)):
    acc += float(w)
    if r <= acc:
      return fn(), i
  return fns[-1](), len(fns) - 1


def SplitRecursively(x, num_splits: int, axis: int = -1):
  """Splits tensors (or NestedMaps/lists of them) into num_splits
  equal parts along axis (reference py_utils.py SplitRecursively).
  Returns a list of num_splits structures mirroring x."""
  if isinstance(x, torch.Tensor):
    assert x.shape[axis] % num_splits == 0
    return list(x.chunk(num_splits, dim=axis))
  if isinstance(x, (list, tuple)):
    split_elems = [SplitRecursively(e, num_splits, axis) for e in x]
    return [type(x)(parts[i] for parts in split_elems)
            for i in range(num_splits)]
  if isinstance(x, NestedMap):
    flat = x.Flatten()
    split_flat = [SplitRecursively(e, num_splits, axis) for e in flat]
    return [x.Pack([parts[i] for parts in split_flat])
            for i in range(num_splits)]
  raise TypeError('Unsupported type for SplitRecursively: %r' % type(x))


def CheckNumerics(x: torch.Tensor, message: str = '') -> torch.Tensor:
  """Raises if x contains NaN/Inf (reference py_utils.py:208)."""
  if not torch.isfinite(x).all():
    raise FloatingPointError(f'Tensor has NaN/Inf: {message}')
  return x


def HasNanOrInf(nmap_or_tensor) -> bool:
  if isinstance(nmap_or_tensor, torch.Tensor):
    return not bool(torch.isfinite(nmap_or_tensor).all())
  tensors = [v for v in nmap_or_tensor.Flatten()
             if isinstance(v, torch.Tensor) and v.is_floating_point()]
  return any(not bool(torch.isfinite(t).all()) for t in tensors)


def GlobalGradNorm(grads: List[torch.Tensor]) -> torch.Tensor:
  """L2 norm over a list of grads (reference learner.py:60-75 +
  SumSquared py_utils.py:4242). Multi-tensor via torch._foreach_norm
  (SURVEY K15)."""
  if not grads:
    return torch.zeros(())
  norms = torch._foreach_norm([g.detach() for g in grads], 2)
  return torch.stack([n.float() for n in norms]).norm(2)


def WeightedAvg(values: torch.Tensor, weights: torch.Tensor
                ) -> Tuple[torch.Tensor, torch.Tensor]:
  w = weights.float()
  total = w.sum()
  avg = (values.float() * w).sum() / total.clamp_min(1e-8)
  return avg, total


def WeightedAvgOfMetrics(metrics_list: List[NestedMap]) -> NestedMap:
  """Averages a list of {name: (value, weight)} metric maps
  (reference base_model.py:648)."""
  ret = NestedMap()
  if not metrics_list:
    return ret
  for name in metrics_list[0].keys():
    vals = torch.stack([torch.as_tensor(m[name][0]).float()
                        for m in metrics_list])
    wts = torch.stack([torch.as_tensor(m[name][1]).float()
                       for m in metrics_list])
    total_w = wts.sum()
    avg = (vals * wts).sum() / total_w.clamp_min(1e-8)
    ret[name] = (avg, total_w)
  return ret


def ToScalar(x) -> float:
  if isinstance(x, torch.Tensor):
    return float(x.detach().cpu().item())
  return float(x)


def AddVn(theta: NestedMap, vn_std: float) -> NestedMap:
  """Adds variational (Gaussian) noise to floating theta leaves
  (reference py_utils.py:3738 AddVN); deterministic per step scope."""

  def add_noise(t):
    if isinstance(t, torch.Tensor) and t.is_floating_point() and \
        t.requires_grad:
      u = GraphSafeUniform(t.shape, t.device).clamp(1e-6, 1 - 1e-6)
      normal = torch.erfinv(2 * u - 1) * math.sqrt(2.0)
      return t + vn_std * normal.to(t.dtype)
    return t

  return theta.Transform(add_noise)


def GraphSafeUniform(shape, device, op_seed: Optional[int] = None
                     ) -> torch.Tensor:
  """Uniform [0,1) tensor whose values vary per step via the device
  step-seed buffer — safe inside hipGraph capture (torch.Generator RNG
  is not). Deterministic in (global_seed, step, op_counter)."""
  import math as _math
  n = 1
  for d in shape:
    n *= int(d)
  s1, s2 = GenerateStepSeedPair(op_seed)
  if not (isinstance(device, torch.device) and device.type == 'cuda') and \
      str(device) != 'cuda' and not str(device).startswith('cuda'):
    g = torch.Generator()
    g.manual_seed((s1 * 2654435761 + s2) & 0x7FFFFFFFFFFFFFFF)
    return torch.rand(shape, generator=g)
  from lingvo_amd.ops import dropout as dropout_ops
  buf = dropout_ops._StepSeedBuf(torch.device(device))

  def lshr(x, k):  # logical shift right on int64 (>> sign-extends)
    return (x >> k) & ((1 << (64 - k)) - 1)

  # murmur3 fmix64, matching the dropout kernel's hash (hip/dropout.hip
  # hash_u32). The previous 2-round mixer left consecutive step seeds
  # correlated (~-0.24 over shared indices) — this one measures <0.01.
  idx = torch.arange(n, device=device, dtype=torch.int64)
  h = (idx * 0x9E3779B97F4A7C15) ^ (buf + s1)
  h = h ^ lshr(h, 33)
  h = h * -0xAE502812AA7333    # 0xFF51AFD7ED558CCD as signed int64
  h = h ^ lshr(h, 33)
  h = h * -0x3B314601E57A13AD  # 0xC4CEB9FE1A85EC53 as signed int64
  h = h ^ lshr(h, 33)
  u = (h & 0x7FFFFFFF).float() / float(1 << 31)
  return u.reshape(shape)


def MatmulBias(x: torch.Tensor, w: torch.Tensor,
               b: Optional[torch.Tensor] = None) -> torch.Tensor:
  """x @ w (+ b) with the bias fused into the hipBLASLt GEMM epilogue
  (torch.addmm) instead of a separate elementwise add kernel."""
  shape = x.shape
  x2 = x.reshape(-1, shape[-1])
  if b is None:
    out = torch.matmul(x2, w)
  else:
    out = torch.addmm(b, x2, w)
  return out.reshape(*shape[:-1], w.shape[-1])


class Timer:
  """Accumulating wall-clock timer (reference py_utils.py:6890)."""

  def __init__(self):
    self._total = 0.0
    self._start = None

  def Start(self):
    import time
    self._start = time.perf_counter()
    return self

  def Stop(self):
    import time
    if self._start is not None:
      self._total += time.perf_counter() - self._start
      self._start = None
    return self

  def Duration(self) -> float:
    return self._total

  def __enter__(self):
    return self.Start()

  def __exit__(self, *a):
    self.Stop()


def SinkhornAssignment(scores: torch.Tensor, tau: float = 0.1,
                       n_iters: int = 20) -> torch.Tensor:
  """Differentiable (soft) assignment via Sinkhorn normalization
  (reference lingvo/core/differentiable_assignment.py): iterated
  log-space row/column normalization of scores/tau converges to a
  doubly-stochastic matrix that approaches the argmax permutation as
  tau -> 0. scores [..., N, M] -> soft assignment of the same shape.
  """
  log_alpha = scores / tau
  for _ in range(n_iters):
    log_alpha = log_alpha - torch.logsumexp(log_alpha, dim=-1,
                                            keepdim=True)
    log_alpha = log_alpha - torch.logsumexp(log_alpha, dim=-2,
                                            keepdim=True)
  return torch.exp(log_alpha)


def AssertShapeMatch(tensor: torch.Tensor, pattern) -> torch.Tensor:
  """Shape assert (reference x_ops.cc:26 AssertShapeMatch): -1 entries
  are wildcards. Returns the tensor for chaining."""
  shape = tuple(tensor.shape)
  assert len(shape) == len(pattern), (shape, pattern)
  for got, want in zip(shape, pattern):
    assert want == -1 or got == want, (shape, pattern)
  return tensor


def AssertIdShape(*tensors: torch.Tensor) -> None:
  """All tensors share one shape (reference assert_kernels.cc)."""
  shapes = {tuple(t.shape) for t in tensors}
  assert len(shapes) == 1, shapes


def EstimateFlops(fn, *args) -> int:
  """Measured flop count of one call via the torch profiler
  (the runtime stand-in for the reference's symbolic FPropMeta /
  computation_cost.py estimates — counts real matmul/conv flops)."""
  from torch.profiler import profile, ProfilerActivity
  with profile(activities=[ProfilerActivity.CPU],
               with_flops=True) as prof:
    fn(*args)
  return int(sum(e.flops for e in prof.key_averages() if e.flops))


def RandomPermutationSequence(num: int, batch: int,
                              op_seed: Optional[int] = None
                              ) -> torch.Tensor:
  """Deterministic batch of random permutations of range(num)
  (reference x_ops random permutation sequence op): [batch, num] longs,
  reproducible under StepSeedScope."""
  keys = GraphSafeUniform((batch, num), 'cpu', op_seed)
  return keys.argsort(dim=1)


class CachedCall:
  """Caches a zero-arg callable's result (reference CachedCall op):
  the fn runs once; later calls return the cached tensor/value."""

  def __init__(self, fn):
    self._fn = fn
    self._has = False
    self._val = None

  def __call__(self):
    if not self._has:
      self._val = self._fn()
      self._has = True
    return self._val
