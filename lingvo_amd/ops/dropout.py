"""Fused deterministic dropout wrapper (seed-only state).

The mask is a function of (op_seed, step_seed, element index). op_seed is
baked into the kernel launch; step_seed lives in a small device buffer so
the SAME captured hipGraph draws fresh masks every replay — the buffer is
rewritten once per step by SetStepSeed (called from StepSeedScope).
"""

from __future__ import annotations

from typing import Dict, Optional

import torch

from lingvo_amd.ops import _loader

_STEP_SEED_BUFS: Dict[int, torch.Tensor] = {}


def _StepSeedBuf(device: torch.device) -> torch.Tensor:
  idx = device.index or 0
  if idx not in _STEP_SEED_BUFS:
    _STEP_SEED_BUFS[idx] = torch.zeros(1, dtype=torch.int64,
                                       device=device)
  return _STEP_SEED_BUFS[idx]


def SetStepSeed(step: int, global_seed: int = 0) -> None:
  """Updates the device step-seed buffers (cheap; once per train step)."""
  if not torch.cuda.is_available():
    return
  val = ((step + 1) * 0x9E3779B97F4A7C15 ^ (global_seed * 2654435761)) \
      & 0x7FFFFFFFFFFFFFFF
  for buf in _STEP_SEED_BUFS.values():
    buf.fill_(val)
  if not _STEP_SEED_BUFS:
    dev = torch.device('cuda', torch.cuda.current_device())
    _StepSeedBuf(dev).fill_(val)


ACT_CODES = {'NONE': 0, 'SWISH': 1, 'SILU': 1, 'RELU': 2}


class _DropoutFn(torch.autograd.Function):

  @staticmethod
  def forward(ctx, x, residual, seed, keep, act, scale, pad, d):
    ext = _loader.get_ext(required=True)
    buf = _StepSeedBuf(x.device)
    y = ext.dropout_fwd(x, residual, seed, buf, keep, act, scale, pad, d)
    ctx.seed = seed
    ctx.keep = keep
    ctx.act = act
    ctx.scale = scale
    ctx.d = d
    ctx.has_res = residual is not None
    ctx.dev = x.device
    saved = []
    if act:
      saved.append(x)
    ctx.has_pad = pad is not None
    if pad is not None:
      saved.append(pad)
    ctx.save_for_backward(*saved)
    return y

  @staticmethod
  def backward(ctx, dy):
    ext = _loader.get_ext(required=True)
    dy = dy.contiguous()
    saved = list(ctx.saved_tensors)
    x = saved.pop(0) if ctx.act else None
    pad = saved.pop(0) if ctx.has_pad else None
    dx = ext.dropout_bwd(dy, x, ctx.seed, _StepSeedBuf(ctx.dev), ctx.keep,
                         ctx.act, ctx.scale, pad, ctx.d)
    dres = dy if ctx.has_res else None
    return dx, dres, None, None, None, None, None, None


def dropout(x: torch.Tensor, keep_prob: float, seed: int,
            residual: Optional[torch.Tensor] = None,
            act: str = 'NONE', scale: float = 1.0,
            paddings: Optional[torch.Tensor] = None) -> torch.Tensor:
  """y = dropout(act(x) * scale * (1 - pad)) (+ residual). One HIP
  kernel each way on GPU bf16 — the FFN activation, residual weight
  and ApplyPadding mask all fuse into the same pass. act in
  {'NONE','SWISH','RELU'}; residual excludes act. paddings is [B, T]
  (or [rows]) with 1.0 == padded, row = flat_index // D where
  D = x.shape[-1] (D % 8 == 0)."""
  act_code = ACT_CODES[act.upper()]
  assert not (act_code and residual is not None)
  d = x.shape[-1]
  if (x.is_cuda and x.numel() % 8 == 0 and
      (paddings is None or d % 8 == 0)):
    orig = x.dtype
    pad = None
    if paddings is not None:
      pad = paddings.to(torch.bfloat16).contiguous()
      assert pad.numel() * d == x.numel()
    y = _DropoutFn.apply(
        x.to(torch.bfloat16).contiguous(),
        None if residual is None else
        residual.to(torch.bfloat16).contiguous(), seed, keep_prob,
        act_code, scale, pad, d)
    return y.to(orig) if orig != torch.bfloat16 else y
  if act_code == 1:
    x = torch.nn.functional.silu(x)
  elif act_code == 2:
    x = torch.relu(x)
  if scale != 1.0:
    x = x * scale
  if paddings is not None:
    x = x * (1.0 - paddings.to(x.dtype)).reshape(
        *paddings.shape, *([1] * (x.dim() - paddings.dim())))
  g = torch.Generator(device=x.device)
  g.manual_seed(seed & 0x7FFFFFFFFFFFFFFF)
  mask = (torch.rand(x.shape, generator=g, device=x.device,
                     dtype=torch.float32) < keep_prob)
  y = x * mask.to(x.dtype) / keep_prob
  return y if residual is None else y + residual
