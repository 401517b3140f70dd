"""Flash-attention microbenchmark at Conformer-L bench shapes.

  python tools/fa_bench.py [--shape conformer|lm] [--iters 50]

Times fwd and bwd separately (bwd via retained graph + backward on a
fixed grad), prints achieved TFLOP/s against the attention flop count
2*B*N*T*S*H*2 (QK^T + PV) per direction (bwd ~2.5x fwd flops).
"""

from __future__ import annotations

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                '..'))

import torch


def bench(fn, iters, warmup=10):
  for _ in range(warmup):
    fn()
  torch.cuda.synchronize()
  t0 = time.perf_counter()
  for _ in range(iters):
    fn()
  torch.cuda.synchronize()
  return (time.perf_counter() - t0) / iters


def main():
  ap = argparse.ArgumentParser()
  ap.add_argument('--shape', default='conformer',
                  choices=['conformer', 'lm', 'long'])
  ap.add_argument('--iters', type=int, default=50)
  ap.add_argument('--nobias', action='store_true')
  args = ap.parse_args()
  from lingvo_amd.ops import _loader
  from lingvo_amd.ops.flash_attn import flash_attention
  ext = _loader.get_ext(required=True)

  if args.shape == 'conformer':
    B, T, N, H, clip = 128, 300, 8, 64, 127
  elif args.shape == 'lm':
    B, T, N, H, clip = 16, 1024, 16, 128, 127
  else:
    B, T, N, H, clip = 8, 4096, 16, 64, 127

  g = torch.Generator(device='cuda').manual_seed(3)
  q = torch.randn(B, T, N, H, device='cuda', dtype=torch.bfloat16,
                  generator=g)
  k = torch.randn_like(q)
  v = torch.randn_like(q)
  klen = torch.randint(int(0.8 * T), T + 1, (B,), device='cuda',
                       generator=g).to(torch.int32)
  bias = (None if args.nobias else
          torch.randn(N, 2 * clip + 1, device='cuda',
                      dtype=torch.bfloat16, generator=g))
  scale = H ** -0.5

  flops_dir = 4.0 * B * N * T * T * H  # QK^T + PV (2 GEMMs, 2 flops/MAC)

  o, lse = ext.fa_fwd(q, k, v, klen, bias, None, None, -1, -1, clip,
                      scale, 0, 0)
  dout = torch.randn_like(o)

  t_fwd = bench(lambda: ext.fa_fwd(q, k, v, klen, bias, None, None, -1,
                                   -1, clip, scale, 0, 0), args.iters)
  t_bwd = bench(lambda: ext.fa_bwd(dout, q, k, v, o, lse, klen, bias,
                                   None, None, bias is not None, -1, -1,
                                   clip, scale, 0, 0), args.iters)
  print(f'shape={args.shape} B={B} T={T} N={N} H={H} '
        f'bias={"clip" + str(clip) if bias is not None else "none"}')
  print(f'fwd: {t_fwd * 1e3:8.3f} ms  {flops_dir / t_fwd / 1e12:7.1f} TF/s')
  print(f'bwd: {t_bwd * 1e3:8.3f} ms  '
        f'{2.5 * flops_dir / t_bwd / 1e12:7.1f} TF/s '
        f'(bwd/fwd = {t_bwd / t_fwd:.2f}x)')


if __name__ == '__main__':
  main()
