"""GPU numerics: depthwise conv1d HIP kernel vs CPU torch reference."""

import pytest
import torch

gpu = pytest.mark.gpu


@gpu
@pytest.mark.parametrize('case', [
    dict(B=2, T=50, D=64, K=3, causal=True),
    dict(B=2, T=100, D=512, K=32, causal=False),
    dict(B=1, T=37, D=256, K=31, causal=True),
])
def test_dwconv1d_fwd_bwd(case):
  from lingvo_amd.ops import conv1d as conv_ops
  torch.manual_seed(0)
  B, T, D, K = case['B'], case['T'], case['D'], case['K']
  x = torch.randn(B, T, D, device='cuda',
                  dtype=torch.bfloat16).requires_grad_(True)
  w = (torch.randn(K, D, device='cuda') * 0.3).requires_grad_(True)
  bias = torch.randn(D, device='cuda').requires_grad_(True)
  y = conv_ops.depthwise_conv1d(x, w, bias, causal=case['causal'])
  g = torch.randn_like(y)
  y.backward(g)

  xr = x.detach().float().cpu().requires_grad_(True)
  wr = w.detach().float().cpu().requires_grad_(True)
  br = bias.detach().float().cpu().requires_grad_(True)
  yr = conv_ops.depthwise_conv1d(xr, wr, br, causal=case['causal'])
  yr.backward(g.float().cpu())

  assert (y.float().cpu() - yr).abs().max() < 0.05
  assert (x.grad.float().cpu() - xr.grad).abs().max() < 0.05
  scale = max(1.0, float(wr.grad.abs().max()))
  assert (w.grad.cpu() - wr.grad).abs().max() / scale < 0.02
  assert (bias.grad.cpu() - br.grad).abs().max() / \
      max(1.0, float(br.grad.abs().max())) < 0.02
