"""GPU numerics: fused softmax-xent vs torch cross_entropy."""

import pytest
import torch
import torch.nn.functional as F

gpu = pytest.mark.gpu


@gpu
@pytest.mark.parametrize('rv', [(64, 1024), (128, 32000), (37, 1000)])
def test_logits_xent_fwd_bwd(rv):
  from lingvo_amd.ops import softmax_xent
  rows, v = rv
  torch.manual_seed(0)
  d = 128
  x = torch.randn(rows, d, device='cuda',
                  dtype=torch.bfloat16).requires_grad_(True)
  w = (torch.randn(d, v, device='cuda', dtype=torch.bfloat16) *
       0.05).requires_grad_(True)
  b = torch.zeros(v, device='cuda', dtype=torch.bfloat16,
                  requires_grad=True)
  labels = torch.randint(0, v, (rows,), device='cuda')
  loss = softmax_xent.logits_xent(x, w, b, labels)
  gw = torch.rand(rows, device='cuda')
  (loss * gw).sum().backward()

  xr = x.detach().float().requires_grad_(True)
  wr = w.detach().float().requires_grad_(True)
  br = b.detach().float().requires_grad_(True)
  logits = xr @ wr + br
  ref = F.cross_entropy(logits, labels, reduction='none')
  (ref * gw).sum().backward()

  assert (loss - ref).abs().max() < 0.03
  for got, want in [(x.grad, xr.grad), (w.grad, wr.grad), (b.grad, br.grad)]:
    scale = max(0.1, float(want.abs().max()))
    assert (got.float() - want).abs().max() / scale < 0.05
