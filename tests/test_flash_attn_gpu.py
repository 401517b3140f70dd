"""GPU numerics tests: flash-attention HIP kernel vs fp32 torch reference.

MFMA fragment layout is verified first (mfma_probe) with random
asymmetric inputs — transpose-detecting per the CDNA guide methodology.
"""

import math

import pytest
import torch

gpu = pytest.mark.gpu


@gpu
def test_mfma_probe_layout():
  import lingvo_amd.ops._lingvo_ops as ext
  torch.manual_seed(0)
  a = torch.randn(16, 32, device='cuda').to(torch.bfloat16)
  b = torch.randn(32, 16, device='cuda').to(torch.bfloat16)
  bT = b.t().contiguous()  # K-contiguous layout the kernels use
  c = ext.mfma_probe(a, bT)
  ref = a.float() @ b.float()
  assert (c - ref).abs().max() < 0.1, (
      f'MFMA layout mismatch: max err {(c - ref).abs().max()}')


def _run_case(B, T, S, N, NKV, H, klen=None, bias=False, win_l=-1, win_r=-1,
              seed=0, tol=2e-2):
  from lingvo_amd.ops import flash_attn as fa
  torch.manual_seed(seed)
  q = torch.randn(B, T, N, H, device='cuda', dtype=torch.bfloat16)
  k = torch.randn(B, S, NKV, H, device='cuda', dtype=torch.bfloat16)
  v = torch.randn(B, S, NKV, H, device='cuda', dtype=torch.bfloat16)
  kl = None
  if klen is not None:
    kl = torch.tensor(klen, device='cuda', dtype=torch.int32)
  bias_t = None
  if bias:
    bias_t = torch.randn(N, 255, device='cuda') * 0.5
  out = fa.flash_attention(q, k, v, kl, bias_t, win_l, win_r, 127)
  ref = fa._ref_attention(q, k, v, kl, bias_t, win_l, win_r, 127,
                          1.0 / math.sqrt(H))
  err = (out.float() - ref).abs().max().item()
  assert err < tol, f'fwd max err {err}'
  return q, k, v, kl, bias_t


@gpu
@pytest.mark.parametrize('case', [
    dict(B=2, T=128, S=128, N=4, NKV=4, H=64),
    dict(B=1, T=200, S=200, N=2, NKV=2, H=64),       # non-multiple of 64
    dict(B=2, T=128, S=128, N=4, NKV=4, H=128),
    dict(B=2, T=128, S=128, N=4, NKV=4, H=64, win_r=0),   # causal
    dict(B=2, T=192, S=192, N=2, NKV=2, H=64, win_l=40, win_r=20),  # local
    dict(B=2, T=128, S=128, N=4, NKV=1, H=64),       # MQA
    dict(B=2, T=128, S=128, N=4, NKV=2, H=64),       # GQA
    dict(B=2, T=96, S=160, N=2, NKV=2, H=64),        # cross attention
    dict(B=2, T=128, S=128, N=2, NKV=2, H=64, klen=[100, 128]),  # padding
    dict(B=1, T=128, S=128, N=2, NKV=2, H=64, bias=True),  # rel bias
    dict(B=1, T=128, S=128, N=2, NKV=2, H=64, bias=True, win_l=64,
         win_r=0),
])
def test_flash_fwd(case):
  _run_case(**case)


@gpu
@pytest.mark.parametrize('case', [
    dict(B=2, T=128, S=128, N=2, NKV=2, H=64),
    dict(B=1, T=200, S=200, N=2, NKV=2, H=64, win_r=0),
    dict(B=2, T=128, S=128, N=2, NKV=2, H=128),
    dict(B=2, T=128, S=128, N=4, NKV=2, H=64),       # GQA
    dict(B=2, T=128, S=128, N=2, NKV=2, H=64, klen=[77, 128]),
    dict(B=1, T=128, S=128, N=2, NKV=2, H=64, bias=True, win_l=48,
         win_r=0),
])
def test_flash_bwd(case):
  from lingvo_amd.ops import flash_attn as fa
  torch.manual_seed(3)
  B, T, S, N, NKV, H = (case['B'], case['T'], case['S'], case['N'],
                        case['NKV'], case['H'])
  q = torch.randn(B, T, N, H, device='cuda',
                  dtype=torch.bfloat16).requires_grad_(True)
  k = torch.randn(B, S, NKV, H, device='cuda',
                  dtype=torch.bfloat16).requires_grad_(True)
  v = torch.randn(B, S, NKV, H, device='cuda',
                  dtype=torch.bfloat16).requires_grad_(True)
  kl = None
  if case.get('klen'):
    kl = torch.tensor(case['klen'], device='cuda', dtype=torch.int32)
  bias_t = None
  if case.get('bias'):
    bias_t = (torch.randn(N, 255, device='cuda') * 0.3).requires_grad_(True)
  win_l = case.get('win_l', -1)
  win_r = case.get('win_r', -1)

  out = fa.flash_attention(q, k, v, kl, bias_t, win_l, win_r, 127)
  g = torch.randn_like(out)
  out.backward(g)

  qr = q.detach().float().requires_grad_(True)
  kr = k.detach().float().requires_grad_(True)
  vr = v.detach().float().requires_grad_(True)
  br = None
  if bias_t is not None:
    br = bias_t.detach().clone().requires_grad_(True)
  ref = fa._ref_attention(qr, kr, vr, kl, br, win_l, win_r, 127,
                          1.0 / math.sqrt(H))
  ref.backward(g.float())

  for name, got, want in [('dq', q.grad, qr.grad), ('dk', k.grad, kr.grad),
                          ('dv', v.grad, vr.grad)]:
    err = (got.float() - want).abs().max().item()
    rel = err / max(1e-3, want.abs().max().item())
    assert rel < 0.06, f'{name} max rel err {rel} (abs {err})'
  if bias_t is not None:
    err = (bias_t.grad.float() - br.grad).abs().max().item()
    rel = err / max(1e-3, br.grad.abs().max().item())
    assert rel < 0.06, f'dbias rel err {rel}'


def test_cpu_reference_self_consistent():
  """CPU path smoke (runs in the no-GPU container)."""
  from lingvo_amd.ops import flash_attn as fa
  q = torch.randn(2, 16, 2, 64)
  k = torch.randn(2, 16, 2, 64)
  v = torch.randn(2, 16, 2, 64)
  out = fa.flash_attention(q, k, v, win_r=0)
  assert out.shape == q.shape
  # causal: first position attends only to key 0
  ref0 = v[:, 0]
  assert torch.allclose(out[:, 0], ref0, atol=1e-4)


@gpu
def test_flash_segment_mask_fwd_bwd():
  """Packed-input segment masking matches the fp32 reference."""
  from lingvo_amd.ops import flash_attn as fa
  torch.manual_seed(11)
  B, T, N, H = 2, 128, 2, 64
  q = torch.randn(B, T, N, H, device='cuda',
                  dtype=torch.bfloat16).requires_grad_(True)
  k = torch.randn(B, T, N, H, device='cuda',
                  dtype=torch.bfloat16).requires_grad_(True)
  v = torch.randn(B, T, N, H, device='cuda',
                  dtype=torch.bfloat16).requires_grad_(True)
  # 3 segments per row (packed layout)
  seg = torch.zeros(B, T, dtype=torch.int32, device='cuda')
  seg[:, 40:90] = 1
  seg[:, 90:] = 2
  out = fa.flash_attention(q, k, v, win_r=0, q_segment_ids=seg,
                           k_segment_ids=seg)
  g = torch.randn_like(out)
  out.backward(g)

  qr = q.detach().float().requires_grad_(True)
  kr = k.detach().float().requires_grad_(True)
  vr = v.detach().float().requires_grad_(True)
  import math as m
  ref = fa._ref_attention(qr, kr, vr, None, None, -1, 0, 127,
                          1.0 / m.sqrt(H), seg, seg)
  ref.backward(g.float())
  assert (out.float() - ref.detach()).abs().max() < 2e-2
  for got, want in [(q.grad, qr.grad), (k.grad, kr.grad),
                    (v.grad, vr.grad)]:
    rel = (got.float() - want).abs().max() / max(1e-3,
                                                 want.abs().max().item())
    assert rel < 0.06
  # cross-segment independence: mutating segment 2 keys leaves
  # segment-0 outputs unchanged
  k2 = k.detach().clone()
  k2[:, 95:] = 3.0
  out2 = fa.flash_attention(q.detach(), k2, v.detach(), win_r=0,
                            q_segment_ids=seg, k_segment_ids=seg)
  assert torch.equal(out[:, :40].detach(), out2[:, :40])


@gpu
def test_flash_chunk_mask_matches_ref():
  """Native chunk-mask mode vs the fp32 reference (fwd + grads)."""
  from lingvo_amd.ops import flash_attn as fa
  torch.manual_seed(6)
  B, T, N, H = 3, 200, 2, 64
  for chunk, lc, causal in [(64, 0, False), (64, 1, True), (48, 2, False)]:
    q = torch.randn(B, T, N, H, device='cuda',
                    dtype=torch.bfloat16).requires_grad_()
    k = torch.randn_like(q).requires_grad_()
    v = torch.randn_like(q).requires_grad_()
    klen = torch.tensor([200, 150, 90], device='cuda',
                        dtype=torch.int32)
    win_r = 0 if causal else -1
    out = fa.flash_attention(q, k, v, klen=klen, win_r=win_r,
                             chunk_size=chunk, left_chunks=lc)
    g = torch.randn_like(out)
    out.backward(g)
    grads = [q.grad.float().clone(), k.grad.float().clone(),
             v.grad.float().clone()]
    q.grad = k.grad = v.grad = None

    ref = fa._ref_attention(q.detach().float().requires_grad_(),
                            k.detach().float().requires_grad_(),
                            v.detach().float().requires_grad_(),
                            klen, None, -1, win_r, 127, H ** -0.5,
                            chunk_size=chunk, left_chunks=lc)
    assert (out.float() - ref.detach()).abs().max() < 0.05, (chunk, lc)
    qr = q.detach().float().requires_grad_()
    kr = k.detach().float().requires_grad_()
    vr = v.detach().float().requires_grad_()
    fa._ref_attention(qr, kr, vr, klen, None, -1, win_r, 127,
                      H ** -0.5, chunk_size=chunk,
                      left_chunks=lc).backward(g.float())
    for got, want in zip(grads, [qr.grad, kr.grad, vr.grad]):
      assert (got - want).abs().max() < 0.1, (chunk, lc)
