"""LM step breakdown + flash-attn microbench for the LM shape."""
import sys, time, torch
sys.path.insert(0, '/root/repo')
from lingvo_amd.core import registry
from lingvo_amd.ops import flash_attn as fa

def timeit(fn, n=10):
  for _ in range(3): fn()
  torch.cuda.synchronize(); t0 = time.perf_counter()
  for _ in range(n): fn()
  torch.cuda.synchronize(); return (time.perf_counter() - t0) / n * 1000

# flash attn microbench at LM shape: B=8 T=1024 N=16 H=128
B, T, N, H = 8, 1024, 16, 128
q = torch.randn(B, T, N, H, device='cuda', dtype=torch.bfloat16)
k, v = torch.randn_like(q), torch.randn_like(q)
fwd_ms = timeit(lambda: fa.flash_attention(q, k, v, win_r=0))
flops = 4 * B * N * T * T * H / 2  # causal halves it
print(f'fa fwd causal B8 T1024 N16 H128: {fwd_ms:.2f} ms = {flops/fwd_ms/1e9:.0f} TF')
qg = q.clone().requires_grad_(True); kg = k.clone().requires_grad_(True); vg = v.clone().requires_grad_(True)
def fb():
  out = fa.flash_attention(qg, kg, vg, win_r=0)
  out.backward(torch.ones_like(out))
  qg.grad = kg.grad = vg.grad = None
fb_ms = timeit(fb, n=5)
print(f'fa fwd+bwd: {fb_ms:.2f} ms = {3.5*flops/fb_ms/1e9:.0f} TF-equiv')

# conformer-shape attn: B=64 T=300 N=8 H=64 with bias
B2, T2, N2, H2 = 64, 300, 8, 64
q2 = torch.randn(B2, T2, N2, H2, device='cuda', dtype=torch.bfloat16)
k2, v2 = torch.randn_like(q2), torch.randn_like(q2)
bias = torch.randn(N2, 255, device='cuda')
f2 = timeit(lambda: fa.flash_attention(q2, k2, v2, None, bias))
fl2 = 4 * B2 * N2 * T2 * T2 * H2
print(f'fa fwd conformer-shape: {f2:.2f} ms = {fl2/f2/1e9:.0f} TF')

# LM phases
model_p = registry.GetParams('lm.one_billion_wds.OneBWdsTransformerLm', 'Train')
model_p.task.random_seed = 1
model = model_p.Instantiate().to('cuda:0')
task = model.GetTask()
task.MaybeConvertBf16Weights()
b = task.GetInputBatch().Transform(lambda t: t.to('cuda:0') if isinstance(t, torch.Tensor) else t)
from lingvo_amd.core import py_utils
th = task.theta
def fwd():
  with py_utils.StepSeedScope(1, 0):
    m, _ = task.FProp(th, b)
  return m
print(f'LM full fwd: {timeit(fwd, n=5):.1f} ms')
act = task.lm.FProp(th.lm, b.ids, b.paddings)
print(f'LM stack+emb fwd only: {timeit(lambda: task.lm.FProp(th.lm, b.ids, b.paddings), n=5):.1f} ms')
act = act.detach()
print(f'LM softmax xent fwd: {timeit(lambda: task.lm.XentLoss(th.lm, act, b.labels, b.weights), n=5):.1f} ms')
def step():
  task.TrainStep(b)
print(f'LM full train step (eager): {timeit(step, n=5):.1f} ms')
