// Fused LSTM gate pointwise math for gfx950 (reference cell
// math: lingvo/core/rnn_cell.py:213 LSTMCellSimple).
//
//   i=sig(g1), f=sig(g2+fgb), o=sig(g3), cand=tanh(g0)
//   c1 = clamp(f*c0 + i*cand, +-cap);  m1 = o * tanh(c1)
//
// One kernel fwd / one bwd (recomputing activations from saved gates)
// replaces the ~12 eager pointwise kernels per step of the teacher-
// forced decoder loop. Gate order matches LSTMCellSimple: [i_i(cand),
// i_g(i), f_g(f), o_g(o)].

#include <ATen/cuda/CUDAContext.h>
#include <torch/extension.h>

#include "common.h"

namespace {

__device__ __forceinline__ float sigf(float x) {
  return 1.f / (1.f + __expf(-x));
}

__global__ void lstm_gates_fwd(const unsigned short* __restrict__ gates,
                               const unsigned short* __restrict__ c0,
                               unsigned short* __restrict__ c1,
                               unsigned short* __restrict__ m1, long n,
                               int h, float fgb, float cap) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    const long row = i / h;
    const long col = i % h;
    const unsigned short* gr = gates + row * 4 * h;
    float cand = tanhf(bf16_bits_to_float(gr[col]));
    float ig = sigf(bf16_bits_to_float(gr[h + col]));
    float fg = sigf(bf16_bits_to_float(gr[2 * h + col]) + fgb);
    float og = sigf(bf16_bits_to_float(gr[3 * h + col]));
    float c = fg * bf16_bits_to_float(c0[i]) + ig * cand;
    if (cap > 0.f) c = fminf(cap, fmaxf(-cap, c));
    c1[i] = float_to_bf16_bits(c);
    m1[i] = float_to_bf16_bits(og * tanhf(c));
  }
}

__global__ void lstm_gates_bwd(const unsigned short* __restrict__ gates,
                               const unsigned short* __restrict__ c0,
                               const unsigned short* __restrict__ c1,
                               const unsigned short* __restrict__ dm1,
                               const unsigned short* __restrict__ dc1_ext,
                               unsigned short* __restrict__ dgates,
                               unsigned short* __restrict__ dc0, long n,
                               int h, float fgb, float cap) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    const long row = i / h;
    const long col = i % h;
    const unsigned short* gr = gates + row * 4 * h;
    float cand = tanhf(bf16_bits_to_float(gr[col]));
    float ig = sigf(bf16_bits_to_float(gr[h + col]));
    float fg = sigf(bf16_bits_to_float(gr[2 * h + col]) + fgb);
    float og = sigf(bf16_bits_to_float(gr[3 * h + col]));
    float c0v = bf16_bits_to_float(c0[i]);
    float cv = bf16_bits_to_float(c1[i]);  // post-clamp value
    float tc = tanhf(cv);
    float dm = bf16_bits_to_float(dm1[i]);
    float dc = dc1_ext ? bf16_bits_to_float(dc1_ext[i]) : 0.f;
    float do_ = dm * tc;
    dc += dm * og * (1.f - tc * tc);
    if (cap > 0.f) {
      float pre = fg * c0v + ig * cand;  // pre-clamp
      if (pre > cap || pre < -cap) dc = 0.f;
    }
    float di = dc * cand;
    float dcand = dc * ig;
    float df = dc * c0v;
    unsigned short* dgr = dgates + row * 4 * h;
    dgr[col] = float_to_bf16_bits(dcand * (1.f - cand * cand));
    dgr[h + col] = float_to_bf16_bits(di * ig * (1.f - ig));
    dgr[2 * h + col] = float_to_bf16_bits(df * fg * (1.f - fg));
    dgr[3 * h + col] = float_to_bf16_bits(do_ * og * (1.f - og));
    dc0[i] = float_to_bf16_bits(dc * fg);
  }
}

}  // namespace

std::vector<torch::Tensor> lstm_gates_fwd_op(torch::Tensor gates,
                                             torch::Tensor c0, double fgb,
                                             double cap) {
  TORCH_CHECK(gates.is_cuda() && gates.is_contiguous() &&
              gates.scalar_type() == torch::kBFloat16);
  const long n = c0.numel();
  const int h = c0.size(-1);
  auto c1 = torch::empty_like(c0);
  auto m1 = torch::empty_like(c0);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(lstm_gates_fwd, dim3(memory_bound_grid(n, 256)),
                     dim3(256), 0, stream,
                     (const unsigned short*)gates.data_ptr(),
                     (const unsigned short*)c0.data_ptr(),
                     (unsigned short*)c1.data_ptr(),
                     (unsigned short*)m1.data_ptr(), n, h, (float)fgb,
                     (float)cap);
  return {c1, m1};
}

std::vector<torch::Tensor> lstm_gates_bwd_op(
    torch::Tensor gates, torch::Tensor c0, torch::Tensor c1,
    torch::Tensor dm1, c10::optional<torch::Tensor> dc1, double fgb,
    double cap) {
  const long n = c0.numel();
  const int h = c0.size(-1);
  auto dgates = torch::empty_like(gates);
  auto dc0 = torch::empty_like(c0);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(lstm_gates_bwd, dim3(memory_bound_grid(n, 256)),
                     dim3(256), 0, stream,
                     (const unsigned short*)gates.data_ptr(),
                     (const unsigned short*)c0.data_ptr(),
                     (const unsigned short*)c1.data_ptr(),
                     (const unsigned short*)dm1.data_ptr(),
                     dc1.has_value()
                         ? (const unsigned short*)dc1->data_ptr()
                         : nullptr,
                     (unsigned short*)dgates.data_ptr(),
                     (unsigned short*)dc0.data_ptr(), n, h, (float)fgb,
                     (float)cap);
  return {dgates, dc0};
}
