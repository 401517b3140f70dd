// Depthwise 1-D time convolution fwd+bwd for gfx950 (SURVEY.md K7;
// reference op: lingvo/core/conv_layers_with_time_padding.py:608
// DepthwiseConv2DLayer / :717 CausalDepthwiseConv2DLayer with time-major
// paddings).
//
// x: [B, T, D] bf16 contiguous, w: [K, D] bf16, y: [B, T, D].
//   y[b,t,d] = sum_j x[b, t + j - pad, d] * w[j, d]   (+ bias[d])
// pad = K-1 for causal, (K-1)/2 for SAME. Out-of-range taps read 0.
// Inputs are pre-masked by paddings on the Python side; outputs are
// re-masked there too.
//
// Memory-bound: vectorized ushort8 IO, one thread per 8 channels of one
// (b, t) row, grid-stride; taps hit L1/L2 (row re-use across t).

#include <ATen/cuda/CUDAContext.h>
#include <torch/extension.h>

#include "common.h"

namespace {

constexpr int MAXK = 33;

__global__ void dwconv_fwd(const unsigned short* __restrict__ x,
                           const unsigned short* __restrict__ w,
                           const unsigned short* __restrict__ bias,
                           unsigned short* __restrict__ y, int B, int T,
                           int D, int K, int pad) {
  const long nvec = (long)B * T * (D / 8);
  const int dvec = D / 8;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long)gridDim.x * blockDim.x) {
    const int dv = (int)(i % dvec);
    const long bt = i / dvec;
    const int t = (int)(bt % T);
    const long b = bt / T;
    float acc[8];
    if (bias) {
      ushortx8 bv = *reinterpret_cast<const ushortx8*>(bias + dv * 8);
#pragma unroll
      for (int e = 0; e < 8; ++e) acc[e] = bf16_bits_to_float(bv[e]);
    } else {
#pragma unroll
      for (int e = 0; e < 8; ++e) acc[e] = 0.f;
    }
    for (int j = 0; j < K; ++j) {
      int ts = t + j - pad;
      if (ts < 0 || ts >= T) continue;
      ushortx8 xv = *reinterpret_cast<const ushortx8*>(
          x + (b * T + ts) * D + dv * 8);
      ushortx8 wv = *reinterpret_cast<const ushortx8*>(w + j * D + dv * 8);
#pragma unroll
      for (int e = 0; e < 8; ++e)
        acc[e] += bf16_bits_to_float(xv[e]) * bf16_bits_to_float(wv[e]);
    }
    ushortx8 ov;
#pragma unroll
    for (int e = 0; e < 8; ++e) ov[e] = float_to_bf16_bits(acc[e]);
    *reinterpret_cast<ushortx8*>(y + (b * T + t) * D + dv * 8) = ov;
  }
}

__global__ void dwconv_bwd_dx(const unsigned short* __restrict__ dy,
                              const unsigned short* __restrict__ w,
                              unsigned short* __restrict__ dx, int B, int T,
                              int D, int K, int pad) {
  const long nvec = (long)B * T * (D / 8);
  const int dvec = D / 8;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long)gridDim.x * blockDim.x) {
    const int dv = (int)(i % dvec);
    const long bt = i / dvec;
    const int t = (int)(bt % T);
    const long b = bt / T;
    float acc[8];
#pragma unroll
    for (int e = 0; e < 8; ++e) acc[e] = 0.f;
    for (int j = 0; j < K; ++j) {
      int ty = t - j + pad;  // y position whose tap j touched x[t]
      if (ty < 0 || ty >= T) continue;
      ushortx8 gv = *reinterpret_cast<const ushortx8*>(
          dy + (b * T + ty) * D + dv * 8);
      ushortx8 wv = *reinterpret_cast<const ushortx8*>(w + j * D + dv * 8);
#pragma unroll
      for (int e = 0; e < 8; ++e)
        acc[e] += bf16_bits_to_float(gv[e]) * bf16_bits_to_float(wv[e]);
    }
    ushortx8 ov;
#pragma unroll
    for (int e = 0; e < 8; ++e) ov[e] = float_to_bf16_bits(acc[e]);
    *reinterpret_cast<ushortx8*>(dx + (b * T + t) * D + dv * 8) = ov;
  }
}

// dw[j,d] = sum_{b,t} dy[b,t,d] * x[b,t+j-pad,d]; db[d] = sum dy.
// One thread per channel d within a (d-block, row-stripe) grid; K+1 fp32
// register accumulators; one atomicAdd per (thread, tap) at the end.
__global__ void dwconv_bwd_dw(const unsigned short* __restrict__ dy,
                              const unsigned short* __restrict__ x,
                              float* __restrict__ dw_acc,
                              float* __restrict__ db_acc, int B, int T,
                              int D, int K, int pad, int nstripes) {
  const int d = blockIdx.x * blockDim.x + threadIdx.x;
  if (d >= D) return;
  const int stripe = blockIdx.y;
  const long rows = (long)B * T;
  // Contiguous row chunks: consecutive iterations share K-1 of the K
  // x-taps, so the taps stay L1-resident.
  const long chunk = (rows + nstripes - 1) / nstripes;
  const long r0 = stripe * chunk;
  const long r1 = min(rows, r0 + chunk);
  float dw[MAXK];
  for (int j = 0; j < K; ++j) dw[j] = 0.f;
  float db = 0.f;
  for (long r = r0; r < r1; ++r) {
    const int t = (int)(r % T);
    const long b = r / T;
    float g = bf16_bits_to_float(dy[r * D + d]);
    db += g;
    const long xbase = (b * T) * (long)D + d;
    for (int j = 0; j < K; ++j) {
      int ts = t + j - pad;
      float xv = (ts < 0 || ts >= T)
                     ? 0.f
                     : bf16_bits_to_float(x[xbase + (long)ts * D]);
      dw[j] += g * xv;
    }
  }
  for (int j = 0; j < K; ++j) {
    if (dw[j] != 0.f) atomicAdd(dw_acc + (long)j * D + d, dw[j]);
  }
  if (db != 0.f) atomicAdd(db_acc + d, db);
}

}  // namespace

torch::Tensor dwconv1d_fwd(torch::Tensor x, torch::Tensor w,
                           c10::optional<torch::Tensor> bias, int64_t pad) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 3 &&
              x.scalar_type() == torch::kBFloat16, "x must be bf16 [B,T,D]");
  TORCH_CHECK(x.size(2) % 8 == 0, "D must be divisible by 8");
  const int B = x.size(0), T = x.size(1), D = x.size(2), K = w.size(0);
  TORCH_CHECK(K < MAXK, "kernel size < ", MAXK);
  auto y = torch::empty_like(x);
  auto stream = at::cuda::getCurrentCUDAStream();
  long nvec = (long)B * T * (D / 8);
  int grid = memory_bound_grid(nvec, 256);
  hipLaunchKernelGGL(dwconv_fwd, dim3(grid), dim3(256), 0, stream,
                     (const unsigned short*)x.data_ptr(),
                     (const unsigned short*)w.data_ptr(),
                     bias.has_value() ? (const unsigned short*)
                                            bias->data_ptr()
                                      : nullptr,
                     (unsigned short*)y.data_ptr(), B, T, D, K, (int)pad);
  return y;
}

std::vector<torch::Tensor> dwconv1d_bwd(torch::Tensor dy, torch::Tensor x,
                                        torch::Tensor w, int64_t pad) {
  const int B = x.size(0), T = x.size(1), D = x.size(2), K = w.size(0);
  auto dx = torch::empty_like(x);
  auto opts = x.options().dtype(torch::kFloat32);
  auto dw = torch::zeros({K, D}, opts);
  auto db = torch::zeros({D}, opts);
  auto stream = at::cuda::getCurrentCUDAStream();
  long nvec = (long)B * T * (D / 8);
  hipLaunchKernelGGL(dwconv_bwd_dx, dim3(memory_bound_grid(nvec, 256)),
                     dim3(256), 0, stream,
                     (const unsigned short*)dy.data_ptr(),
                     (const unsigned short*)w.data_ptr(),
                     (unsigned short*)dx.data_ptr(), B, T, D, K, (int)pad);
  int nstripes = (int)std::min<long>(512, std::max<long>(1, (long)B * T / 8));
  dim3 grid_w((D + 255) / 256, nstripes);
  hipLaunchKernelGGL(dwconv_bwd_dw, grid_w, dim3(256), 0, stream,
                     (const unsigned short*)dy.data_ptr(),
                     (const unsigned short*)x.data_ptr(),
                     dw.data_ptr<float>(), db.data_ptr<float>(), B, T, D, K,
                     (int)pad, nstripes);
  return {dx, dw, db};
}
