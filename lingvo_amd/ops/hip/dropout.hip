// Fused deterministic dropout (+ optional residual add) for gfx950.
// Replaces the reference's DeterministicDropout (py_utils.py:3978): the
// mask is a pure function of (seed, element index), so backward
// recomputes it from the saved seed — no mask tensor is stored and
// fwd/bwd are one kernel each instead of rand+cmp+mul chains.

#include <ATen/cuda/CUDAContext.h>
#include <torch/extension.h>

#include "common.h"

namespace {

// Stateless hash RNG (xxhash-style avalanche): adequate for dropout.
__device__ __forceinline__ unsigned int hash_u32(unsigned long long seed,
                                                 unsigned long long idx) {
  unsigned long long h = seed ^ (idx * 0x9E3779B97F4A7C15ull);
  h ^= h >> 33;
  h *= 0xFF51AFD7ED558CCDull;
  h ^= h >> 33;
  h *= 0xC4CEB9FE1A85EC53ull;
  h ^= h >> 33;
  return (unsigned int)h;
}

template <bool RESIDUAL>
__global__ void dropout_fwd_kernel(const unsigned short* __restrict__ x,
                                   const unsigned short* __restrict__ res,
                                   unsigned short* __restrict__ y,
                                   long nvec, unsigned long long seed,
                                   const long* __restrict__ step_seed,
                                   float keep, float inv_keep) {
  const unsigned int thresh = (unsigned int)(keep * 4294967296.0);
  if (step_seed) seed ^= (unsigned long long)(*step_seed);
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long)gridDim.x * blockDim.x) {
    ushortx8 v = *reinterpret_cast<const ushortx8*>(x + i * 8);
    ushortx8 o;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      bool kept = hash_u32(seed, i * 8 + e) < thresh;
      float f = kept ? bf16_bits_to_float(v[e]) * inv_keep : 0.f;
      o[e] = float_to_bf16_bits(f);
    }
    if (RESIDUAL) {
      ushortx8 r = *reinterpret_cast<const ushortx8*>(res + i * 8);
#pragma unroll
      for (int e = 0; e < 8; ++e)
        o[e] = float_to_bf16_bits(bf16_bits_to_float(o[e]) +
                                  bf16_bits_to_float(r[e]));
    }
    *reinterpret_cast<ushortx8*>(y + i * 8) = o;
  }
}

__global__ void dropout_bwd_kernel(const unsigned short* __restrict__ dy,
                                   unsigned short* __restrict__ dx,
                                   long nvec, unsigned long long seed,
                                   const long* __restrict__ step_seed,
                                   float keep, float inv_keep) {
  const unsigned int thresh = (unsigned int)(keep * 4294967296.0);
  if (step_seed) seed ^= (unsigned long long)(*step_seed);
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long)gridDim.x * blockDim.x) {
    ushortx8 v = *reinterpret_cast<const ushortx8*>(dy + i * 8);
    ushortx8 o;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      bool kept = hash_u32(seed, i * 8 + e) < thresh;
      float f = kept ? bf16_bits_to_float(v[e]) * inv_keep : 0.f;
      o[e] = float_to_bf16_bits(f);
    }
    *reinterpret_cast<ushortx8*>(dx + i * 8) = o;
  }
}

}  // namespace

torch::Tensor dropout_fwd(torch::Tensor x, c10::optional<torch::Tensor> res,
                          int64_t seed, c10::optional<torch::Tensor> step_seed,
                          double keep) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() &&
              x.scalar_type() == torch::kBFloat16 && x.numel() % 8 == 0);
  auto y = torch::empty_like(x);
  long nvec = x.numel() / 8;
  auto stream = at::cuda::getCurrentCUDAStream();
  const long* ssp = step_seed.has_value() ?
      step_seed->data_ptr<long>() : nullptr;
  if (res.has_value()) {
    hipLaunchKernelGGL((dropout_fwd_kernel<true>),
                       dim3(memory_bound_grid(nvec, 256)), dim3(256), 0,
                       stream, (const unsigned short*)x.data_ptr(),
                       (const unsigned short*)res->data_ptr(),
                       (unsigned short*)y.data_ptr(), nvec,
                       (unsigned long long)seed, ssp, (float)keep,
                       (float)(1.0 / keep));
  } else {
    hipLaunchKernelGGL((dropout_fwd_kernel<false>),
                       dim3(memory_bound_grid(nvec, 256)), dim3(256), 0,
                       stream, (const unsigned short*)x.data_ptr(), nullptr,
                       (unsigned short*)y.data_ptr(), nvec,
                       (unsigned long long)seed, ssp, (float)keep,
                       (float)(1.0 / keep));
  }
  return y;
}

torch::Tensor dropout_bwd(torch::Tensor dy, int64_t seed,
                          c10::optional<torch::Tensor> step_seed,
                          double keep) {
  auto dx = torch::empty_like(dy);
  long nvec = dy.numel() / 8;
  auto stream = at::cuda::getCurrentCUDAStream();
  const long* ssp = step_seed.has_value() ?
      step_seed->data_ptr<long>() : nullptr;
  hipLaunchKernelGGL(dropout_bwd_kernel,
                     dim3(memory_bound_grid(nvec, 256)), dim3(256), 0,
                     stream, (const unsigned short*)dy.data_ptr(),
                     (unsigned short*)dx.data_ptr(), nvec,
                     (unsigned long long)seed, ssp, (float)keep,
                     (float)(1.0 / keep));
  return dx;
}
