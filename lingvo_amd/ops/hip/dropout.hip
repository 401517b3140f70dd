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

// ACT: 0 = identity, 1 = SiLU/Swish, 2 = ReLU — applied BEFORE the
// dropout mask, fusing the FFN activation pass into this kernel.
template <int ACT>
__device__ __forceinline__ float apply_act(float v) {
  if (ACT == 1) return v / (1.f + __expf(-v)) ;
  if (ACT == 2) return v > 0.f ? v : 0.f;
  return v;
}

template <int ACT>
__device__ __forceinline__ float act_grad(float v) {
  if (ACT == 1) {
    const float s = 1.f / (1.f + __expf(-v));
    return s * (1.f + v * (1.f - s));
  }
  if (ACT == 2) return v > 0.f ? 1.f : 0.f;
  return 1.f;
}

// scale folds a residual weight (e.g. the conformer's 0.5 half-FFN)
// and pad ([B*T] bf16, 1.0 == padded; row = element/D) folds the
// ApplyPadding mask into the same pass.
template <bool RESIDUAL, int ACT>
__global__ void dropout_fwd_kernel(const unsigned short* __restrict__ x,
                                   const unsigned short* __restrict__ res,
                                   unsigned short* __restrict__ y,
                                   long nvec, unsigned long long seed,
                                   const long* __restrict__ step_seed,
                                   float keep, float inv_keep,
                                   float scale,
                                   const unsigned short* __restrict__ pad,
                                   long dvec) {
  const unsigned int thresh = (unsigned int)(keep * 4294967296.0);
  if (step_seed) seed ^= (unsigned long long)(*step_seed);
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long)gridDim.x * blockDim.x) {
    float sc = scale;
    if (pad) sc *= 1.f - bf16_bits_to_float(pad[i / dvec]);
    ushortx8 v = *reinterpret_cast<const ushortx8*>(x + i * 8);
    ushortx8 o;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      bool kept = hash_u32(seed, i * 8 + e) < thresh;
      float f = kept ? apply_act<ACT>(bf16_bits_to_float(v[e])) * inv_keep
                           * sc
                     : 0.f;
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

template <int ACT>
__global__ void dropout_bwd_kernel(const unsigned short* __restrict__ dy,
                                   const unsigned short* __restrict__ x,
                                   unsigned short* __restrict__ dx,
                                   long nvec, unsigned long long seed,
                                   const long* __restrict__ step_seed,
                                   float keep, float inv_keep,
                                   float scale,
                                   const unsigned short* __restrict__ pad,
                                   long dvec) {
  const unsigned int thresh = (unsigned int)(keep * 4294967296.0);
  if (step_seed) seed ^= (unsigned long long)(*step_seed);
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long)gridDim.x * blockDim.x) {
    float sc = scale;
    if (pad) sc *= 1.f - bf16_bits_to_float(pad[i / dvec]);
    ushortx8 v = *reinterpret_cast<const ushortx8*>(dy + i * 8);
    ushortx8 xv;
    if (ACT != 0) xv = *reinterpret_cast<const ushortx8*>(x + i * 8);
    ushortx8 o;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      bool kept = hash_u32(seed, i * 8 + e) < thresh;
      float f = kept ? bf16_bits_to_float(v[e]) * inv_keep * sc : 0.f;
      if (ACT != 0) f *= act_grad<ACT>(bf16_bits_to_float(xv[e]));
      o[e] = float_to_bf16_bits(f);
    }
    *reinterpret_cast<ushortx8*>(dx + i * 8) = o;
  }
}

}  // namespace

torch::Tensor dropout_fwd(torch::Tensor x, c10::optional<torch::Tensor> res,
                          int64_t seed, c10::optional<torch::Tensor> step_seed,
                          double keep, int64_t act, double scale,
                          c10::optional<torch::Tensor> pad, int64_t d) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() &&
              x.scalar_type() == torch::kBFloat16 && x.numel() % 8 == 0);
  TORCH_CHECK(!(res.has_value() && act != 0),
              "residual+activation fusion unsupported");
  TORCH_CHECK(!pad.has_value() || (d > 0 && d % 8 == 0),
              "padding fusion needs D % 8 == 0");
  auto y = torch::empty_like(x);
  long nvec = x.numel() / 8;
  auto stream = at::cuda::getCurrentCUDAStream();
  const long* ssp = step_seed.has_value() ?
      step_seed->data_ptr<long>() : nullptr;
  const unsigned short* pp = pad.has_value() ?
      (const unsigned short*)pad->data_ptr() : nullptr;
  const long dvec = d > 0 ? d / 8 : 1;
#define LAUNCH_FWD(RES, ACT, RESPTR)                                         \
  hipLaunchKernelGGL((dropout_fwd_kernel<RES, ACT>),                         \
                     dim3(memory_bound_grid(nvec, 256)), dim3(256), 0,       \
                     stream, (const unsigned short*)x.data_ptr(), RESPTR,    \
                     (unsigned short*)y.data_ptr(), nvec,                    \
                     (unsigned long long)seed, ssp, (float)keep,             \
                     (float)(1.0 / keep), (float)scale, pp, dvec)
  if (res.has_value()) {
    LAUNCH_FWD(true, 0, (const unsigned short*)res->data_ptr());
  } else if (act == 1) {
    LAUNCH_FWD(false, 1, nullptr);
  } else if (act == 2) {
    LAUNCH_FWD(false, 2, nullptr);
  } else {
    LAUNCH_FWD(false, 0, nullptr);
  }
#undef LAUNCH_FWD
  return y;
}

torch::Tensor dropout_bwd(torch::Tensor dy, c10::optional<torch::Tensor> x,
                          int64_t seed, c10::optional<torch::Tensor> step_seed,
                          double keep, int64_t act, double scale,
                          c10::optional<torch::Tensor> pad, int64_t d) {
  TORCH_CHECK(act == 0 || x.has_value(),
              "activation-fused dropout bwd needs the pre-activation input");
  auto dx = torch::empty_like(dy);
  long nvec = dy.numel() / 8;
  auto stream = at::cuda::getCurrentCUDAStream();
  const long* ssp = step_seed.has_value() ?
      step_seed->data_ptr<long>() : nullptr;
  const unsigned short* xp = x.has_value() ?
      (const unsigned short*)x->data_ptr() : nullptr;
  const unsigned short* pp = pad.has_value() ?
      (const unsigned short*)pad->data_ptr() : nullptr;
  const long dvec = d > 0 ? d / 8 : 1;
#define LAUNCH_BWD(ACT)                                                      \
  hipLaunchKernelGGL((dropout_bwd_kernel<ACT>),                              \
                     dim3(memory_bound_grid(nvec, 256)), dim3(256), 0,       \
                     stream, (const unsigned short*)dy.data_ptr(), xp,       \
                     (unsigned short*)dx.data_ptr(), nvec,                   \
                     (unsigned long long)seed, ssp, (float)keep,             \
                     (float)(1.0 / keep), (float)scale, pp, dvec)
  if (act == 1) {
    LAUNCH_BWD(1);
  } else if (act == 2) {
    LAUNCH_BWD(2);
  } else {
    LAUNCH_BWD(0);
  }
#undef LAUNCH_BWD
  return dx;
}
