// Common helpers for lingvo_amd gfx950 (CDNA4) HIP kernels.
//
// Conventions (see /opt/skills guides):
//  - wavefront = 64 lanes; block sizes are multiples of 64.
//  - bf16 global loads are vectorized as ushort8 (16 B/lane).
//  - fp32 accumulation everywhere; bf16 stores use RNE via __float2bfloat16.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define WAVE_SIZE 64

#define LV_CHECK_HIP(expr)                                       \
  do {                                                           \
    hipError_t _e = (expr);                                      \
    if (_e != hipSuccess) {                                      \
      TORCH_CHECK(false, "HIP error: ", hipGetErrorString(_e));  \
    }                                                            \
  } while (0)

typedef __attribute__((ext_vector_type(4))) float floatx4;
typedef __attribute__((ext_vector_type(8))) float floatx8;
typedef __attribute__((ext_vector_type(8))) unsigned short ushortx8;
typedef __attribute__((ext_vector_type(4))) unsigned short ushortx4;
typedef __attribute__((ext_vector_type(2))) unsigned short ushortx2;

__device__ __forceinline__ float bf16_bits_to_float(unsigned short u) {
  union {
    unsigned int i;
    float f;
  } v;
  v.i = ((unsigned int)u) << 16;
  return v.f;
}

__device__ __forceinline__ unsigned short float_to_bf16_bits(float f) {
  __hip_bfloat16 h = __float2bfloat16(f);  // RNE
  union {
    __hip_bfloat16 h;
    unsigned short u;
  } v;
  v.h = h;
  return v.u;
}

// ---- wave-level reductions (64-wide) -------------------------------------
__device__ __forceinline__ float wave_reduce_sum(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    x += __shfl_xor(x, off, WAVE_SIZE);
  }
  return x;
}

__device__ __forceinline__ float wave_reduce_max(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    x = fmaxf(x, __shfl_xor(x, off, WAVE_SIZE));
  }
  return x;
}

// Block-level reduce over NWAVES waves using LDS scratch (caller provides
// __shared__ float scratch[NWAVES]); result valid on all threads.
template <int NWAVES>
__device__ __forceinline__ float block_reduce_sum(float x, float* scratch) {
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wid = threadIdx.x / WAVE_SIZE;
  x = wave_reduce_sum(x);
  if (lane == 0) scratch[wid] = x;
  __syncthreads();
  float total = 0.f;
#pragma unroll
  for (int i = 0; i < NWAVES; ++i) total += scratch[i];
  __syncthreads();
  return total;
}

__device__ __forceinline__ int cdiv_dev(int a, int b) { return (a + b - 1) / b; }

static inline int cdiv(long a, long b) { return (int)((a + b - 1) / b); }

// Grid sizing for memory-bound grid-stride kernels: cap at ~8 blocks/CU
// (guide §6 Guideline 11).
static inline int memory_bound_grid(long total_work, int block_size,
                                    int cap_blocks = 2048) {
  long want = (total_work + block_size - 1) / block_size;
  return (int)(want < cap_blocks ? (want > 0 ? want : 1) : cap_blocks);
}
