// MFMA helpers for gfx950 v_mfma_f32_16x16x32_bf16.
//
// Fragment layouts (guide cdna_hip_programming.md §3, HW-verified C/D
// mapping; A/B derived from the same 16-lane grouping and verified by the
// mfma_probe gpu test in tests/test_mfma_probe.py):
//   A (M=16, K=32): lane holds A[row = lane&15][k = (lane>>4)*8 + j], j=0..7
//   B (K=32, N=16): lane holds B[k = (lane>>4)*8 + j][col = lane&15]
//   C/D (16x16):    lane holds C[row = (lane>>4)*4 + r][col = lane&15], r=0..3
//
// I.e. A fragments load 8 contiguous K-elements per lane from a row-major
// [M][K] tile; B fragments load 8 contiguous K-elements per lane from a
// K-contiguous [N][K] tile ("column-major of the math matrix").
#pragma once

#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

__device__ __forceinline__ f32x4 mfma16x16x32_bf16(bf16x8 a, bf16x8 b,
                                                   f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

// Pack 8 floats to a bf16x8 fragment (RNE).
__device__ __forceinline__ bf16x8 pack_bf16x8(const float* f) {
  bf16x8 out;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    unsigned short u = float_to_bf16_bits(f[j]);
    union {
      unsigned short u;
      __bf16 h;
    } cv;
    cv.u = u;
    out[j] = cv.h;
  }
  return out;
}

__device__ __forceinline__ bf16x8 load_bf16x8_bits(const unsigned short* p) {
  union {
    ushortx8 u;
    bf16x8 h;
  } cv;
  cv.u = *reinterpret_cast<const ushortx8*>(p);
  return cv.h;
}

// XOR swizzle on the byte offset of a 16B-granular LDS row layout: spreads
// the 16B slots of 8 consecutive rows across banks (guide §6 Guideline 4).
__device__ __forceinline__ int swz(int row, int byte_in_row) {
  return byte_in_row ^ ((row & 7) << 4);
}
