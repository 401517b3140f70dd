// Layout-verification probe: C[MxN] = A[MxK] @ B[KxN] for one MFMA tile
// (M=N=16, K=32) using the fragment layouts declared in mfma.h. The GPU
// test compares against torch.matmul with random asymmetric inputs
// (transpose-detecting per guide methodology rule 16).

#include <ATen/cuda/CUDAContext.h>
#include <torch/extension.h>

#include "mfma.h"

namespace {

__global__ void mfma_probe_kernel(const unsigned short* __restrict__ a,
                                  const unsigned short* __restrict__ b,
                                  float* __restrict__ c) {
  const int lane = threadIdx.x & 63;
  // A[row][k], row-major [16][32]: lane -> row=lane&15, k0=(lane>>4)*8
  float af[8], bf[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    af[j] = bf16_bits_to_float(a[(lane & 15) * 32 + (lane >> 4) * 8 + j]);
    // B[k][col], we read from a K-contiguous [N][K] buffer: bT[col][k]
    bf[j] = bf16_bits_to_float(b[(lane & 15) * 32 + (lane >> 4) * 8 + j]);
  }
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = mfma16x16x32_bf16(pack_bf16x8(af), pack_bf16x8(bf), acc);
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    c[((lane >> 4) * 4 + r) * 16 + (lane & 15)] = acc[r];
  }
}

}  // namespace

torch::Tensor mfma_probe(torch::Tensor a, torch::Tensor bT) {
  // a: [16, 32] bf16 row-major; bT: [16, 32] bf16 = B^T (K-contiguous).
  TORCH_CHECK(a.is_cuda() && a.is_contiguous() && bT.is_contiguous());
  TORCH_CHECK(a.size(0) == 16 && a.size(1) == 32);
  auto c = torch::empty({16, 16}, a.options().dtype(torch::kFloat32));
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, stream,
                     (const unsigned short*)a.data_ptr(),
                     (const unsigned short*)bT.data_ptr(),
                     c.data_ptr<float>());
  return c;
}

// ---------------------------------------------------------------------------
// ds_read_tr16_b64 exploration probe. Measured gfx950 semantics
// (tools/tr16_diag.hip + tr16_diag2.hip raw-pattern dumps):
//   out(lane l, elem j) = LDS[ addr supplied by lane (l&~15)|((l&3)+4j) ]
// within each 16-lane group — a cooperative cross-lane gather where
// every source element is REPLICATED into 4 output lanes (lanes 0,4,8,
// 12 of a group produce identical results). A single read therefore
// covers only 4 distinct output columns, NOT the 16-distinct-column
// MFMA B fragment this probe originally assumed; building B-frags needs
// a different (multi-read / different-layout) composition, as in
// HipKittens' 192-read kernels. Since phase-skip attribution measured
// the transposed-staging cost this would remove at only ~4% of
// flash-attention backward, the kernels keep their scalar-write
// transposed staging; this probe + the diag tools document the measured
// instruction behavior for future schedule work.
// ---------------------------------------------------------------------------

typedef __attribute__((__vector_size__(4 * sizeof(__bf16)))) __bf16 bf16x4_v;
#define LV_LDS __attribute__((address_space(3)))

// panel: tiled [K/4][4 col-blocks][4][4] layout (16-col logical panel).
__device__ __forceinline__ bf16x8 lds_b_frag_tr16(const char* panel,
                                                  int k0) {
  const int lane = threadIdx.x & 63;
  const int g = lane >> 4;
  const int cl = lane & 15;
  const int kb = (k0 + g * 8) / 4;       // first 4-row block
  const int cb = cl / 4;
  const char* addr =
      panel + (((kb * 4 + cb) * 16) + (cl & 3)) * 2;
  bf16x4_v lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (LV_LDS bf16x4_v*)(addr));
  bf16x4_v hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (LV_LDS bf16x4_v*)(addr + 4 * 16 * 2));
  bf16x8 out;
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    out[j] = lo[j];
    out[j + 4] = hi[j];
  }
  return out;
}

namespace {

__global__ void tr16_probe_kernel(const unsigned short* __restrict__ a,
                                  const unsigned short* __restrict__ b,
                                  float* __restrict__ c) {
  // Stage b [32][16] row-major into the TILED [8][4][4][4] layout.
  __shared__ unsigned short panel[32 * 16];
  for (int i = threadIdx.x; i < 32 * 16; i += 64) {
    const int row = i / 16;
    const int col = i % 16;
    panel[((row / 4) * 4 + col / 4) * 16 + (row % 4) * 4 + (col % 4)] =
        b[i];
  }
  __syncthreads();
  const int lane = threadIdx.x & 63;
  float af[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    af[j] = bf16_bits_to_float(a[(lane & 15) * 32 + (lane >> 4) * 8 + j]);
  }
  bf16x8 bfrag = lds_b_frag_tr16(reinterpret_cast<const char*>(panel), 0);
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = mfma16x16x32_bf16(pack_bf16x8(af), bfrag, acc);
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    c[((lane >> 4) * 4 + r) * 16 + (lane & 15)] = acc[r];
  }
}

}  // namespace

torch::Tensor tr16_probe(torch::Tensor a, torch::Tensor b) {
  // a: [16, 32] bf16 row-major (A); b: [32, 16] bf16 row-major (B).
  // Returns A @ B [16, 16] with B read through ds_read_tr16_b64.
  TORCH_CHECK(a.is_cuda() && a.is_contiguous() && b.is_contiguous());
  TORCH_CHECK(a.size(0) == 16 && a.size(1) == 32);
  TORCH_CHECK(b.size(0) == 32 && b.size(1) == 16);
  auto c = torch::empty({16, 16}, a.options().dtype(torch::kFloat32));
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(tr16_probe_kernel, dim3(1), dim3(64), 0, stream,
                     (const unsigned short*)a.data_ptr(),
                     (const unsigned short*)b.data_ptr(),
                     c.data_ptr<float>());
  return c;
}
