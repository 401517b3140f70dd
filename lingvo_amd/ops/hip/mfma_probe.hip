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
