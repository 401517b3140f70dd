// MoE top-2 gating position kernel (reference
// Top2GatingOnLogits capacity cumsum, gshard_layers.py:1932).
//
// Replaces the one-hot cumsum ([N, E] int32 materialization + scan)
// with a per-expert ordered block scan: pos1[i] = number of earlier
// tokens whose top1 is the same expert; pos2[i] continues after all
// top1 assignments (count1[e] offset) — bit-identical to the reference
// ordering, deterministic by construction. One block per expert;
// block-wide exclusive prefix sums over 256-token chunks.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

// 256-thread exclusive prefix sum of 0/1 flags; returns (excl, total).
__device__ void block_scan(int flag, int* excl, int* total,
                           float* scratch) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x / WAVE_SIZE;
  // Wave-level inclusive scan via shifts.
  int v = flag;
  for (int off = 1; off < WAVE_SIZE; off <<= 1) {
    int up = __shfl_up(v, off, WAVE_SIZE);
    if (lane >= off) v += up;
  }
  const int wave_total = __shfl(v, WAVE_SIZE - 1, WAVE_SIZE);
  if (lane == WAVE_SIZE - 1) scratch[wid] = (float)v;
  __syncthreads();
  int base = 0;
  for (int w = 0; w < wid; ++w) base += (int)scratch[w];
  int all = 0;
  for (int w = 0; w < 4; ++w) all += (int)scratch[w];
  __syncthreads();
  *excl = base + v - flag;
  *total = all;
}

__global__ __launch_bounds__(256) void moe_positions_kernel(
    const int* __restrict__ top1, const int* __restrict__ top2,
    int* __restrict__ pos1, int* __restrict__ pos2,
    int* __restrict__ count1, long n) {
  const int e = blockIdx.x;
  __shared__ float scratch[4];
  int running = 0;
  // Pass 1: top1 positions.
  for (long base = 0; base < n; base += 256) {
    const long i = base + threadIdx.x;
    const int flag = (i < n && top1[i] == e) ? 1 : 0;
    int excl, total;
    block_scan(flag, &excl, &total, scratch);
    if (flag) pos1[i] = running + excl;
    running += total;
    __syncthreads();
  }
  if (threadIdx.x == 0) count1[e] = running;
  // Pass 2: top2 positions continue after the top1 block.
  for (long base = 0; base < n; base += 256) {
    const long i = base + threadIdx.x;
    const int flag = (i < n && top2[i] == e) ? 1 : 0;
    int excl, total;
    block_scan(flag, &excl, &total, scratch);
    if (flag) pos2[i] = running + excl;
    running += total;
    __syncthreads();
  }
}

}  // namespace

std::vector<torch::Tensor> moe_positions(torch::Tensor top1,
                                         torch::Tensor top2,
                                         int64_t num_experts) {
  TORCH_CHECK(top1.is_cuda() && top1.scalar_type() == torch::kInt32);
  TORCH_CHECK(top2.scalar_type() == torch::kInt32);
  const long n = top1.numel();
  auto pos1 = torch::zeros_like(top1);
  auto pos2 = torch::zeros_like(top1);
  auto count1 = torch::zeros({num_experts},
                             top1.options().dtype(torch::kInt32));
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(moe_positions_kernel, dim3((unsigned)num_experts),
                     dim3(256), 0, stream, top1.data_ptr<int>(),
                     top2.data_ptr<int>(), pos1.data_ptr<int>(),
                     pos2.data_ptr<int>(), count1.data_ptr<int>(), n);
  return {pos1, pos2, count1};
}
