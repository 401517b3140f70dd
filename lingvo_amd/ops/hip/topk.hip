// Row-wise top-k for beam-search pruning (gfx950) — K12 in SURVEY §2.9.
// scores [R, V] bf16 -> (vals fp32 [R, k], idx int32 [R, k]) sorted
// descending. k <= 32. One 256-thread block per row: each thread keeps
// an insertion-sorted local top-k over its strided slice, then one
// wave merges the 256 partial lists from LDS in k max-scan rounds.
//
// Replaces torch.topk on [num_hyps*B, V] inside the decode loop
// (reference beam-search op does a CPU partial sort,
// beam_search_step_op_kernels.cc).

#include <ATen/cuda/CUDAContext.h>
#include <torch/extension.h>

#include "common.h"

namespace {

constexpr int TK_BLOCK = 256;
constexpr int TK_MAXK = 32;

template <typename T>
__device__ __forceinline__ float tk_load(const T* p);
template <>
__device__ __forceinline__ float tk_load<unsigned short>(
    const unsigned short* p) {
  return bf16_bits_to_float(*p);
}
template <>
__device__ __forceinline__ float tk_load<float>(const float* p) {
  return *p;
}

// KT is compile-time so the per-thread top-KT lists live in REGISTERS:
// dynamic indexing would spill them to scratch (4x slower, measured).
template <typename T, int KT>
__global__ __launch_bounds__(TK_BLOCK) void topk_rows_kernel(
    const T* __restrict__ scores, float* __restrict__ out_v,
    int* __restrict__ out_i, long V, int k) {
  __shared__ float cand_v[TK_BLOCK * KT];
  __shared__ int cand_i[TK_BLOCK * KT];
  const long row = blockIdx.x;
  const int tid = threadIdx.x;
  const T* src = scores + row * V;

  // Phase 1: per-thread top-KT via bubble insertion (static indices).
  float lv[KT];
  int li[KT];
#pragma unroll
  for (int j = 0; j < KT; ++j) {
    lv[j] = -INFINITY;
    li[j] = -1;
  }
  for (long i = tid; i < V; i += TK_BLOCK) {
    const float v = tk_load<T>(src + i);
    if (v <= lv[KT - 1]) continue;
    lv[KT - 1] = v;
    li[KT - 1] = (int)i;
#pragma unroll
    for (int j = KT - 1; j > 0; --j) {
      if (lv[j] > lv[j - 1]) {
        const float tv = lv[j];
        lv[j] = lv[j - 1];
        lv[j - 1] = tv;
        const int ti = li[j];
        li[j] = li[j - 1];
        li[j - 1] = ti;
      }
    }
  }
#pragma unroll
  for (int j = 0; j < KT; ++j) {
    cand_v[tid * KT + j] = lv[j];
    cand_i[tid * KT + j] = li[j];
  }
  __syncthreads();

  // Phase 2: wave 0 merges. Each selection round: every lane scans 4
  // candidate slots (256*k/64 <= 128 -> per-lane head pointers), take
  // the global max via wave reduce, advance that list's head.
  if (tid >= WAVE_SIZE) return;
  // Each lane owns 4 source lists (their heads).
  int head[4] = {0, 0, 0, 0};
  for (int sel = 0; sel < k; ++sel) {
    float best = -INFINITY;
    int best_list = -1;
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      const int list = tid * 4 + c;
      if (head[c] < KT) {
        const float v = cand_v[list * KT + head[c]];
        if (v > best) {
          best = v;
          best_list = c;
        }
      }
    }
    // wave argmax: reduce (value, lane) pairs.
    float rbest = best;
    int rlane = tid;
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      const float ov = __shfl_down(rbest, off);
      const int ol = __shfl_down(rlane, off);
      if (ov > rbest) {
        rbest = ov;
        rlane = ol;
      }
    }
    rbest = __shfl(rbest, 0);
    rlane = __shfl(rlane, 0);
    if (tid == rlane) {
#pragma unroll
      for (int c = 0; c < 4; ++c) {
        if (c == best_list) {
          out_v[row * k + sel] = best;
          out_i[row * k + sel] = cand_i[(tid * 4 + c) * KT + head[c]];
          ++head[c];
        }
      }
    }
  }
}

}  // namespace

std::vector<torch::Tensor> topk_rows(torch::Tensor scores, int64_t k) {
  TORCH_CHECK(scores.is_cuda() && scores.is_contiguous() &&
              scores.dim() == 2 &&
              (scores.scalar_type() == torch::kBFloat16 ||
               scores.scalar_type() == torch::kFloat32));
  TORCH_CHECK(k >= 1 && k <= TK_MAXK && k <= scores.size(1));
  const long R = scores.size(0), V = scores.size(1);
  auto vals = torch::empty({R, k}, scores.options().dtype(torch::kFloat32));
  auto idx = torch::empty({R, k}, scores.options().dtype(torch::kInt32));
  auto stream = at::cuda::getCurrentCUDAStream();
#define TK_LAUNCH(T, KT, PTR)                                             \
  hipLaunchKernelGGL((topk_rows_kernel<T, KT>), dim3((unsigned)R),         \
                     dim3(TK_BLOCK), 0, stream, PTR,                       \
                     vals.data_ptr<float>(), idx.data_ptr<int>(), V,       \
                     (int)k)
#define TK_DISPATCH(T, PTR)                                                \
  do {                                                                     \
    if (k <= 8) TK_LAUNCH(T, 8, PTR);                                      \
    else if (k <= 16) TK_LAUNCH(T, 16, PTR);                               \
    else if (k <= 24) TK_LAUNCH(T, 24, PTR);                               \
    else TK_LAUNCH(T, 32, PTR);                                            \
  } while (0)
  if (scores.scalar_type() == torch::kBFloat16) {
    TK_DISPATCH(unsigned short, (const unsigned short*)scores.data_ptr());
  } else {
    TK_DISPATCH(float, scores.data_ptr<float>());
  }
#undef TK_DISPATCH
#undef TK_LAUNCH
  return {vals, idx};
}
