// Embedding gather fwd + deterministic scatter-add bwd (SURVEY K10;
// reference lingvo/core/layers.py:2679 SimpleEmbeddingLayer).
//
// fwd: out[r, :] = scale * table[ids[r], :] — vectorized gather.
// bwd: host sorts the ids (torch.sort, stable) and computes segment
// boundaries; the kernel reduces each segment's rows IN INDEX ORDER
// into fp32 before one bf16 store per (vocab row, chunk) — bitwise
// deterministic across runs, unlike atomic scatter-add.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

__global__ void emb_gather_kernel(const unsigned short* __restrict__ table,
                                  const long* __restrict__ ids,
                                  unsigned short* __restrict__ out, long rows,
                                  int d, float scale) {
  const long total = rows * (d / 8);
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const long r = i / (d / 8);
    const int c8 = (int)(i % (d / 8)) * 8;
    const long v = ids[r];
    ushortx8 x = *reinterpret_cast<const ushortx8*>(table + v * d + c8);
    if (scale != 1.f) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        x[j] = float_to_bf16_bits(bf16_bits_to_float(x[j]) * scale);
      }
    }
    *reinterpret_cast<ushortx8*>(out + r * d + c8) = x;
  }
}

// One block per (segment, d-chunk of 256): accumulates the segment's
// rows in sorted order (deterministic) into fp32, adds into dtable.
__global__ void emb_scatter_kernel(const unsigned short* __restrict__ dy,
                                   const long* __restrict__ perm,
                                   const long* __restrict__ seg_start,
                                   const long* __restrict__ seg_id,
                                   unsigned short* __restrict__ dtable,
                                   long num_segs, int d, float scale) {
  const int dchunks = (d + 255) / 256;
  const long seg = blockIdx.x / dchunks;
  if (seg >= num_segs) return;
  const int c = (int)(blockIdx.x % dchunks) * 256 + threadIdx.x;
  if (c >= d) return;
  const long lo = seg_start[seg];
  const long hi = seg_start[seg + 1];
  float acc = 0.f;
  for (long i = lo; i < hi; ++i) {
    acc += bf16_bits_to_float(dy[perm[i] * d + c]);
  }
  const long v = seg_id[seg];
  long off = v * d + c;
  dtable[off] = float_to_bf16_bits(
      bf16_bits_to_float(dtable[off]) + acc * scale);
}

}  // namespace

torch::Tensor emb_gather(torch::Tensor table, torch::Tensor ids,
                         double scale) {
  TORCH_CHECK(table.is_cuda() && table.is_contiguous() &&
              table.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(ids.scalar_type() == torch::kLong);
  const int d = table.size(1);
  TORCH_CHECK(d % 8 == 0, "D % 8");
  auto ids_flat = ids.contiguous().reshape(-1);
  const long rows = ids_flat.numel();
  auto out = torch::empty({rows, (long)d}, table.options());
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(emb_gather_kernel,
                     dim3(memory_bound_grid(rows * d / 8, 256)), dim3(256),
                     0, stream, (const unsigned short*)table.data_ptr(),
                     ids_flat.data_ptr<long>(),
                     (unsigned short*)out.data_ptr(), rows, d,
                     (float)scale);
  auto shape = ids.sizes().vec();
  shape.push_back(d);
  return out.reshape(shape);
}

torch::Tensor emb_scatter_add(torch::Tensor dy, torch::Tensor ids,
                              int64_t vocab, double scale) {
  TORCH_CHECK(dy.is_cuda() && dy.scalar_type() == torch::kBFloat16);
  auto ids_flat = ids.contiguous().reshape(-1);
  const long rows = ids_flat.numel();
  const int d = dy.size(-1);
  auto dy2 = dy.contiguous().reshape({rows, (long)d});
  // Stable sort -> segments of equal ids (host-side torch ops,
  // deterministic).
  auto sorted = ids_flat.sort(/*dim=*/0, /*descending=*/false);
  auto sorted_ids = std::get<0>(sorted);
  auto perm = std::get<1>(sorted);
  auto uniq = at::unique_consecutive(sorted_ids, false, true);
  auto seg_ids = std::get<0>(uniq);
  auto counts = std::get<2>(uniq);
  auto seg_start = torch::zeros({seg_ids.numel() + 1},
                                counts.options());
  seg_start.slice(0, 1, seg_ids.numel() + 1).copy_(counts.cumsum(0));
  auto dtable = torch::zeros({vocab, (long)d}, dy.options());
  const long num_segs = seg_ids.numel();
  const int dchunks = (d + 255) / 256;
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(emb_scatter_kernel,
                     dim3((unsigned)(num_segs * dchunks)), dim3(256), 0,
                     stream, (const unsigned short*)dy2.data_ptr(),
                     perm.data_ptr<long>(), seg_start.data_ptr<long>(),
                     seg_ids.data_ptr<long>(),
                     (unsigned short*)dtable.data_ptr(), num_segs, d,
                     (float)scale);
  return dtable;
}
