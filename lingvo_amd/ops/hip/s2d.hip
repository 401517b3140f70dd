// Space-to-depth transforms for the stride-2 conv frontend (gfx950).
//
// The conv-subsampling 3x3/s2 conv runs as 4 flat GEMMs over a
// space-to-depth'd, zero-padded buffer X [B*(Ho+1)*(Wo+1), 4C] (see
// layers/conformer.py _Conv3x3S2Nhwc). Composing that buffer from
// torch strided-slice copies costs 4 badly-coalesced passes per
// direction (~0.86 ms each at the bench shape); these kernels do each
// transform in ONE output-coalesced pass with ushortx8 vectors.
//
// Block order: s0=(0,1), s1=(1,1), s2=(1,0), s3=(0,0) in (a,b) cell
// coordinates — a = row parity, b = col parity.

#include <ATen/cuda/CUDAContext.h>
#include <torch/extension.h>

#include "common.h"

namespace {

// (a, b) for block index s.
__constant__ const int kBlkA[4] = {0, 1, 1, 0};
__constant__ const int kBlkB[4] = {1, 1, 0, 0};

// x [B, 2Ho, 2Wo, C] -> X [B, Ho+1, Wo+1, 4C]; row 0 / col 0 zero.
__global__ void s2d_fwd_kernel(const unsigned short* __restrict__ x,
                               unsigned short* __restrict__ out, long B,
                               long Ho, long Wo, long C) {
  const long Hp = Ho + 1, Wp = Wo + 1;
  const long cvec = C / 8;
  const long nvec = B * Hp * Wp * 4 * cvec;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long)gridDim.x * blockDim.x) {
    long t = i;
    const long c8 = t % cvec;
    t /= cvec;
    const int s = (int)(t % 4);
    t /= 4;
    const long wp = t % Wp;
    t /= Wp;
    const long hp = t % Hp;
    const long b = t / Hp;
    ushortx8 v;
    if (hp == 0 || wp == 0) {
      for (int e = 0; e < 8; ++e) v[e] = 0;
    } else {
      const long h = 2 * (hp - 1) + kBlkA[s];
      const long w = 2 * (wp - 1) + kBlkB[s];
      v = *reinterpret_cast<const ushortx8*>(
          x + ((b * (2 * Ho) + h) * (2 * Wo) + w) * C + c8 * 8);
    }
    *reinterpret_cast<ushortx8*>(out + i * 8) = v;
  }
}

// dX [B, Ho+1, Wo+1, 4C] -> dx [B, 2Ho, 2Wo, C] (inverse s2d; padded
// row/col dropped).
__global__ void s2d_inv_kernel(const unsigned short* __restrict__ dX,
                               unsigned short* __restrict__ dx, long B,
                               long Ho, long Wo, long C) {
  const long Hp = Ho + 1, Wp = Wo + 1;
  const long cvec = C / 8;
  const long nvec = B * (2 * Ho) * (2 * Wo) * cvec;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long)gridDim.x * blockDim.x) {
    long t = i;
    const long c8 = t % cvec;
    t /= cvec;
    const long w = t % (2 * Wo);
    t /= (2 * Wo);
    const long h = t % (2 * Ho);
    const long b = t / (2 * Ho);
    const int a = (int)(h & 1), bb = (int)(w & 1);
    // block for (a, b): s0=(0,1) s1=(1,1) s2=(1,0) s3=(0,0)
    const int s = a ? (bb ? 1 : 2) : (bb ? 0 : 3);
    const long src = (((b * Hp + h / 2 + 1) * Wp + w / 2 + 1) * 4 + s) *
                         C + c8 * 8;
    *reinterpret_cast<ushortx8*>(dx + i * 8) =
        *reinterpret_cast<const ushortx8*>(dX + src);
  }
}

// dout [B, Ho, Wo, Co] -> dO [B, Ho+1, Wo+1, Co] with zero border.
__global__ void pad_scatter_kernel(const unsigned short* __restrict__ src,
                                   unsigned short* __restrict__ out,
                                   long B, long Ho, long Wo, long Co) {
  const long Hp = Ho + 1, Wp = Wo + 1;
  const long cvec = Co / 8;
  const long nvec = B * Hp * Wp * cvec;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long)gridDim.x * blockDim.x) {
    long t = i;
    const long c8 = t % cvec;
    t /= cvec;
    const long wp = t % Wp;
    t /= Wp;
    const long hp = t % Hp;
    const long b = t / Hp;
    ushortx8 v;
    if (hp == 0 || wp == 0) {
      for (int e = 0; e < 8; ++e) v[e] = 0;
    } else {
      v = *reinterpret_cast<const ushortx8*>(
          src + ((b * Ho + hp - 1) * Wo + wp - 1) * Co + c8 * 8);
    }
    *reinterpret_cast<ushortx8*>(out + i * 8) = v;
  }
}

}  // namespace

torch::Tensor s2d_fwd(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 4 &&
              x.scalar_type() == torch::kBFloat16 && x.size(3) % 8 == 0 &&
              x.size(1) % 2 == 0 && x.size(2) % 2 == 0);
  const long B = x.size(0), Ho = x.size(1) / 2, Wo = x.size(2) / 2,
             C = x.size(3);
  auto out = torch::empty({B, Ho + 1, Wo + 1, 4 * C}, x.options());
  long nvec = out.numel() / 8;
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(s2d_fwd_kernel, dim3(memory_bound_grid(nvec, 256)),
                     dim3(256), 0, stream,
                     (const unsigned short*)x.data_ptr(),
                     (unsigned short*)out.data_ptr(), B, Ho, Wo, C);
  return out;
}

torch::Tensor s2d_inv(torch::Tensor dX, int64_t C) {
  TORCH_CHECK(dX.is_cuda() && dX.is_contiguous() && dX.dim() == 4 &&
              dX.scalar_type() == torch::kBFloat16 && C % 8 == 0 &&
              dX.size(3) == 4 * C);
  const long B = dX.size(0), Ho = dX.size(1) - 1, Wo = dX.size(2) - 1;
  auto dx = torch::empty({B, 2 * Ho, 2 * Wo, C}, dX.options());
  long nvec = dx.numel() / 8;
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(s2d_inv_kernel, dim3(memory_bound_grid(nvec, 256)),
                     dim3(256), 0, stream,
                     (const unsigned short*)dX.data_ptr(),
                     (unsigned short*)dx.data_ptr(), B, Ho, Wo, C);
  return dx;
}

torch::Tensor pad_scatter(torch::Tensor dout) {
  TORCH_CHECK(dout.is_cuda() && dout.is_contiguous() && dout.dim() == 4 &&
              dout.scalar_type() == torch::kBFloat16 &&
              dout.size(3) % 8 == 0);
  const long B = dout.size(0), Ho = dout.size(1), Wo = dout.size(2),
             Co = dout.size(3);
  auto out = torch::empty({B, Ho + 1, Wo + 1, Co}, dout.options());
  long nvec = out.numel() / 8;
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(pad_scatter_kernel,
                     dim3(memory_bound_grid(nvec, 256)), dim3(256), 0,
                     stream, (const unsigned short*)dout.data_ptr(),
                     (unsigned short*)out.data_ptr(), B, Ho, Wo, Co);
  return out;
}
