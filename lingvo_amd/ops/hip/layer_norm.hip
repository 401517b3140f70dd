// Fused LayerNorm / RMSNorm forward+backward for gfx950 (SURVEY.md K5;
// reference op: lingvo/core/layers.py:4927 LayerNorm).
//
// Layout: x is [rows, D] contiguous (rows = prod(leading dims)). One wave
// per row; each lane covers D/64 columns so column-partials for
// dscale/dbias stay in registers across the wave's grid-stride rows and
// are reduced by a second small kernel (no atomics).
//
// Convention matches the Python layer: y = xhat * (1 + scale) + bias.

#include <ATen/cuda/CUDAContext.h>
#include <torch/extension.h>

#include "common.h"

namespace {

constexpr int kWavesPerBlock = 4;
constexpr int kBlock = WAVE_SIZE * kWavesPerBlock;

// ---- forward -------------------------------------------------------------
// NVEC = columns per lane (D / 64). bf16 in/out, fp32 stats saved.
template <int NVEC, bool RMS>
__global__ void ln_fwd_bf16(const unsigned short* __restrict__ x,
                            const unsigned short* __restrict__ scale,
                            const unsigned short* __restrict__ bias,
                            unsigned short* __restrict__ y,
                            float* __restrict__ mean_out,
                            float* __restrict__ rstd_out, int rows, int D,
                            float eps) {
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wave = (blockIdx.x * kWavesPerBlock) + (threadIdx.x / WAVE_SIZE);
  const int nwaves = gridDim.x * kWavesPerBlock;

  // Per-lane affine params (same columns every row).
  float w[NVEC], b[NVEC];
#pragma unroll
  for (int v = 0; v < NVEC; v += 8) {
    ushortx8 ws = *reinterpret_cast<const ushortx8*>(scale + lane * NVEC + v);
#pragma unroll
    for (int j = 0; j < 8; ++j) w[v + j] = 1.f + bf16_bits_to_float(ws[j]);
    if (!RMS) {
      ushortx8 bs = *reinterpret_cast<const ushortx8*>(bias + lane * NVEC + v);
#pragma unroll
      for (int j = 0; j < 8; ++j) b[v + j] = bf16_bits_to_float(bs[j]);
    }
  }

  for (int row = wave; row < rows; row += nwaves) {
    const unsigned short* xr = x + (long)row * D + lane * NVEC;
    float xs[NVEC];
    float sum = 0.f, sumsq = 0.f;
#pragma unroll
    for (int v = 0; v < NVEC; v += 8) {
      ushortx8 xv = *reinterpret_cast<const ushortx8*>(xr + v);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = bf16_bits_to_float(xv[j]);
        xs[v + j] = f;
        sum += f;
        sumsq += f * f;
      }
    }
    sum = wave_reduce_sum(sum);
    sumsq = wave_reduce_sum(sumsq);
    const float inv_d = 1.f / (float)D;
    float mu = RMS ? 0.f : sum * inv_d;
    float var = sumsq * inv_d - mu * mu;
    float rstd = rsqrtf(var + eps);
    if (lane == 0) {
      if (!RMS) mean_out[row] = mu;
      rstd_out[row] = rstd;
    }
    unsigned short* yr = y + (long)row * D + lane * NVEC;
#pragma unroll
    for (int v = 0; v < NVEC; v += 8) {
      ushortx8 ov;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float xhat = (xs[v + j] - mu) * rstd;
        float o = xhat * w[v + j] + (RMS ? 0.f : b[v + j]);
        ov[j] = float_to_bf16_bits(o);
      }
      *reinterpret_cast<ushortx8*>(yr + v) = ov;
    }
  }
}

// ---- backward ------------------------------------------------------------
// dx in one pass; per-wave column partials for dscale/dbias written to
// partials [2, nwaves, D] (dscale at slab 0, dbias at slab 1).
template <int NVEC, bool RMS>
__global__ void ln_bwd_bf16(const unsigned short* __restrict__ dy,
                            const unsigned short* __restrict__ x,
                            const unsigned short* __restrict__ scale,
                            const float* __restrict__ mean,
                            const float* __restrict__ rstd,
                            unsigned short* __restrict__ dx,
                            float* __restrict__ partials, int rows, int D) {
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wave = (blockIdx.x * kWavesPerBlock) + (threadIdx.x / WAVE_SIZE);
  const int nwaves = gridDim.x * kWavesPerBlock;

  float w[NVEC];
#pragma unroll
  for (int v = 0; v < NVEC; v += 8) {
    ushortx8 ws = *reinterpret_cast<const ushortx8*>(scale + lane * NVEC + v);
#pragma unroll
    for (int j = 0; j < 8; ++j) w[v + j] = 1.f + bf16_bits_to_float(ws[j]);
  }
  float dscale_acc[NVEC];
  float dbias_acc[NVEC];
#pragma unroll
  for (int v = 0; v < NVEC; ++v) {
    dscale_acc[v] = 0.f;
    dbias_acc[v] = 0.f;
  }

  const float inv_d = 1.f / (float)D;
  for (int row = wave; row < rows; row += nwaves) {
    const unsigned short* xr = x + (long)row * D + lane * NVEC;
    const unsigned short* dyr = dy + (long)row * D + lane * NVEC;
    float mu = RMS ? 0.f : mean[row];
    float rs = rstd[row];
    float xhat[NVEC], g[NVEC];
    float s1 = 0.f, s2 = 0.f;  // sum(dxhat), sum(dxhat*xhat)
#pragma unroll
    for (int v = 0; v < NVEC; v += 8) {
      ushortx8 xv = *reinterpret_cast<const ushortx8*>(xr + v);
      ushortx8 dv = *reinterpret_cast<const ushortx8*>(dyr + v);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float xh = (bf16_bits_to_float(xv[j]) - mu) * rs;
        float dyf = bf16_bits_to_float(dv[j]);
        float dxhat = dyf * w[v + j];
        xhat[v + j] = xh;
        g[v + j] = dxhat;
        s1 += dxhat;
        s2 += dxhat * xh;
        dscale_acc[v + j] += dyf * xh;
        dbias_acc[v + j] += dyf;
      }
    }
    s1 = wave_reduce_sum(s1) * inv_d;
    s2 = wave_reduce_sum(s2) * inv_d;
    unsigned short* dxr = dx + (long)row * D + lane * NVEC;
#pragma unroll
    for (int v = 0; v < NVEC; v += 8) {
      ushortx8 ov;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float val = RMS ? rs * (g[v + j] - xhat[v + j] * s2)
                        : rs * (g[v + j] - s1 - xhat[v + j] * s2);
        ov[j] = float_to_bf16_bits(val);
      }
      *reinterpret_cast<ushortx8*>(dxr + v) = ov;
    }
  }

  float* ds = partials + (long)wave * D + lane * NVEC;
  float* db = partials + (long)nwaves * D + (long)wave * D + lane * NVEC;
#pragma unroll
  for (int v = 0; v < NVEC; ++v) {
    ds[v] = dscale_acc[v];
    db[v] = dbias_acc[v];
  }
}

// Reduce partials [2, nwaves, D] -> dscale [D], dbias [D] (fp32).
// Split-K over the waves dim (grid.y); coalesced column reads; one fp32
// atomicAdd per (thread, output). dscale/dbias must be zero-initialized.
__global__ void ln_bwd_reduce(const float* __restrict__ partials,
                              float* __restrict__ dscale,
                              float* __restrict__ dbias, int nwaves, int D) {
  int col = blockIdx.x * blockDim.x + threadIdx.x;
  if (col >= D) return;
  const int chunk = (nwaves + gridDim.y - 1) / gridDim.y;
  const int w0 = blockIdx.y * chunk;
  const int w1 = min(nwaves, w0 + chunk);
  float s = 0.f, b = 0.f;
  for (int w = w0; w < w1; ++w) {
    s += partials[(long)w * D + col];
    b += partials[(long)(nwaves + w) * D + col];
  }
  if (gridDim.y == 1) {
    dscale[col] = s;
    dbias[col] = b;
  } else {
    atomicAdd(dscale + col, s);
    atomicAdd(dbias + col, b);
  }
}

// ---- large-D path: one block (256 threads) per row, 8 cols per thread ----
// The wave-per-row kernel's per-lane D/64-wide register arrays spill to
// scratch for D >= 2048 (observed 388 B/lane); here each thread owns 8
// columns per pass so register use is constant in D.
template <bool RMS, int ITERS>
__global__ __launch_bounds__(kBlock) void ln_fwd_big(
    const unsigned short* __restrict__ x,
    const unsigned short* __restrict__ scale,
    const unsigned short* __restrict__ bias,
    unsigned short* __restrict__ y, float* __restrict__ mean_out,
    float* __restrict__ rstd_out, int rows, int D, float eps) {
  __shared__ float scratch[kWavesPerBlock];
  const int tid = threadIdx.x;
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const unsigned short* xr = x + (long)row * D;
    float sum = 0.f, sumsq = 0.f;
#pragma unroll
    for (int it = 0; it < ITERS; ++it) {
      int base = (it * kBlock + tid) * 8;
      if (base < D) {
        ushortx8 v = *reinterpret_cast<const ushortx8*>(xr + base);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float f = bf16_bits_to_float(v[j]);
          sum += f;
          sumsq += f * f;
        }
      }
    }
    sum = block_reduce_sum<kWavesPerBlock>(sum, scratch);
    sumsq = block_reduce_sum<kWavesPerBlock>(sumsq, scratch);
    float mu = RMS ? 0.f : sum / D;
    float rstd = rsqrtf(sumsq / D - mu * mu + eps);
    if (tid == 0) {
      if (!RMS) mean_out[row] = mu;
      rstd_out[row] = rstd;
    }
    unsigned short* yr = y + (long)row * D;
#pragma unroll
    for (int it = 0; it < ITERS; ++it) {
      int base = (it * kBlock + tid) * 8;
      if (base < D) {
        ushortx8 v = *reinterpret_cast<const ushortx8*>(xr + base);
        ushortx8 ws = *reinterpret_cast<const ushortx8*>(scale + base);
        ushortx8 ov;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float xhat = (bf16_bits_to_float(v[j]) - mu) * rstd;
          float w = 1.f + bf16_bits_to_float(ws[j]);
          float bb =
              RMS ? 0.f
                  : bf16_bits_to_float(
                        reinterpret_cast<const unsigned short*>(bias)[base +
                                                                      j]);
          ov[j] = float_to_bf16_bits(xhat * w + bb);
        }
        *reinterpret_cast<ushortx8*>(yr + base) = ov;
      }
    }
    __syncthreads();
  }
}

template <bool RMS, int ITERS>
__global__ __launch_bounds__(kBlock) void ln_bwd_big(
    const unsigned short* __restrict__ dy,
    const unsigned short* __restrict__ x,
    const unsigned short* __restrict__ scale,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    unsigned short* __restrict__ dx, float* __restrict__ partials,
    int rows, int D) {
  __shared__ float scratch[kWavesPerBlock];
  const int tid = threadIdx.x;
  float ds_acc[ITERS][8];
  float db_acc[ITERS][8];
#pragma unroll
  for (int it = 0; it < ITERS; ++it)
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      ds_acc[it][j] = 0.f;
      db_acc[it][j] = 0.f;
    }
  const float inv_d = 1.f / (float)D;
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const unsigned short* xr = x + (long)row * D;
    const unsigned short* dyr = dy + (long)row * D;
    float mu = RMS ? 0.f : mean[row];
    float rs = rstd[row];
    float s1 = 0.f, s2 = 0.f;
#pragma unroll
    for (int it = 0; it < ITERS; ++it) {
      int base = (it * kBlock + tid) * 8;
      if (base < D) {
        ushortx8 xv = *reinterpret_cast<const ushortx8*>(xr + base);
        ushortx8 dv = *reinterpret_cast<const ushortx8*>(dyr + base);
        ushortx8 ws = *reinterpret_cast<const ushortx8*>(scale + base);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float xh = (bf16_bits_to_float(xv[j]) - mu) * rs;
          float dyf = bf16_bits_to_float(dv[j]);
          float w = 1.f + bf16_bits_to_float(ws[j]);
          float dxhat = dyf * w;
          s1 += dxhat;
          s2 += dxhat * xh;
          ds_acc[it][j] += dyf * xh;
          db_acc[it][j] += dyf;
        }
      }
    }
    s1 = block_reduce_sum<kWavesPerBlock>(s1, scratch) * inv_d;
    s2 = block_reduce_sum<kWavesPerBlock>(s2, scratch) * inv_d;
    unsigned short* dxr = dx + (long)row * D;
#pragma unroll
    for (int it = 0; it < ITERS; ++it) {
      int base = (it * kBlock + tid) * 8;
      if (base < D) {
        ushortx8 xv = *reinterpret_cast<const ushortx8*>(xr + base);
        ushortx8 dv = *reinterpret_cast<const ushortx8*>(dyr + base);
        ushortx8 ws = *reinterpret_cast<const ushortx8*>(scale + base);
        ushortx8 ov;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float xh = (bf16_bits_to_float(xv[j]) - mu) * rs;
          float dxhat = bf16_bits_to_float(dv[j]) *
                        (1.f + bf16_bits_to_float(ws[j]));
          float val = RMS ? rs * (dxhat - xh * s2)
                          : rs * (dxhat - s1 - xh * s2);
          ov[j] = float_to_bf16_bits(val);
        }
        *reinterpret_cast<ushortx8*>(dxr + base) = ov;
      }
    }
    __syncthreads();
  }
  // Column partials: [2, nblocks, D].
  float* ds = partials + (long)blockIdx.x * D;
  float* db = partials + (long)(gridDim.x + blockIdx.x) * D;
#pragma unroll
  for (int it = 0; it < ITERS; ++it) {
    int base = (it * kBlock + tid) * 8;
    if (base < D) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        ds[base + j] = ds_acc[it][j];
        db[base + j] = db_acc[it][j];
      }
    }
  }
}

// ---- generic-D fallback (scalar, one block per row) -----------------------
template <bool RMS>
__global__ void ln_fwd_generic(const unsigned short* __restrict__ x,
                               const unsigned short* __restrict__ scale,
                               const unsigned short* __restrict__ bias,
                               unsigned short* __restrict__ y,
                               float* __restrict__ mean_out,
                               float* __restrict__ rstd_out, int rows, int D,
                               float eps) {
  __shared__ float scratch[kWavesPerBlock];
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const unsigned short* xr = x + (long)row * D;
    float sum = 0.f, sumsq = 0.f;
    for (int i = threadIdx.x; i < D; i += kBlock) {
      float f = bf16_bits_to_float(xr[i]);
      sum += f;
      sumsq += f * f;
    }
    sum = block_reduce_sum<kWavesPerBlock>(sum, scratch);
    sumsq = block_reduce_sum<kWavesPerBlock>(sumsq, scratch);
    float mu = RMS ? 0.f : sum / D;
    float rstd = rsqrtf(sumsq / D - mu * mu + eps);
    if (threadIdx.x == 0) {
      if (!RMS) mean_out[row] = mu;
      rstd_out[row] = rstd;
    }
    unsigned short* yr = y + (long)row * D;
    for (int i = threadIdx.x; i < D; i += kBlock) {
      float xhat = (bf16_bits_to_float(xr[i]) - mu) * rstd;
      float w = 1.f + bf16_bits_to_float(scale[i]);
      float b = RMS ? 0.f : bf16_bits_to_float(bias[i]);
      yr[i] = float_to_bf16_bits(xhat * w + b);
    }
    __syncthreads();
  }
}

template <bool RMS>
__global__ void ln_bwd_generic(const unsigned short* __restrict__ dy,
                               const unsigned short* __restrict__ x,
                               const unsigned short* __restrict__ scale,
                               const float* __restrict__ mean,
                               const float* __restrict__ rstd,
                               unsigned short* __restrict__ dx,
                               float* __restrict__ partials, int rows,
                               int D) {
  __shared__ float scratch[kWavesPerBlock];
  float* ds = partials + (long)blockIdx.x * D;
  float* db = partials + (long)(gridDim.x + blockIdx.x) * D;
  for (int i = threadIdx.x; i < D; i += kBlock) {
    ds[i] = 0.f;
    db[i] = 0.f;
  }
  __syncthreads();
  const float inv_d = 1.f / (float)D;
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const unsigned short* xr = x + (long)row * D;
    const unsigned short* dyr = dy + (long)row * D;
    float mu = RMS ? 0.f : mean[row];
    float rs = rstd[row];
    float s1 = 0.f, s2 = 0.f;
    for (int i = threadIdx.x; i < D; i += kBlock) {
      float xh = (bf16_bits_to_float(xr[i]) - mu) * rs;
      float dyf = bf16_bits_to_float(dyr[i]);
      float w = 1.f + bf16_bits_to_float(scale[i]);
      float dxhat = dyf * w;
      s1 += dxhat;
      s2 += dxhat * xh;
      ds[i] += dyf * xh;
      db[i] += dyf;
    }
    s1 = block_reduce_sum<kWavesPerBlock>(s1, scratch) * inv_d;
    s2 = block_reduce_sum<kWavesPerBlock>(s2, scratch) * inv_d;
    unsigned short* dxr = dx + (long)row * D;
    for (int i = threadIdx.x; i < D; i += kBlock) {
      float xh = (bf16_bits_to_float(xr[i]) - mu) * rs;
      float w = 1.f + bf16_bits_to_float(scale[i]);
      float dxhat = dyr ? bf16_bits_to_float(dyr[i]) * w : 0.f;
      float val = RMS ? rs * (dxhat - xh * s2) : rs * (dxhat - s1 - xh * s2);
      dxr[i] = float_to_bf16_bits(val);
    }
    __syncthreads();
  }
}

inline bool use_vec_path(long D) {
  long nvec = D / WAVE_SIZE;
  // Larger NVEC spills to scratch (D/64-wide per-lane arrays); D >= 1536
  // takes the block-per-row big path instead.
  return D % (WAVE_SIZE * 8) == 0 && (nvec == 8 || nvec == 16);
}

inline int big_iters(long D) {  // 0 = not eligible
  if (D % 8 != 0) return 0;
  long iters = (D + 8 * 256 - 1) / (8 * 256);
  return iters <= 4 ? (int)iters : 0;
}

}  // namespace

// ---- host wrappers --------------------------------------------------------
std::vector<torch::Tensor> layer_norm_fwd(torch::Tensor x, torch::Tensor scale,
                                          torch::Tensor bias, double eps,
                                          bool rms) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous(), "x must be contiguous CUDA");
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "ln kernel expects bf16");
  const long D = x.size(-1);
  const long rows = x.numel() / D;
  auto y = torch::empty_like(x);
  auto opts = x.options().dtype(torch::kFloat32);
  auto mean = torch::empty({rms ? 0 : rows}, opts);
  auto rstd = torch::empty({rows}, opts);
  auto stream = at::cuda::getCurrentCUDAStream();

  const unsigned short* xp = (const unsigned short*)x.data_ptr();
  const unsigned short* sp = (const unsigned short*)scale.data_ptr();
  const unsigned short* bp =
      rms ? nullptr : (const unsigned short*)bias.data_ptr();
  unsigned short* yp = (unsigned short*)y.data_ptr();
  float* mp = rms ? nullptr : mean.data_ptr<float>();
  float* rp = rstd.data_ptr<float>();

  if (use_vec_path(D)) {
    int grid = memory_bound_grid(rows, kWavesPerBlock);
    const int nvec = (int)(D / WAVE_SIZE);
#define LN_FWD_CASE(NV)                                                     \
  case NV:                                                                  \
    if (rms)                                                                \
      hipLaunchKernelGGL((ln_fwd_bf16<NV, true>), dim3(grid), dim3(kBlock), \
                         0, stream, xp, sp, bp, yp, mp, rp, (int)rows,      \
                         (int)D, (float)eps);                               \
    else                                                                    \
      hipLaunchKernelGGL((ln_fwd_bf16<NV, false>), dim3(grid),              \
                         dim3(kBlock), 0, stream, xp, sp, bp, yp, mp, rp,   \
                         (int)rows, (int)D, (float)eps);                    \
    break;
    switch (nvec) {
      LN_FWD_CASE(8)
      LN_FWD_CASE(16)
      LN_FWD_CASE(24)
      LN_FWD_CASE(32)
      LN_FWD_CASE(64)
      default:
        TORCH_CHECK(false, "unhandled NVEC");
    }
#undef LN_FWD_CASE
  } else if (big_iters(D)) {
    int grid = (int)std::min<long>(rows, 2048);
#define LN_FWD_BIG(IT)                                                     \
  case IT:                                                                 \
    if (rms)                                                               \
      hipLaunchKernelGGL((ln_fwd_big<true, IT>), dim3(grid), dim3(kBlock), \
                         0, stream, xp, sp, bp, yp, mp, rp, (int)rows,     \
                         (int)D, (float)eps);                              \
    else                                                                   \
      hipLaunchKernelGGL((ln_fwd_big<false, IT>), dim3(grid),              \
                         dim3(kBlock), 0, stream, xp, sp, bp, yp, mp, rp,  \
                         (int)rows, (int)D, (float)eps);                   \
    break;
    switch (big_iters(D)) {
      LN_FWD_BIG(1)
      LN_FWD_BIG(2)
      LN_FWD_BIG(3)
      LN_FWD_BIG(4)
    }
#undef LN_FWD_BIG
  } else {
    int grid = memory_bound_grid(rows * kBlock, kBlock, 1024);
    if (rms)
      hipLaunchKernelGGL((ln_fwd_generic<true>), dim3(grid), dim3(kBlock), 0,
                         stream, xp, sp, bp, yp, mp, rp, (int)rows, (int)D,
                         (float)eps);
    else
      hipLaunchKernelGGL((ln_fwd_generic<false>), dim3(grid), dim3(kBlock), 0,
                         stream, xp, sp, bp, yp, mp, rp, (int)rows, (int)D,
                         (float)eps);
  }
  return {y, mean, rstd};
}

std::vector<torch::Tensor> layer_norm_bwd(torch::Tensor dy, torch::Tensor x,
                                          torch::Tensor scale,
                                          torch::Tensor mean,
                                          torch::Tensor rstd, bool rms) {
  const long D = x.size(-1);
  const long rows = x.numel() / D;
  auto dx = torch::empty_like(x);
  auto opts = x.options().dtype(torch::kFloat32);
  auto stream = at::cuda::getCurrentCUDAStream();

  const unsigned short* dyp = (const unsigned short*)dy.data_ptr();
  const unsigned short* xp = (const unsigned short*)x.data_ptr();
  const unsigned short* sp = (const unsigned short*)scale.data_ptr();
  const float* mp = rms ? nullptr : mean.data_ptr<float>();
  const float* rp = rstd.data_ptr<float>();
  unsigned short* dxp = (unsigned short*)dx.data_ptr();

  torch::Tensor dscale = torch::zeros({D}, opts);
  torch::Tensor dbias = torch::zeros({D}, opts);

  if (use_vec_path(D)) {
    int grid = memory_bound_grid(rows, kWavesPerBlock, 512);
    int nwaves = grid * kWavesPerBlock;
    auto partials = torch::empty({2L * nwaves, D}, opts);
    const int nvec = (int)(D / WAVE_SIZE);
#define LN_BWD_CASE(NV)                                                  \
  case NV:                                                               \
    if (rms)                                                             \
      hipLaunchKernelGGL((ln_bwd_bf16<NV, true>), dim3(grid),            \
                         dim3(kBlock), 0, stream, dyp, xp, sp, mp, rp,   \
                         dxp, partials.data_ptr<float>(), (int)rows,     \
                         (int)D);                                        \
    else                                                                 \
      hipLaunchKernelGGL((ln_bwd_bf16<NV, false>), dim3(grid),           \
                         dim3(kBlock), 0, stream, dyp, xp, sp, mp, rp,   \
                         dxp, partials.data_ptr<float>(), (int)rows,     \
                         (int)D);                                        \
    break;
    switch (nvec) {
      LN_BWD_CASE(8)
      LN_BWD_CASE(16)
      LN_BWD_CASE(24)
      LN_BWD_CASE(32)
      LN_BWD_CASE(64)
      default:
        TORCH_CHECK(false, "unhandled NVEC");
    }
#undef LN_BWD_CASE
    int splitk = (int)std::min<long>(64, std::max<long>(1, nwaves / 8));
    hipLaunchKernelGGL(ln_bwd_reduce, dim3(cdiv(D, 256), splitk), dim3(256),
                       0, stream, partials.data_ptr<float>(),
                       dscale.data_ptr<float>(), dbias.data_ptr<float>(),
                       nwaves, (int)D);
  } else if (big_iters(D)) {
    int grid = (int)std::min<long>(rows, 1024);
    auto partials = torch::empty({2L * grid, D}, opts);
#define LN_BWD_BIG(IT)                                                    \
  case IT:                                                                \
    if (rms)                                                              \
      hipLaunchKernelGGL((ln_bwd_big<true, IT>), dim3(grid),              \
                         dim3(kBlock), 0, stream, dyp, xp, sp, mp, rp,    \
                         dxp, partials.data_ptr<float>(), (int)rows,      \
                         (int)D);                                         \
    else                                                                  \
      hipLaunchKernelGGL((ln_bwd_big<false, IT>), dim3(grid),             \
                         dim3(kBlock), 0, stream, dyp, xp, sp, mp, rp,    \
                         dxp, partials.data_ptr<float>(), (int)rows,      \
                         (int)D);                                         \
    break;
    switch (big_iters(D)) {
      LN_BWD_BIG(1)
      LN_BWD_BIG(2)
      LN_BWD_BIG(3)
      LN_BWD_BIG(4)
    }
#undef LN_BWD_BIG
    int splitk3 = (int)std::min<long>(64, std::max<long>(1, grid / 8));
    hipLaunchKernelGGL(ln_bwd_reduce, dim3(cdiv(D, 256), splitk3),
                       dim3(256), 0, stream, partials.data_ptr<float>(),
                       dscale.data_ptr<float>(), dbias.data_ptr<float>(),
                       grid, (int)D);
  } else {
    int grid = memory_bound_grid(rows, 1, 256);
    auto partials = torch::empty({2L * grid, D}, opts);
    if (rms)
      hipLaunchKernelGGL((ln_bwd_generic<true>), dim3(grid), dim3(kBlock), 0,
                         stream, dyp, xp, sp, mp, rp, dxp,
                         partials.data_ptr<float>(), (int)rows, (int)D);
    else
      hipLaunchKernelGGL((ln_bwd_generic<false>), dim3(grid), dim3(kBlock), 0,
                         stream, dyp, xp, sp, mp, rp, dxp,
                         partials.data_ptr<float>(), (int)rows, (int)D);
    int splitk2 = (int)std::min<long>(64, std::max<long>(1, grid / 8));
    hipLaunchKernelGGL(ln_bwd_reduce, dim3(cdiv(D, 256), splitk2), dim3(256),
                       0, stream, partials.data_ptr<float>(),
                       dscale.data_ptr<float>(), dbias.data_ptr<float>(),
                       grid, (int)D);
  }
  return {dx, dscale, dbias};
}
