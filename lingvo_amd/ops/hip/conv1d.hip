// Depthwise 1-D time convolution fwd+bwd for gfx950 (SURVEY.md K7;
// reference op: lingvo/core/conv_layers_with_time_padding.py:608
// DepthwiseConv2DLayer / :717 CausalDepthwiseConv2DLayer with time-major
// paddings).
//
// x: [B, T, D] bf16 contiguous, w: [K, D] bf16, y: [B, T, D].
//   y[b,t,d] = sum_j x[b, t + j - pad, d] * w[j, d]   (+ bias[d])
// pad = K-1 for causal, (K-1)/2 for SAME. Out-of-range taps read 0.
// Inputs are pre-masked by paddings on the Python side; outputs are
// re-masked there too.
//
// Memory-bound: vectorized ushort8 IO, one thread per 8 channels of one
// (b, t) row, grid-stride; taps hit L1/L2 (row re-use across t).

#include <ATen/cuda/CUDAContext.h>
#include <torch/extension.h>

#include "common.h"

namespace {

constexpr int MAXK = 33;

// KT > 0: compile-time tap count (full unroll + software pipelining);
// KT == 0: runtime K fallback.
template <int KT>
__global__ void dwconv_fwd(const unsigned short* __restrict__ x,
                           const unsigned short* __restrict__ w,
                           const unsigned short* __restrict__ bias,
                           unsigned short* __restrict__ y, int B, int T,
                           int D, int Krt, int pad) {
  const int K = KT > 0 ? KT : Krt;
  const long nvec = (long)B * T * (D / 8);
  const int dvec = D / 8;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long)gridDim.x * blockDim.x) {
    const int dv = (int)(i % dvec);
    const long bt = i / dvec;
    const int t = (int)(bt % T);
    const long b = bt / T;
    float acc[8];
    if (bias) {
      ushortx8 bv = *reinterpret_cast<const ushortx8*>(bias + dv * 8);
#pragma unroll
      for (int e = 0; e < 8; ++e) acc[e] = bf16_bits_to_float(bv[e]);
    } else {
#pragma unroll
      for (int e = 0; e < 8; ++e) acc[e] = 0.f;
    }
#pragma unroll
    for (int j = 0; j < (KT > 0 ? KT : 1); ++j) {
      if (KT == 0) break;
      int ts = t + j - pad;
      if (ts < 0 || ts >= T) continue;
      ushortx8 xv = *reinterpret_cast<const ushortx8*>(
          x + (b * T + ts) * D + dv * 8);
      ushortx8 wv = *reinterpret_cast<const ushortx8*>(w + j * D + dv * 8);
#pragma unroll
      for (int e = 0; e < 8; ++e)
        acc[e] += bf16_bits_to_float(xv[e]) * bf16_bits_to_float(wv[e]);
    }
    if (KT == 0) {
      for (int j = 0; j < K; ++j) {
        int ts = t + j - pad;
        if (ts < 0 || ts >= T) continue;
        ushortx8 xv = *reinterpret_cast<const ushortx8*>(
            x + (b * T + ts) * D + dv * 8);
        ushortx8 wv =
            *reinterpret_cast<const ushortx8*>(w + j * D + dv * 8);
#pragma unroll
        for (int e = 0; e < 8; ++e)
          acc[e] += bf16_bits_to_float(xv[e]) * bf16_bits_to_float(wv[e]);
      }
    }
    ushortx8 ov;
#pragma unroll
    for (int e = 0; e < 8; ++e) ov[e] = float_to_bf16_bits(acc[e]);
    *reinterpret_cast<ushortx8*>(y + (b * T + t) * D + dv * 8) = ov;
  }
}

// Register-window fwd variant (A/B, default OFF — measured SLOWER:
// 0.77 vs 0.44 ms at the bench shape): thread computes R consecutive
// t rows of one 8-channel vec with taps in registers and a shifting x
// window, cutting x re-reads from KT to (KT+R-1)/R per output. The
// loss shows the plain kernel is NOT L1-bound; the ~128-VGPR tap
// array costs more occupancy than the traffic saving returns. Kept
// behind LINGVO_DWCONV_REG=1 as measured methodology.
template <int KT, int R>
__global__ __launch_bounds__(256) void dwconv_fwd_reg(
    const unsigned short* __restrict__ x,
    const unsigned short* __restrict__ w,
    const unsigned short* __restrict__ bias,
    unsigned short* __restrict__ y, int B, int T, int D, int pad) {
  const int dvec = D / 8;
  const int tblks = (T + R - 1) / R;
  const long nwork = (long)B * tblks * dvec;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nwork;
       i += (long)gridDim.x * blockDim.x) {
    const int dv = (int)(i % dvec);
    const long bt = i / dvec;
    const int t0 = (int)(bt % tblks) * R;
    const long b = bt / tblks;
    // Tap registers for this channel octet.
    ushortx8 wv[KT];
#pragma unroll
    for (int j = 0; j < KT; ++j) {
      wv[j] = *reinterpret_cast<const ushortx8*>(w + j * D + dv * 8);
    }
    float acc[R][8];
#pragma unroll
    for (int r = 0; r < R; ++r) {
      if (bias) {
        ushortx8 bv = *reinterpret_cast<const ushortx8*>(bias + dv * 8);
#pragma unroll
        for (int e = 0; e < 8; ++e) acc[r][e] = bf16_bits_to_float(bv[e]);
      } else {
#pragma unroll
        for (int e = 0; e < 8; ++e) acc[r][e] = 0.f;
      }
    }
    const long rowbase = b * T;
    auto loadx = [&](int t) -> ushortx8 {
      ushortx8 v;
      if (t >= 0 && t < T) {
        v = *reinterpret_cast<const ushortx8*>(
            x + (rowbase + t) * D + dv * 8);
      } else {
#pragma unroll
        for (int e = 0; e < 8; ++e) v[e] = 0;
      }
      return v;
    };
    // Window W[r] = x[t0 + r + j - pad] for the current tap j.
    ushortx8 W[R];
#pragma unroll
    for (int r = 0; r < R; ++r) W[r] = loadx(t0 + r - pad);
#pragma unroll
    for (int j = 0; j < KT; ++j) {
#pragma unroll
      for (int r = 0; r < R; ++r) {
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          acc[r][e] += bf16_bits_to_float(W[r][e]) *
                       bf16_bits_to_float(wv[j][e]);
        }
      }
      if (j + 1 < KT) {
#pragma unroll
        for (int r = 0; r + 1 < R; ++r) W[r] = W[r + 1];
        W[R - 1] = loadx(t0 + (R - 1) + (j + 1) - pad);
      }
    }
#pragma unroll
    for (int r = 0; r < R; ++r) {
      const int t = t0 + r;
      if (t >= T) continue;
      ushortx8 ov;
#pragma unroll
      for (int e = 0; e < 8; ++e) ov[e] = float_to_bf16_bits(acc[r][e]);
      *reinterpret_cast<ushortx8*>(y + (rowbase + t) * D + dv * 8) = ov;
    }
  }
}

template <int KT>
__global__ void dwconv_bwd_dx(const unsigned short* __restrict__ dy,
                              const unsigned short* __restrict__ w,
                              unsigned short* __restrict__ dx, int B, int T,
                              int D, int Krt, int pad) {
  const int K = KT > 0 ? KT : Krt;
  const long nvec = (long)B * T * (D / 8);
  const int dvec = D / 8;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long)gridDim.x * blockDim.x) {
    const int dv = (int)(i % dvec);
    const long bt = i / dvec;
    const int t = (int)(bt % T);
    const long b = bt / T;
    float acc[8];
#pragma unroll
    for (int e = 0; e < 8; ++e) acc[e] = 0.f;
#pragma unroll
    for (int j = 0; j < (KT > 0 ? KT : 1); ++j) {
      if (KT == 0) break;
      int ty = t - j + pad;  // y position whose tap j touched x[t]
      if (ty < 0 || ty >= T) continue;
      ushortx8 gv = *reinterpret_cast<const ushortx8*>(
          dy + (b * T + ty) * D + dv * 8);
      ushortx8 wv = *reinterpret_cast<const ushortx8*>(w + j * D + dv * 8);
#pragma unroll
      for (int e = 0; e < 8; ++e)
        acc[e] += bf16_bits_to_float(gv[e]) * bf16_bits_to_float(wv[e]);
    }
    if (KT == 0) {
      for (int j = 0; j < K; ++j) {
        int ty = t - j + pad;
        if (ty < 0 || ty >= T) continue;
        ushortx8 gv = *reinterpret_cast<const ushortx8*>(
            dy + (b * T + ty) * D + dv * 8);
        ushortx8 wv =
            *reinterpret_cast<const ushortx8*>(w + j * D + dv * 8);
#pragma unroll
        for (int e = 0; e < 8; ++e)
          acc[e] += bf16_bits_to_float(gv[e]) * bf16_bits_to_float(wv[e]);
      }
    }
    ushortx8 ov;
#pragma unroll
    for (int e = 0; e < 8; ++e) ov[e] = float_to_bf16_bits(acc[e]);
    *reinterpret_cast<ushortx8*>(dx + (b * T + t) * D + dv * 8) = ov;
  }
}

// dw[j,d] = sum_{b,t} dy[b,t,d] * x[b,t+j-pad,d]; db[d] = sum dy.
// LDS-staged: each block owns 128 channels x a 64-row chunk; x and dy
// tiles are staged once (2-pass global traffic total), then 256 threads
// (d, tap-parity) accumulate from LDS. K/2 fp32 tap accumulators per
// thread; one atomicAdd per (tap, channel) per chunk-group.
constexpr int DW_CHUNK = 64;
constexpr int DW_DBLK = 128;

__global__ __launch_bounds__(256) void dwconv_bwd_dw(
    const unsigned short* __restrict__ dy,
    const unsigned short* __restrict__ x,
    float* __restrict__ dw_part,  // [ngroups, K, D] plain stores
    float* __restrict__ db_part,  // [ngroups, D]
    int B, int T, int D, int K, int pad,
    int ngroups) {
  // Thread decomposition (fully vectorized LDS reads): 16 d-groups of
  // 8 channels x 4 tap-quarters of 8 taps x 4 row-quarters of 16 rows.
  // Every x/dy access is a ushortx8 load; accumulators live in
  // registers across all chunks and flush once.
  // Row stride padded by 8 ushorts (4 banks): with an unpadded 128-
  // ushort (64-bank) stride, the 4 tap-quarter groups of a wave read
  // x_s rows 8 apart that land on IDENTICAL banks -> 4-way conflict on
  // every inner-loop LDS read.
  __shared__ unsigned short x_s[DW_CHUNK + MAXK - 1][DW_DBLK + 8];
  __shared__ unsigned short dy_s[DW_CHUNK][DW_DBLK + 8];
  const int tid = threadIdx.x;
  const int dg = tid & 15;          // d-group (8 channels)
  const int tq = (tid >> 4) & 3;    // tap quarter (taps 8tq..8tq+7)
  const int rq = tid >> 6;          // row quarter (rows 16rq..16rq+15)
  const int d0 = blockIdx.x * DW_DBLK + dg * 8;
  const long rows = (long)B * T;
  const int xrows = DW_CHUNK + K - 1;

  float dw8[8][8];
#pragma unroll
  for (int jj = 0; jj < 8; ++jj) {
#pragma unroll
    for (int e = 0; e < 8; ++e) dw8[jj][e] = 0.f;
  }
  float db8[8] = {0, 0, 0, 0, 0, 0, 0, 0};

  const long nchunks = (rows + DW_CHUNK - 1) / DW_CHUNK;
  for (long c = blockIdx.y; c < nchunks; c += ngroups) {
    const long r0 = c * DW_CHUNK;
    for (int i = tid; i < xrows * (DW_DBLK / 8); i += 256) {
      int row = i / (DW_DBLK / 8);
      int d8 = (i % (DW_DBLK / 8)) * 8;
      long fr = r0 - pad + row;
      ushortx8 v;
      int gd = blockIdx.x * DW_DBLK + d8;
      if (fr >= 0 && fr < rows && gd + 7 < D) {
        v = *reinterpret_cast<const ushortx8*>(x + fr * D + gd);
      } else {
        for (int e = 0; e < 8; ++e) v[e] = 0;
      }
      *reinterpret_cast<ushortx8*>(&x_s[row][d8]) = v;
    }
    for (int i = tid; i < DW_CHUNK * (DW_DBLK / 8); i += 256) {
      int row = i / (DW_DBLK / 8);
      int d8 = (i % (DW_DBLK / 8)) * 8;
      long fr = r0 + row;
      ushortx8 v;
      int gd = blockIdx.x * DW_DBLK + d8;
      if (fr < rows && gd + 7 < D) {
        v = *reinterpret_cast<const ushortx8*>(dy + fr * D + gd);
      } else {
        for (int e = 0; e < 8; ++e) v[e] = 0;
      }
      *reinterpret_cast<ushortx8*>(&dy_s[row][d8]) = v;
    }
    __syncthreads();

    const int nrows = (int)min((long)DW_CHUNK, rows - r0);
    const int lr_hi = min((rq + 1) * 16, nrows);
    for (int lr = rq * 16; lr < lr_hi; ++lr) {
      ushortx8 gv = *reinterpret_cast<const ushortx8*>(&dy_s[lr][dg * 8]);
      float g[8];
#pragma unroll
      for (int e = 0; e < 8; ++e) g[e] = bf16_bits_to_float(gv[e]);
      const int t = (int)((r0 + lr) % T);
      if (tq == 0) {
#pragma unroll
        for (int e = 0; e < 8; ++e) db8[e] += g[e];
      }
#pragma unroll
      for (int jj = 0; jj < 8; ++jj) {
        const int j = tq * 8 + jj;
        if (j >= K) break;
        const int ts = t + j - pad;
        if (ts < 0 || ts >= T) continue;
        ushortx8 xv =
            *reinterpret_cast<const ushortx8*>(&x_s[lr + j][dg * 8]);
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          dw8[jj][e] += g[e] * bf16_bits_to_float(xv[e]);
        }
      }
    }
    __syncthreads();
  }

  // Cross-wave (rq) reduction in LDS, then ONE plain store per element
  // into this block's [K, DW_DBLK] partial slice. The previous global
  // atomicAdd flush serialized ~4k adds per dw address across the grid
  // and dominated the kernel's runtime.
  __shared__ float dw_lds[MAXK - 1][DW_DBLK];
  __shared__ float db_lds[DW_DBLK];
  for (int i = tid; i < K * DW_DBLK; i += 256) {
    dw_lds[i / DW_DBLK][i % DW_DBLK] = 0.f;
  }
  for (int i = tid; i < DW_DBLK; i += 256) db_lds[i] = 0.f;
  __syncthreads();
#pragma unroll
  for (int jj = 0; jj < 8; ++jj) {
    const int j = tq * 8 + jj;
    if (j >= K) break;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      if (dw8[jj][e] != 0.f) atomicAdd(&dw_lds[j][dg * 8 + e], dw8[jj][e]);
    }
  }
  if (tq == 0) {
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      if (db8[e] != 0.f) atomicAdd(&db_lds[dg * 8 + e], db8[e]);
    }
  }
  __syncthreads();
  const long gbase = (long)blockIdx.y * K * D;
  for (int i = tid; i < K * DW_DBLK; i += 256) {
    const int j = i / DW_DBLK;
    const int d = blockIdx.x * DW_DBLK + i % DW_DBLK;
    if (d < D) dw_part[gbase + (long)j * D + d] = dw_lds[j][i % DW_DBLK];
  }
  for (int i = tid; i < DW_DBLK; i += 256) {
    const int d = blockIdx.x * DW_DBLK + i;
    if (d < D) db_part[(long)blockIdx.y * D + d] = db_lds[i];
  }
}

// Sums the per-group partials: dw[j][d] = sum_g dw_part[g][j][d].
__global__ void dwconv_bwd_dw_reduce(const float* __restrict__ dw_part,
                                     const float* __restrict__ db_part,
                                     float* __restrict__ dw,
                                     float* __restrict__ db, long kd, int D,
                                     int ngroups) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < kd + D;
       i += (long)gridDim.x * blockDim.x) {
    float sum = 0.f;
    if (i < kd) {
      for (int g = 0; g < ngroups; ++g) sum += dw_part[(long)g * kd + i];
      dw[i] = sum;
    } else {
      const long d = i - kd;
      for (int g = 0; g < ngroups; ++g) sum += db_part[(long)g * D + d];
      db[d] = sum;
    }
  }
}

}  // namespace

torch::Tensor dwconv1d_fwd(torch::Tensor x, torch::Tensor w,
                           c10::optional<torch::Tensor> bias, int64_t pad) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 3 &&
              x.scalar_type() == torch::kBFloat16, "x must be bf16 [B,T,D]");
  TORCH_CHECK(x.size(2) % 8 == 0, "D must be divisible by 8");
  const int B = x.size(0), T = x.size(1), D = x.size(2), K = w.size(0);
  TORCH_CHECK(K < MAXK, "kernel size < ", MAXK);
  auto y = torch::empty_like(x);
  auto stream = at::cuda::getCurrentCUDAStream();
  long nvec = (long)B * T * (D / 8);
  int grid = memory_bound_grid(nvec, 256);
  // Measured: full K unroll (KT=32) raises register pressure and costs
  // ~50% (8.2 -> 12.5 ms/step at the bench shape). Runtime-K path wins
  // for the naive form; LINGVO_DWCONV_REG=1 selects the register-window
  // variant (A/B knob).
  static const bool use_reg = []() {
    const char* e = getenv("LINGVO_DWCONV_REG");
    return e && e[0] == '1';
  }();
  if (use_reg && K == 32) {
    long nwork = (long)B * ((T + 3) / 4) * (D / 8);
    hipLaunchKernelGGL((dwconv_fwd_reg<32, 4>),
                       dim3(memory_bound_grid(nwork, 256)), dim3(256), 0,
                       stream, (const unsigned short*)x.data_ptr(),
                       (const unsigned short*)w.data_ptr(),
                       bias.has_value() ? (const unsigned short*)
                                              bias->data_ptr()
                                        : nullptr,
                       (unsigned short*)y.data_ptr(), B, T, D, (int)pad);
    return y;
  }
  auto fwd_kern = dwconv_fwd<0>;
  hipLaunchKernelGGL(fwd_kern, dim3(grid), dim3(256), 0, stream,
                     (const unsigned short*)x.data_ptr(),
                     (const unsigned short*)w.data_ptr(),
                     bias.has_value() ? (const unsigned short*)
                                            bias->data_ptr()
                                      : nullptr,
                     (unsigned short*)y.data_ptr(), B, T, D, K, (int)pad);
  return y;
}

std::vector<torch::Tensor> dwconv1d_bwd(torch::Tensor dy, torch::Tensor x,
                                        torch::Tensor w, int64_t pad) {
  const int B = x.size(0), T = x.size(1), D = x.size(2), K = w.size(0);
  auto dx = torch::empty_like(x);
  auto opts = x.options().dtype(torch::kFloat32);
  auto dw = torch::zeros({K, D}, opts);
  auto db = torch::zeros({D}, opts);
  auto stream = at::cuda::getCurrentCUDAStream();
  long nvec = (long)B * T * (D / 8);
  auto dx_kern = dwconv_bwd_dx<0>;
  hipLaunchKernelGGL(dx_kern, dim3(memory_bound_grid(nvec, 256)),
                     dim3(256), 0, stream,
                     (const unsigned short*)dy.data_ptr(),
                     (const unsigned short*)w.data_ptr(),
                     (unsigned short*)dx.data_ptr(), B, T, D, K, (int)pad);
  long nchunks = ((long)B * T + DW_CHUNK - 1) / DW_CHUNK;
  int dblks = (D + DW_DBLK - 1) / DW_DBLK;
  int ngroups = (int)std::min<long>(std::max<long>(1, 1024 / dblks),
                                    nchunks);
  auto dw_part = torch::empty({(long)ngroups * K * D}, opts);
  auto db_part = torch::empty({(long)ngroups * D}, opts);
  dim3 grid_w(dblks, ngroups);
  hipLaunchKernelGGL(dwconv_bwd_dw, grid_w, dim3(256), 0, stream,
                     (const unsigned short*)dy.data_ptr(),
                     (const unsigned short*)x.data_ptr(),
                     dw_part.data_ptr<float>(), db_part.data_ptr<float>(),
                     B, T, D, K, (int)pad, ngroups);
  const long kd = (long)K * D;
  hipLaunchKernelGGL(dwconv_bwd_dw_reduce,
                     dim3((int)std::min<long>((kd + D + 255) / 256, 1024)),
                     dim3(256), 0, stream, dw_part.data_ptr<float>(),
                     db_part.data_ptr<float>(), dw.data_ptr<float>(),
                     db.data_ptr<float>(), kd, D, ngroups);
  return {dx, dw, db};
}
