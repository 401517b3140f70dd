// Fused LAS attention-decoder step kernels for gfx950 (SURVEY K4/K11;
// reference decode-path einsums lingvo/core/batch_major_attention.py:920,
// 1069 and tasks/asr/decoder.py teacher-forced loop).
//
// The teacher-forced recurrence is latency-bound: library GEMMs at M=B
// (=128) occupy ~10 of 256 CUs and cost ~107 us each. These kernels
// replace them:
//  - smallm_gemm: C[M,N] (+)= alpha * A[M,K] @ Wt[N,K]^T (+ pre): MFMA
//    16x16x32, ONE workgroup per 16-column n-tile so the whole chip is
//    busy; A stays L2-resident across workgroups. ~5-15 us per call.
//  - attend_fwd/attend_bwd: the dot-attention step (logits + masked
//    softmax + context, and its full backward incl. dEnc accumulation)
//    as one kernel per direction, one workgroup per batch row.
//
// Consumed by models/asr.py DecoderRecurrence (custom autograd Function
// that also batches all weight-gradient GEMMs across the L timesteps).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"
#include "mfma.h"

namespace {

// ---------------------------------------------------------------------------
// smallm_gemm: out[M,N] = alpha * (A[M,K] @ Wt[N,K]^T) + pre
//   pre_mode: 0 none, 1 row-vector [N], 2 full matrix [M,N],
//             3 accumulate into existing out
// A row-major [M, K] (contiguous), Wt row-major [N, ldw] with the
// mathematical W^T in its first K columns. M <= 128, K % 32 == 0.
// Grid: (ceil(N/16)); block 256 = 4 waves; wave w owns m-tiles
// {2w, 2w+1} (rows 32w .. 32w+31).
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void smallm_gemm_kernel(
    const unsigned short* __restrict__ a, const unsigned short* __restrict__ wt,
    const unsigned short* __restrict__ pre, unsigned short* __restrict__ out,
    int m, int n, int k, long ldw, float alpha, int pre_mode) {
  // K-SPLIT layout: every wave computes ALL 8 m-tiles over its quarter
  // of K (no per-iteration block syncs; the serial K chain is 4x
  // shorter than an M-split), then waves 1-3 spill partials to LDS and
  // wave 0 reduces + writes. The recurrence calls this back-to-back
  // with dependent inputs, so per-call LATENCY is the metric.
  const int nt = blockIdx.x;          // n-tile (16 cols)
  const int col0 = nt * 16;
  const int m0 = blockIdx.y * 128;    // m-block (rows handled here)
  const int wid = threadIdx.x / WAVE_SIZE;
  const int lane = threadIdx.x & 63;
  const int g = lane >> 4;            // k-group 0..3
  const int cl = lane & 15;           // row-in-frag / col-in-frag
  const int col = col0 + cl;
  const bool col_ok = col < n;

  constexpr int MT = 8;               // m-tiles per block (128 rows)
  __shared__ float red[3][128][16];   // waves 1-3 partials

  const int kq = (k / 32 + 3) / 4 * 32;  // per-wave K quota (mult of 32)
  const int k_lo = wid * kq;
  const int k_hi = min(k, k_lo + kq);

  f32x4 acc[MT];
#pragma unroll
  for (int mt = 0; mt < MT; ++mt) acc[mt] = {0.f, 0.f, 0.f, 0.f};

  for (int k0 = k_lo; k0 < k_hi; k0 += 32) {
    bf16x8 bf;
    if (col_ok) {
      bf = load_bf16x8_bits(wt + (long)col * ldw + k0 + g * 8);
    } else {
      float z[8] = {0, 0, 0, 0, 0, 0, 0, 0};
      bf = pack_bf16x8(z);
    }
#pragma unroll
    for (int mt = 0; mt < MT; ++mt) {
      const int row = m0 + mt * 16 + cl;
      bf16x8 af;
      if (row < m) {
        af = load_bf16x8_bits(a + (long)row * k + k0 + g * 8);
      } else {
        float z[8] = {0, 0, 0, 0, 0, 0, 0, 0};
        af = pack_bf16x8(z);
      }
      acc[mt] = mfma16x16x32_bf16(af, bf, acc[mt]);
    }
  }

  // Cross-wave reduction: C layout row = mt*16 + g*4 + r, col = cl.
  if (wid > 0) {
#pragma unroll
    for (int mt = 0; mt < MT; ++mt) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        red[wid - 1][mt * 16 + g * 4 + r][cl] = acc[mt][r];
      }
    }
  }
  __syncthreads();
  if (wid != 0) return;
#pragma unroll
  for (int mt = 0; mt < MT; ++mt) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int lrow = mt * 16 + g * 4 + r;  // LDS-local row
      const int row = m0 + lrow;
      if (row >= m || !col_ok) continue;
      float v = (acc[mt][r] + red[0][lrow][cl] + red[1][lrow][cl] +
                 red[2][lrow][cl]) * alpha;
      long off = (long)row * n + col;
      if (pre_mode == 1) {
        v += bf16_bits_to_float(pre[col]);
      } else if (pre_mode == 2) {
        v += bf16_bits_to_float(pre[off]);
      } else if (pre_mode == 3) {
        v += bf16_bits_to_float(out[off]);
      }
      out[off] = float_to_bf16_bits(v);
    }
  }
}

// ---------------------------------------------------------------------------
// attend_fwd: per batch row b:
//   logits[s] = scale * dot(q[b], enc[b,s]) + (-1e30 if padded)
//   probs = softmax(logits); ctx[b] = sum_s probs[s] * enc[b,s]
// q [B, D] bf16; enc [B, S, D] bf16; pad [B, S] float; out probs [B, S]
// fp32, ctx [B, D] bf16. Grid: B blocks of 256 (4 waves).
// Dynamic LDS: q (D bf16) + logits (S f32) + 4*D f32 ctx partials.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void attend_fwd_kernel(
    const unsigned short* __restrict__ q, const unsigned short* __restrict__ enc,
    const float* __restrict__ pad, float* __restrict__ probs,
    unsigned short* __restrict__ ctx, int b_total, int s_len, int d,
    float scale) {
  const int b = blockIdx.x;
  const int wid = threadIdx.x / WAVE_SIZE;
  const int lane = threadIdx.x & 63;
  extern __shared__ char smem[];
  unsigned short* q_s = (unsigned short*)smem;            // [D]
  float* logit_s = (float*)(q_s + d);                     // [S]
  float* ctx_s = logit_s + s_len;                         // [4][D]
  float* red = ctx_s + 4 * d;                             // [4]

  for (int i = threadIdx.x; i < d; i += 256) q_s[i] = q[(long)b * d + i];
  __syncthreads();

  // Cache this lane's q slice in registers (vectorized over the s loop).
  constexpr int MAXDV = 4;  // supports d up to 2048
  const int ndv = d / (WAVE_SIZE * 8) + (d % (WAVE_SIZE * 8) ? 1 : 0);
  float qreg[MAXDV][8];
  for (int v = 0; v < ndv; ++v) {
    const int i = lane * 8 + v * WAVE_SIZE * 8;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      qreg[v][j] = i < d ? bf16_bits_to_float(q_s[i + j]) : 0.f;
    }
  }

  // Pass 1: logits. Each wave processes 4 consecutive s per iteration
  // (independent loads pipeline the HBM/L2 latency that a one-s-at-a-
  // time loop serializes).
  const unsigned short* eb = enc + (long)b * s_len * d;
  for (int s0 = wid * 4; s0 < s_len; s0 += 16) {
    float part[4] = {0.f, 0.f, 0.f, 0.f};
    for (int v = 0; v < ndv; ++v) {
      const int i = lane * 8 + v * WAVE_SIZE * 8;
      if (i < d) {
        bf16x8 ev[4];
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          if (s0 + u < s_len) {
            ev[u] = load_bf16x8_bits(eb + (long)(s0 + u) * d + i);
          }
        }
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          if (s0 + u < s_len) {
#pragma unroll
            for (int j = 0; j < 8; ++j) {
              part[u] += qreg[v][j] * (float)ev[u][j];
            }
          }
        }
      }
    }
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      if (s0 + u < s_len) {
        float t = wave_reduce_sum(part[u]);
        if (lane == 0) {
          logit_s[s0 + u] = t * scale +
              (pad[(long)b * s_len + s0 + u] > 0.5f ? -1e30f : 0.f);
        }
      }
    }
  }
  __syncthreads();

  // Softmax over logits (block-wide).
  float lmax = -1e30f;
  for (int s = threadIdx.x; s < s_len; s += 256) {
    lmax = fmaxf(lmax, logit_s[s]);
  }
  lmax = wave_reduce_max(lmax);
  if (lane == 0) red[wid] = lmax;
  __syncthreads();
  lmax = fmaxf(fmaxf(red[0], red[1]), fmaxf(red[2], red[3]));
  float lsum = 0.f;
  for (int s = threadIdx.x; s < s_len; s += 256) {
    float e = __expf(logit_s[s] - lmax);
    logit_s[s] = e;
    lsum += e;
  }
  lsum = wave_reduce_sum(lsum);
  __syncthreads();
  if (lane == 0) red[wid] = lsum;
  __syncthreads();
  lsum = red[0] + red[1] + red[2] + red[3];
  const float inv = lsum > 0.f ? 1.f / lsum : 0.f;

  // Pass 2: probs out + ctx accumulation (per-wave private partials).
  for (int i = threadIdx.x; i < 4 * d; i += 256) ctx_s[i] = 0.f;
  __syncthreads();
  float* my_ctx = ctx_s + wid * d;
  for (int s0 = wid * 4; s0 < s_len; s0 += 16) {
    float pv[4];
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      pv[u] = s0 + u < s_len ? logit_s[s0 + u] * inv : 0.f;
      if (lane == 0 && s0 + u < s_len) {
        probs[(long)b * s_len + s0 + u] = pv[u];
      }
    }
    for (int v = 0; v < ndv; ++v) {
      const int i = lane * 8 + v * WAVE_SIZE * 8;
      if (i < d) {
        bf16x8 ev[4];
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          if (s0 + u < s_len) {
            ev[u] = load_bf16x8_bits(eb + (long)(s0 + u) * d + i);
          }
        }
        float accv[8] = {0, 0, 0, 0, 0, 0, 0, 0};
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          if (s0 + u < s_len) {
#pragma unroll
            for (int j = 0; j < 8; ++j) {
              accv[j] += pv[u] * (float)ev[u][j];
            }
          }
        }
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          my_ctx[i + j] += accv[j];
        }
      }
    }
  }
  __syncthreads();
  for (int i = threadIdx.x; i < d; i += 256) {
    ctx[(long)b * d + i] = float_to_bf16_bits(
        ctx_s[i] + ctx_s[d + i] + ctx_s[2 * d + i] + ctx_s[3 * d + i]);
  }
}

// ---------------------------------------------------------------------------
// attend_bwd: per batch row b, given dctx [B,D] (bf16), probs [B,S] f32,
// q [B,D], enc:
//   dprobs[s] = dot(dctx, enc[b,s])
//   t = sum_s probs[s]*dprobs[s];  dl[s] = probs[s]*(dprobs[s]-t)*scale
//   dq[b] = sum_s dl[s]*enc[b,s]
//   denc[b,s] += dl[s]*q[b] + probs[s]*dctx      (fp32 accumulation)
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void attend_bwd_kernel(
    const unsigned short* __restrict__ dctx,
    const float* __restrict__ probs, const unsigned short* __restrict__ q,
    const unsigned short* __restrict__ enc, unsigned short* __restrict__ dq,
    float* __restrict__ denc, int b_total, int s_len, int d, float scale) {
  const int b = blockIdx.x;
  const int wid = threadIdx.x / WAVE_SIZE;
  const int lane = threadIdx.x & 63;
  extern __shared__ char smem[];
  unsigned short* dctx_s = (unsigned short*)smem;         // [D]
  unsigned short* q_s = dctx_s + d;                       // [D]
  float* dl_s = (float*)(q_s + d);                        // [S]
  float* dq_s = dl_s + s_len;                             // [4][D]
  float* red = dq_s + 4 * d;                              // [4]

  for (int i = threadIdx.x; i < d; i += 256) {
    dctx_s[i] = dctx[(long)b * d + i];
    q_s[i] = q[(long)b * d + i];
  }
  __syncthreads();

  // Per-lane register caches of dctx and q slices.
  constexpr int MAXDV = 4;
  const int ndv = d / (WAVE_SIZE * 8) + (d % (WAVE_SIZE * 8) ? 1 : 0);
  float dcreg[MAXDV][8], qreg[MAXDV][8];
  for (int v = 0; v < ndv; ++v) {
    const int i = lane * 8 + v * WAVE_SIZE * 8;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      dcreg[v][j] = i < d ? bf16_bits_to_float(dctx_s[i + j]) : 0.f;
      qreg[v][j] = i < d ? bf16_bits_to_float(q_s[i + j]) : 0.f;
    }
  }

  // Pass 1: dprobs (stored to dl_s) + t reduction.
  const unsigned short* eb = enc + (long)b * s_len * d;
  const float* pb = probs + (long)b * s_len;
  for (int s0 = wid * 4; s0 < s_len; s0 += 16) {
    float part[4] = {0.f, 0.f, 0.f, 0.f};
    for (int v = 0; v < ndv; ++v) {
      const int i = lane * 8 + v * WAVE_SIZE * 8;
      if (i < d) {
        bf16x8 ev[4];
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          if (s0 + u < s_len) {
            ev[u] = load_bf16x8_bits(eb + (long)(s0 + u) * d + i);
          }
        }
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          if (s0 + u < s_len) {
#pragma unroll
            for (int j = 0; j < 8; ++j) {
              part[u] += dcreg[v][j] * (float)ev[u][j];
            }
          }
        }
      }
    }
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      if (s0 + u < s_len) {
        float t = wave_reduce_sum(part[u]);
        if (lane == 0) dl_s[s0 + u] = t;
      }
    }
  }
  __syncthreads();
  float t_part = 0.f;
  for (int s = threadIdx.x; s < s_len; s += 256) {
    t_part += pb[s] * dl_s[s];
  }
  t_part = wave_reduce_sum(t_part);
  if (lane == 0) red[wid] = t_part;
  __syncthreads();
  const float t = red[0] + red[1] + red[2] + red[3];

  // dl[s] = probs[s] * (dprobs[s] - t) * scale.
  __syncthreads();
  for (int s = threadIdx.x; s < s_len; s += 256) {
    dl_s[s] = pb[s] * (dl_s[s] - t) * scale;
  }
  __syncthreads();

  // Pass 2: dq partials + denc writes.
  for (int i = threadIdx.x; i < 4 * d; i += 256) dq_s[i] = 0.f;
  __syncthreads();
  float* my_dq = dq_s + wid * d;
  float* db = denc + (long)b * s_len * d;
  for (int s0 = wid * 4; s0 < s_len; s0 += 16) {
    float dlv[4], ppv[4];
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      dlv[u] = s0 + u < s_len ? dl_s[s0 + u] : 0.f;
      ppv[u] = s0 + u < s_len ? pb[s0 + u] : 0.f;
    }
    for (int v = 0; v < ndv; ++v) {
      const int i = lane * 8 + v * WAVE_SIZE * 8;
      if (i < d) {
        bf16x8 ev[4];
        floatx4 lo[4], hi[4];
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          if (s0 + u < s_len) {
            ev[u] = load_bf16x8_bits(eb + (long)(s0 + u) * d + i);
            floatx4* dbv =
                reinterpret_cast<floatx4*>(db + (long)(s0 + u) * d + i);
            lo[u] = dbv[0];
            hi[u] = dbv[1];
          }
        }
        float dqacc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          if (s0 + u < s_len) {
#pragma unroll
            for (int j = 0; j < 4; ++j) {
              dqacc[j] += dlv[u] * (float)ev[u][j];
              lo[u][j] += dlv[u] * qreg[v][j] + ppv[u] * dcreg[v][j];
            }
#pragma unroll
            for (int j = 4; j < 8; ++j) {
              dqacc[j] += dlv[u] * (float)ev[u][j];
              hi[u][j - 4] += dlv[u] * qreg[v][j] + ppv[u] * dcreg[v][j];
            }
          }
        }
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          if (s0 + u < s_len) {
            floatx4* dbv =
                reinterpret_cast<floatx4*>(db + (long)(s0 + u) * d + i);
            dbv[0] = lo[u];
            dbv[1] = hi[u];
          }
        }
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          my_dq[i + j] += dqacc[j];
        }
      }
    }
  }
  __syncthreads();
  for (int i = threadIdx.x; i < d; i += 256) {
    dq[(long)b * d + i] = float_to_bf16_bits(
        dq_s[i] + dq_s[d + i] + dq_s[2 * d + i] + dq_s[3 * d + i]);
  }
}

}  // namespace

// ---------------------------------------------------------------------------
// Host wrappers
// ---------------------------------------------------------------------------

void smallm_gemm(torch::Tensor a, torch::Tensor wt,
                 c10::optional<torch::Tensor> pre, torch::Tensor out,
                 int64_t k, int64_t wt_col0, double alpha,
                 int64_t pre_mode) {
  TORCH_CHECK(a.is_cuda() && a.is_contiguous() &&
              a.scalar_type() == torch::kBFloat16, "a bf16 contig");
  TORCH_CHECK(wt.is_cuda() && wt.is_contiguous() &&
              wt.scalar_type() == torch::kBFloat16, "wt bf16 contig");
  TORCH_CHECK(out.is_contiguous() && out.scalar_type() == torch::kBFloat16);
  const int m = a.size(0);
  const int n = out.size(1);
  TORCH_CHECK(a.size(1) >= k && k % 32 == 0, "K%32");
  TORCH_CHECK(k == a.size(1), "a must be [M,K] exactly");

  TORCH_CHECK(wt.size(0) >= n, "wt rows >= N");
  const long ldw = wt.size(1);
  TORCH_CHECK(wt_col0 + k <= ldw, "wt K slice OOB");
  const unsigned short* prep = nullptr;
  if (pre_mode == 1 || pre_mode == 2) {
    TORCH_CHECK(pre.has_value() && pre->is_contiguous() &&
                pre->scalar_type() == torch::kBFloat16);
    prep = (const unsigned short*)pre->data_ptr();
  }
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(smallm_gemm_kernel,
                     dim3(cdiv(n, 16), cdiv(m, 128)), dim3(256), 0,
                     stream, (const unsigned short*)a.data_ptr(),
                     (const unsigned short*)wt.data_ptr() + wt_col0,
                     prep, (unsigned short*)out.data_ptr(), m, n, (int)k,
                     ldw, (float)alpha, (int)pre_mode);
}

std::vector<torch::Tensor> attend_fwd(torch::Tensor q, torch::Tensor enc,
                                      torch::Tensor pad, double scale) {
  TORCH_CHECK(q.is_cuda() && q.is_contiguous() &&
              q.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(enc.is_contiguous() &&
              enc.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(pad.is_contiguous() && pad.scalar_type() == torch::kFloat32);
  const int b = q.size(0), d = q.size(1);
  const int s = enc.size(1);
  TORCH_CHECK(d % 8 == 0, "D % 8 == 0");
  auto probs = torch::empty({b, s}, q.options().dtype(torch::kFloat32));
  auto ctx = torch::empty({b, d}, q.options());
  size_t shmem = (size_t)d * 2 + (size_t)s * 4 + (size_t)4 * d * 4 + 16;
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(attend_fwd_kernel, dim3(b), dim3(256), shmem, stream,
                     (const unsigned short*)q.data_ptr(),
                     (const unsigned short*)enc.data_ptr(),
                     pad.data_ptr<float>(), probs.data_ptr<float>(),
                     (unsigned short*)ctx.data_ptr(), b, s, d,
                     (float)scale);
  return {probs, ctx};
}

std::vector<torch::Tensor> attend_bwd(torch::Tensor dctx,
                                      torch::Tensor probs, torch::Tensor q,
                                      torch::Tensor enc, torch::Tensor denc,
                                      double scale) {
  const int b = q.size(0), d = q.size(1);
  const int s = enc.size(1);
  TORCH_CHECK(denc.is_contiguous() &&
              denc.scalar_type() == torch::kFloat32);
  auto dctx_c = dctx.to(torch::kBFloat16).contiguous();
  auto dq = torch::empty({b, d}, q.options());
  size_t shmem = (size_t)d * 4 + (size_t)s * 4 + (size_t)4 * d * 4 + 16;
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(attend_bwd_kernel, dim3(b), dim3(256), shmem, stream,
                     (const unsigned short*)dctx_c.data_ptr(),
                     probs.data_ptr<float>(),
                     (const unsigned short*)q.data_ptr(),
                     (const unsigned short*)enc.data_ptr(),
                     (unsigned short*)dq.data_ptr(),
                     denc.data_ptr<float>(), b, s, d, (float)scale);
  return {dq};
}
