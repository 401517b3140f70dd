// Fused flash-attention forward+backward for gfx950 (SURVEY.md K2-K4;
// reference einsums: lingvo/core/batch_major_attention.py:508-514 and
// AttenProbs:943 — BTNH layouts, causal/local/padding masks, clipped
// relative-position bias, GQA).
//
// Structure (v1, correctness-first; guide §B "fused attention prefill"):
//  - fwd: grid (ceil(T/64), N, B); block = 4 waves; each wave owns a
//    16-row Q strip. K staged in LDS row-major (XOR-swizzled), V staged
//    transposed [H][64]; online softmax in fp32 with per-row (m, l)
//    tracked in the 16-lane C-layout groups; P routed through per-wave
//    LDS for the PV MFMA. Saves LSE for backward.
//  - bwd: grid (ceil(S/64), N_kv, B); each block owns a 64-key tile,
//    loops q-heads of the GQA group and q-tiles; recomputes P^T from
//    LSE; accumulates dK/dV in registers (written once, no atomics);
//    dQ accumulated via fp32 global atomics; optional clipped
//    rel-position bias gradient via LDS accumulation.
//
// Masks: key k is visible from query q iff
//   k < klen[b]  AND  (win_l < 0 OR k >= q - win_l)
//                AND  (win_r < 0 OR k <= q + win_r).
// Causal = win_r 0. LocalSelfAttention (batch_major_attention.py:2656)
// = finite (win_l, win_r).

#include <ATen/cuda/CUDAContext.h>
#include <torch/extension.h>

#include "mfma.h"

namespace {

constexpr int QT = 64;   // q rows per block (fwd) / q rows per inner tile (bwd)
constexpr int KT = 64;   // keys per tile
constexpr int NW = 4;    // waves per block
constexpr int BLOCK = NW * WAVE_SIZE;
constexpr float NEG_INF = -1e30f;

// ---------------------------------------------------------------------------
// LDS staging helpers. Row-major tiles are stored with a per-row XOR
// swizzle on 16B granules (mfma.h swz) so B-fragment ds_read_b128 at
// row-stride >= 128B doesn't bank-conflict (guide §6 G4).
// ---------------------------------------------------------------------------

// Stage ROWS x H bf16 from global (row i at src + i*src_stride elems) into
// LDS row-major with swizzle. Rows >= valid_rows are zeroed.
template <int H, int ROWS, int NT = BLOCK>
__device__ void stage_regular(const unsigned short* src, long src_stride,
                              int valid_rows, char* dst) {
  constexpr int ROWB = H * 2;
  constexpr int TPR = ROWB / 16;             // threads per row (16B each)
  constexpr int RPP = NT / TPR;              // rows per pass
  const int tid = threadIdx.x;
  const int r0 = tid / TPR;
  const int byte0 = (tid % TPR) * 16;
#pragma unroll
  for (int rp = 0; rp < (ROWS + RPP - 1) / RPP; ++rp) {
    int row = r0 + rp * RPP;
    // constexpr-folded for ROWS % RPP == 0 (the fwd tile shapes).
    if ((ROWS % RPP) != 0 && row >= ROWS) break;
    ushortx8 v;
    if (row < valid_rows) {
      v = *reinterpret_cast<const ushortx8*>(src + (long)row * src_stride +
                                             byte0 / 2);
    } else {
      for (int j = 0; j < 8; ++j) v[j] = 0;
    }
    *reinterpret_cast<ushortx8*>(dst + row * ROWB + swz(row, byte0)) = v;
  }
}

// Stage ROWS x H bf16 from global into LDS TRANSPOSED as [H][ROWS]
// (ROWS=64, 128B rows), swizzled per h-row. Scalar 2B writes.
template <int H, int ROWS, int NT = BLOCK, int PITCHB = ROWS * 2>
__device__ void stage_transposed(const unsigned short* src, long src_stride,
                                 int valid_rows, char* dst) {
  // PITCHB must be >= 128 when the swizzle is in play: swz() XORs up to
  // 112 bytes into the row, so a 64-byte pitch would corrupt neighbors.
  constexpr int ROWB = H * 2;
  constexpr int TPR = ROWB / 16;
  constexpr int RPP = NT / TPR;
  const int tid = threadIdx.x;
  const int r0 = tid / TPR;
  const int h0 = (tid % TPR) * 8;
#pragma unroll
  for (int rp = 0; rp < (ROWS + RPP - 1) / RPP; ++rp) {
    int row = r0 + rp * RPP;
    if ((ROWS % RPP) != 0 && row >= ROWS) break;
    ushortx8 v;
    if (row < valid_rows) {
      v = *reinterpret_cast<const ushortx8*>(src + (long)row * src_stride +
                                             h0);
    } else {
      for (int j = 0; j < 8; ++j) v[j] = 0;
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int h = h0 + j;
      *reinterpret_cast<unsigned short*>(
          dst + h * PITCHB + swz(h, row * 2)) = v[j];
    }
  }
}

// ds_read a B-fragment (8 bf16) from a swizzled row-major LDS tile.
__device__ __forceinline__ bf16x8 lds_frag(const char* tile, int row,
                                           int rowb, int byte_in_row) {
  union {
    ushortx8 u;
    bf16x8 h;
  } cv;
  cv.u = *reinterpret_cast<const ushortx8*>(tile + row * rowb +
                                            swz(row, byte_in_row));
  return cv.h;
}

__device__ __forceinline__ bool visible(int q, int k, int klen, int win_l,
                                        int win_r, int chunk_lo,
                                        int chunk_hi) {
  if (k >= klen) return false;
  if (win_l >= 0 && k < q - win_l) return false;
  if (win_r >= 0 && k > q + win_r) return false;
  // Chunkwise mask (reference ChunkwiseSelfAttention,
  // batch_major_attention.py:4008): caller precomputes the visible key
  // range [chunk_lo, chunk_hi) PER QUERY (one division per q, not per
  // element — per-element integer division cost 1.7x fwd time).
  if (k < chunk_lo || k >= chunk_hi) return false;
  return true;
}

// ---------------------------------------------------------------------------
// Forward
// ---------------------------------------------------------------------------
constexpr int FWD_NW = 8;                 // waves per fwd block
constexpr int FWD_BLOCK = FWD_NW * WAVE_SIZE;
constexpr int FQT = FWD_NW * 16;          // q rows per fwd block

template <int H, int KTF, bool SEG>
__global__ __launch_bounds__(FWD_BLOCK) void fa_fwd_kernel(
    const unsigned short* __restrict__ q, const unsigned short* __restrict__ k,
    const unsigned short* __restrict__ v,
    const int* __restrict__ klen_ptr,          // [B] or null
    const unsigned short* __restrict__ bias,   // [N][2C+1] or null
    const int* __restrict__ qseg,              // [B][T] or null (packed)
    const int* __restrict__ kseg,              // [B][S] or null
    unsigned short* __restrict__ o, float* __restrict__ lse, int B, int T,
    int S, int N, int NKV, int win_l, int win_r, int bias_clip, float scale,
    int chunk, int lc) {
  constexpr int ROWB = H * 2;
  constexpr int KH = H / 32;   // mfma K-steps over head dim
  constexpr int HF = H / 16;   // output col frags
  constexpr int NF = KTF / 16; // key frags per strip
  extern __shared__ char smem[];
  char* k_lds = smem;                        // [KTF][H] swz
  char* vt_lds = k_lds + KTF * ROWB;         // [H][KTF] swz
  char* p_lds = vt_lds + H * KTF * 2;        // [FWD_NW][16][KTF] swz

  const int qt = blockIdx.x;
  const int n = blockIdx.y;
  const int b = blockIdx.z;
  const int nkv = n / (N / NKV);
  const int wid = threadIdx.x / WAVE_SIZE;
  const int lane = threadIdx.x & 63;
  const int g = lane >> 4;       // 16-lane group
  const int cl = lane & 15;      // col-in-frag
  const int klen = klen_ptr ? klen_ptr[b] : S;

  const int q0 = qt * FQT + wid * 16;  // this wave's first q row

  // Q fragments (A layout): row = cl, k = g*8 + kk*32. Folded zeros for
  // rows >= T.
  bf16x8 qfrag[KH];
#pragma unroll
  for (int kk = 0; kk < KH; ++kk) {
    int qrow = q0 + cl;
    if (qrow < T) {
      qfrag[kk] = load_bf16x8_bits(
          q + (((long)b * T + qrow) * N + n) * H + kk * 32 + g * 8);
    } else {
      float z[8] = {0, 0, 0, 0, 0, 0, 0, 0};
      qfrag[kk] = pack_bf16x8(z);
    }
  }

  // Per-lane chunk windows for its 4 q rows (one division each).
  int cw_lo[4], cw_hi[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    if (chunk > 0) {
      const int qc = (q0 + g * 4 + r) / chunk;
      cw_lo[r] = (qc - lc) * chunk;
      cw_hi[r] = (qc + 1) * chunk;
    } else {
      cw_lo[r] = 0;
      cw_hi[r] = 0x7fffffff;
    }
  }

  f32x4 acc_o[HF];
#pragma unroll
  for (int hf = 0; hf < HF; ++hf) acc_o[hf] = {0.f, 0.f, 0.f, 0.f};
  float m_run[4], l_run[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    m_run[r] = NEG_INF;
    l_run[r] = 0.f;
  }

  // KV tile range from the window.
  int kmax_excl =
      min(klen, win_r < 0 ? S : min(S, qt * FQT + FQT - 1 + win_r + 1));
  int kmin = win_l < 0 ? 0 : max(0, qt * FQT - win_l);
  if (chunk > 0) {
    // Skip key tiles entirely outside the q-tile's chunk window.
    const int qc_lo = (qt * FQT) / chunk;
    const int qc_hi = (qt * FQT + FQT - 1) / chunk;
    kmin = max(kmin, max(0, (qc_lo - lc)) * chunk);
    kmax_excl = min(kmax_excl, (qc_hi + 1) * chunk);
    if (kmax_excl <= kmin) {
      // No visible keys for this whole q-tile: emit zeros (fully-masked
      // rows output 0, matching the reference) and a NEG_INF lse.
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qrow = q0 + g * 4 + r;
        if (qrow >= T) continue;
        unsigned short* orow = o + (((long)b * T + qrow) * N + n) * H;
#pragma unroll
        for (int hf = 0; hf < HF; ++hf) {
          orow[hf * 16 + cl] = float_to_bf16_bits(0.f);
        }
        if (cl == 0) lse[((long)b * N + n) * T + qrow] = NEG_INF;
      }
      return;
    }
  }
  const int kt_lo = kmin / KTF;
  const int kt_hi = (max(kmax_excl, 1) - 1) / KTF;

  for (int kt = kt_lo; kt <= kt_hi; ++kt) {
    const int kbase = kt * KTF;
    // Stage K and V^T.
    stage_regular<H, KTF, FWD_BLOCK>(
        k + (((long)b * S + kbase) * NKV + nkv) * H, (long)NKV * H,
        klen - kbase, k_lds);
    stage_transposed<H, KTF, FWD_BLOCK>(
        v + (((long)b * S + kbase) * NKV + nkv) * H, (long)NKV * H,
        klen - kbase, vt_lds);
    __syncthreads();

    // S strip: 16 q rows x KTF keys, fp32.
    float s[NF][4];  // [nf over keys][r]
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int nf = 0; nf < NF; ++nf) {
      f32x4 acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kk = 0; kk < KH; ++kk) {
        bf16x8 bfr = lds_frag(k_lds, nf * 16 + cl, ROWB, (kk * 32 + g * 8) * 2);
        acc = mfma16x16x32_bf16(qfrag[kk], bfr, acc);
      }
#pragma unroll
      for (int r = 0; r < 4; ++r) s[nf][r] = acc[r];
    }
    __builtin_amdgcn_s_setprio(0);

    // Scale + mask + bias.
#pragma unroll
    for (int nf = 0; nf < NF; ++nf) {
      const int kcol = kbase + nf * 16 + cl;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qrow = q0 + g * 4 + r;
        float val = s[nf][r] * scale;
        if (bias) {
          int d = qrow - kcol;
          d = d < -bias_clip ? -bias_clip : (d > bias_clip ? bias_clip : d);
          val += bf16_bits_to_float(
              bias[(long)n * (2 * bias_clip + 1) + d + bias_clip]);
        }
        if (!visible(qrow, kcol, klen, win_l, win_r, cw_lo[r], cw_hi[r]) ||
            qrow >= T)
          val = NEG_INF;
        if (SEG) {
          if (kcol < S && qrow < T &&
              qseg[(long)b * T + qrow] != kseg[(long)b * S + kcol])
            val = NEG_INF;
        }
        s[nf][r] = val;
      }
    }

    // Online softmax. Row r lives across the 16 lanes with this lane's g.
    float rowmax[4], rowsum[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float mx = s[0][r];
#pragma unroll
      for (int nf = 1; nf < NF; ++nf) mx = fmaxf(mx, s[nf][r]);
#pragma unroll
      for (int off = 1; off < 16; off <<= 1) mx = fmaxf(mx, __shfl_xor(mx, off));
      rowmax[r] = mx;
    }
    float alpha[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float m_new = fmaxf(m_run[r], rowmax[r]);
      alpha[r] = (m_run[r] == NEG_INF) ? 0.f : __expf(m_run[r] - m_new);
      m_run[r] = m_new;
      float sum = 0.f;
#pragma unroll
      for (int nf = 0; nf < NF; ++nf) {
        float p = (s[nf][r] <= NEG_INF * 0.5f) ? 0.f
                                               : __expf(s[nf][r] - m_new);
        s[nf][r] = p;
        sum += p;
      }
#pragma unroll
      for (int off = 1; off < 16; off <<= 1) sum += __shfl_xor(sum, off);
      rowsum[r] = sum;
      l_run[r] = l_run[r] * alpha[r] + sum;
#pragma unroll
      for (int hf = 0; hf < HF; ++hf) acc_o[hf][r] *= alpha[r];
    }

    // P -> per-wave LDS (bf16, C layout -> row-major [16][KTF], swizzled).
    char* pw = p_lds + wid * 16 * (KTF * 2);
#pragma unroll
    for (int nf = 0; nf < NF; ++nf) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = g * 4 + r;
        int col = nf * 16 + cl;
        *reinterpret_cast<unsigned short*>(
            pw + row * (KTF * 2) + swz(row, col * 2)) =
            float_to_bf16_bits(s[nf][r]);
      }
    }
    // Same-wave DS ordering makes the reads below safe without a barrier.

    // O += P @ V : A = P (rows=q, k=keys), B from Vt (k=keys, col=h).
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk2 = 0; kk2 < KTF / 32; ++kk2) {
      bf16x8 pa = lds_frag(pw, cl, KTF * 2, (kk2 * 32 + g * 8) * 2);
#pragma unroll
      for (int hf = 0; hf < HF; ++hf) {
        bf16x8 vb =
            lds_frag(vt_lds, hf * 16 + cl, KTF * 2, (kk2 * 32 + g * 8) * 2);
        acc_o[hf] = mfma16x16x32_bf16(pa, vb, acc_o[hf]);
      }
    }
    __builtin_amdgcn_s_setprio(0);
    __syncthreads();
  }

  // Epilogue: divide by l, store O (bf16) and LSE (fp32).
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qrow = q0 + g * 4 + r;
    if (qrow >= T) continue;
    float inv_l = l_run[r] > 0.f ? 1.f / l_run[r] : 0.f;
    unsigned short* orow = o + (((long)b * T + qrow) * N + n) * H;
#pragma unroll
    for (int hf = 0; hf < HF; ++hf) {
      orow[hf * 16 + cl] = float_to_bf16_bits(acc_o[hf][r] * inv_l);
    }
    if (cl == 0) {
      lse[((long)b * N + n) * T + qrow] =
          l_run[r] > 0.f ? m_run[r] + __logf(l_run[r]) : NEG_INF;
    }
  }
}

// ---------------------------------------------------------------------------
// Backward: delta = rowsum(dO * O)
// ---------------------------------------------------------------------------
template <int H>
__global__ void fa_bwd_delta(const unsigned short* __restrict__ dout,
                             const unsigned short* __restrict__ o,
                             float* __restrict__ delta, int B, int T, int N) {
  // One wave per (b, t, n) row.
  const long row = (long)blockIdx.x * NW + threadIdx.x / WAVE_SIZE;
  const int lane = threadIdx.x & 63;
  if (row >= (long)B * T * N) return;
  // row enumerates (b, t, n) in memory order of [B,T,N,H].
  const unsigned short* dp = dout + row * H;
  const unsigned short* op = o + row * H;
  float sum = 0.f;
#pragma unroll
  for (int i = lane; i < H; i += WAVE_SIZE) {
    sum += bf16_bits_to_float(dp[i]) * bf16_bits_to_float(op[i]);
  }
  sum = wave_reduce_sum(sum);
  if (lane == 0) {
    long b = row / ((long)T * N);
    long tn = row % ((long)T * N);
    long t = tn / N, n = tn % N;
    delta[(b * N + n) * T + t] = sum;
  }
}

// ---------------------------------------------------------------------------
// Backward main
// ---------------------------------------------------------------------------
template <int H, bool BIAS_GRAD, int KTB, int NWB, int QTB>
__global__ __launch_bounds__(NWB * WAVE_SIZE) void fa_bwd_kernel(
    const unsigned short* __restrict__ dout,
    const unsigned short* __restrict__ q, const unsigned short* __restrict__ k,
    const unsigned short* __restrict__ v, const float* __restrict__ lse,
    const float* __restrict__ delta, const int* __restrict__ klen_ptr,
    const unsigned short* __restrict__ bias,
    const int* __restrict__ qseg, const int* __restrict__ kseg,
    float* __restrict__ dq_acc,
    unsigned short* __restrict__ dk, unsigned short* __restrict__ dv,
    float* __restrict__ dbias, int B, int T, int S, int N, int NKV,
    int win_l, int win_r, int bias_clip, float scale, int skip,
    int chunk, int lc) {
  constexpr int ROWB = H * 2;
  constexpr int KH = H / 32;
  constexpr int HF = H / 16;
  const int nbias = 2 * bias_clip + 1;
  extern __shared__ char smem[];
  // qt/dot/a tiles keep a 128-byte row pitch regardless of QTB: the
  // swizzle XORs up to 112 bytes within a row.
  constexpr int TP = 128;
  char* q_lds = smem;                        // [QTB][H] swz
  char* qt_lds = q_lds + QTB * ROWB;         // [H][QTB] swz, pitch TP
  char* do_lds = qt_lds + H * TP;            // [QTB][H] swz
  char* dot_lds = do_lds + QTB * ROWB;       // [H][QTB] swz, pitch TP
  char* kt_lds = dot_lds + H * TP;           // [H][KTB] swz
  char* ds_lds = kt_lds + H * KTB * 2;       // [QTB][KTB] swz
  char* a_lds = ds_lds + QTB * KTB * 2;      // [NWB][16][QTB] swz, pitch TP
  float* lse_s = (float*)(a_lds + NWB * 16 * TP);       // [QTB]
  float* delta_s = lse_s + QTB;                         // [QTB]
  float* dbias_s = delta_s + QTB;                       // [nbias] if BIAS_GRAD

  const int kt = blockIdx.x;
  const int nkv = blockIdx.y;
  const int b = blockIdx.z;
  const int group = N / NKV;
  const int wid = threadIdx.x / WAVE_SIZE;
  const int lane = threadIdx.x & 63;
  const int g = lane >> 4;
  const int cl = lane & 15;
  const int klen = klen_ptr ? klen_ptr[b] : S;
  const int kbase = kt * KTB;
  const int k0 = kbase + wid * 16;  // wave's first key

  // Stage K^T once (shared by all heads in the group).
  stage_transposed<H, KTB, NWB * WAVE_SIZE>(
      k + (((long)b * S + kbase) * NKV + nkv) * H, (long)NKV * H,
      klen - kbase, kt_lds);

  // K, V fragments (A layout) from global: row = key (cl), k = h.
  bf16x8 kfrag[KH], vfrag[KH];
  {
    int key = k0 + cl;
    bool ok = key < klen;
#pragma unroll
    for (int kk = 0; kk < KH; ++kk) {
      if (ok) {
        long off = (((long)b * S + key) * NKV + nkv) * H + kk * 32 + g * 8;
        kfrag[kk] = load_bf16x8_bits(k + off);
        vfrag[kk] = load_bf16x8_bits(v + off);
      } else {
        float z[8] = {0, 0, 0, 0, 0, 0, 0, 0};
        kfrag[kk] = pack_bf16x8(z);
        vfrag[kk] = pack_bf16x8(z);
      }
    }
  }

  // q-tile range for this key tile.
  int qlo = win_r < 0 ? 0 : max(0, kbase - win_r);
  int qhi = win_l < 0 ? T - 1 : min(T - 1, kbase + KTB - 1 + win_l);
  if (chunk > 0) {
    const int kc_lo = kbase / chunk;
    const int kc_hi = (min(kbase + KTB, S) - 1) / chunk;
    qlo = max(qlo, kc_lo * chunk);
    qhi = min(qhi, (kc_hi + lc + 1) * chunk - 1);
    if (qhi < qlo) {
      // No visible queries: dK/dV of this key tile are zero. Write them
      // (group==1 outputs are uninitialized buffers).
      if (group == 1) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int key = k0 + g * 4 + r;
          if (key >= S) continue;
          long off = (((long)b * S + key) * NKV + nkv) * H;
#pragma unroll
          for (int hf = 0; hf < HF; ++hf) {
            dk[off + hf * 16 + cl] = float_to_bf16_bits(0.f);
            dv[off + hf * 16 + cl] = float_to_bf16_bits(0.f);
          }
        }
      }
      return;
    }
  }
  const int qt_lo = qlo / QTB, qt_hi = max(qhi, 0) / QTB;

  for (int head = 0; head < group; ++head) {
    const int n = nkv * group + head;

    f32x4 acc_dk[HF], acc_dv[HF];
#pragma unroll
    for (int hf = 0; hf < HF; ++hf) {
      acc_dk[hf] = {0.f, 0.f, 0.f, 0.f};
      acc_dv[hf] = {0.f, 0.f, 0.f, 0.f};
    }
    if (BIAS_GRAD) {
      for (int i = threadIdx.x; i < nbias; i += NWB * WAVE_SIZE)
        dbias_s[i] = 0.f;
    }

    for (int qt2 = qt_lo; qt2 <= qt_hi; ++qt2) {
      const int qb = qt2 * QTB;
      __syncthreads();
      stage_regular<H, QTB, NWB * WAVE_SIZE>(
          q + (((long)b * T + qb) * N + n) * H, (long)N * H, T - qb,
          q_lds);
      stage_regular<H, QTB, NWB * WAVE_SIZE>(
          dout + (((long)b * T + qb) * N + n) * H, (long)N * H, T - qb,
          do_lds);
      if (!(skip & 2)) {
        stage_transposed<H, QTB, NWB * WAVE_SIZE, TP>(
            q + (((long)b * T + qb) * N + n) * H, (long)N * H, T - qb,
            qt_lds);
        stage_transposed<H, QTB, NWB * WAVE_SIZE, TP>(
            dout + (((long)b * T + qb) * N + n) * H, (long)N * H, T - qb,
            dot_lds);
      }
      for (int i = threadIdx.x; i < QTB; i += NWB * WAVE_SIZE) {
        int qrow = qb + i;
        lse_s[i] = qrow < T ? lse[((long)b * N + n) * T + qrow] : NEG_INF;
        delta_s[i] = qrow < T ? delta[((long)b * N + n) * T + qrow] : 0.f;
      }
      __syncthreads();

      // Per-lane chunk windows for this q-tile's 4 q columns.
      constexpr int NQ = QTB / 16;
      int bw_lo[NQ], bw_hi[NQ];
#pragma unroll
      for (int nf = 0; nf < NQ; ++nf) {
        if (chunk > 0) {
          const int qc = (qb + nf * 16 + cl) / chunk;
          bw_lo[nf] = (qc - lc) * chunk;
          bw_hi[nf] = (qc + 1) * chunk;
        } else {
          bw_lo[nf] = 0;
          bw_hi[nf] = 0x7fffffff;
        }
      }

      // S^T strip: rows = 16 keys (this wave), cols = QTB queries.
      float pt[NQ][4];  // P^T
      float dlg[NQ][4];  // dLogits
#pragma unroll
      for (int nf = 0; nf < NQ; ++nf) {
        f32x4 acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int kk = 0; kk < KH; ++kk) {
          bf16x8 bq =
              lds_frag(q_lds, nf * 16 + cl, ROWB, (kk * 32 + g * 8) * 2);
          acc = mfma16x16x32_bf16(kfrag[kk], bq, acc);
        }
        const int qcol = qb + nf * 16 + cl;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int key = k0 + g * 4 + r;
          float val = acc[r] * scale;
          if (bias) {
            int d = qcol - key;
            d = d < -bias_clip ? -bias_clip : (d > bias_clip ? bias_clip : d);
            val += bf16_bits_to_float(bias[(long)n * nbias + d + bias_clip]);
          }
          bool vis = visible(qcol, key, klen, win_l, win_r, bw_lo[nf],
                             bw_hi[nf]) && qcol < T;
          if (vis && qseg &&
              qseg[(long)b * T + qcol] != kseg[(long)b * S + key])
            vis = false;
          float l = lse_s[nf * 16 + cl];
          pt[nf][r] =
              (vis && l > NEG_INF * 0.5f) ? __expf(val - l) : 0.f;
        }
      }

      // P^T -> a_lds; dV += P^T @ dO (B from dOt).
      char* aw = a_lds + wid * 16 * TP;
#pragma unroll
      for (int nf = 0; nf < NQ; ++nf) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int row = g * 4 + r;
          int col = nf * 16 + cl;
          *reinterpret_cast<unsigned short*>(
              aw + row * TP + swz(row, col * 2)) =
              float_to_bf16_bits(pt[nf][r]);
        }
      }
#pragma unroll
      for (int kk2 = 0; kk2 < QTB / 32; ++kk2) {
        bf16x8 pa = lds_frag(aw, cl, TP, (kk2 * 32 + g * 8) * 2);
#pragma unroll
        for (int hf = 0; hf < HF; ++hf) {
          bf16x8 bd =
              lds_frag(dot_lds, hf * 16 + cl, TP, (kk2 * 32 + g * 8) * 2);
          acc_dv[hf] = mfma16x16x32_bf16(pa, bd, acc_dv[hf]);
        }
      }

      // dP^T = V @ dO^T (B from do_lds rows).
#pragma unroll
      for (int nf = 0; nf < NQ; ++nf) {
        f32x4 acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int kk = 0; kk < KH; ++kk) {
          bf16x8 bd =
              lds_frag(do_lds, nf * 16 + cl, ROWB, (kk * 32 + g * 8) * 2);
          acc = mfma16x16x32_bf16(vfrag[kk], bd, acc);
        }
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float dlogits = pt[nf][r] * (acc[r] - delta_s[nf * 16 + cl]);
          dlg[nf][r] = dlogits;
        }
      }

      // dS^T (scaled, bf16) -> a_lds (overwrite); dK += dS^T @ Q (B=Qt).
#pragma unroll
      for (int nf = 0; nf < NQ; ++nf) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int row = g * 4 + r;
          int col = nf * 16 + cl;
          *reinterpret_cast<unsigned short*>(
              aw + row * TP + swz(row, col * 2)) =
              float_to_bf16_bits(dlg[nf][r] * scale);
        }
      }
#pragma unroll
      for (int kk2 = 0; kk2 < QTB / 32; ++kk2) {
        bf16x8 da = lds_frag(aw, cl, TP, (kk2 * 32 + g * 8) * 2);
#pragma unroll
        for (int hf = 0; hf < HF; ++hf) {
          bf16x8 bq =
              lds_frag(qt_lds, hf * 16 + cl, TP, (kk2 * 32 + g * 8) * 2);
          acc_dk[hf] = mfma16x16x32_bf16(da, bq, acc_dk[hf]);
        }
      }

      // Full dS tile to LDS (transposed store: ds_lds[q][key]) for dQ.
      if (!(skip & 4))
#pragma unroll
      for (int nf = 0; nf < NQ; ++nf) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int qrow = nf * 16 + cl;
          int key = wid * 16 + g * 4 + r;
          *reinterpret_cast<unsigned short*>(
              ds_lds + qrow * (KTB * 2) + swz(qrow, key * 2)) =
              float_to_bf16_bits(dlg[nf][r] * scale);
        }
      }
      __syncthreads();

      // dQ strips: rows q = qb + dw*16 + .. (A from ds_lds), B = K from
      // kt_lds; atomic-accumulate fp32. Only QTB/16 strips exist, so with
      // NWB > QTB/16 the extra waves skip this phase.
      if (wid < QTB / 16 && !(skip & 1)) {
        f32x4 acc_dq[HF];
#pragma unroll
        for (int hf = 0; hf < HF; ++hf) acc_dq[hf] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int kk2 = 0; kk2 < KTB / 32; ++kk2) {
          bf16x8 da = lds_frag(ds_lds, wid * 16 + cl, KTB * 2,
                               (kk2 * 32 + g * 8) * 2);
#pragma unroll
          for (int hf = 0; hf < HF; ++hf) {
            bf16x8 bk = lds_frag(kt_lds, hf * 16 + cl, KTB * 2,
                                 (kk2 * 32 + g * 8) * 2);
            acc_dq[hf] = mfma16x16x32_bf16(da, bk, acc_dq[hf]);
          }
        }
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int qrow = qb + wid * 16 + g * 4 + r;
          if (qrow >= T) continue;
#pragma unroll
          for (int hf = 0; hf < HF; ++hf) {
            if (acc_dq[hf][r] != 0.f) {
              atomicAdd(
                  dq_acc + (((long)b * T + qrow) * N + n) * H + hf * 16 +
                      cl,
                  acc_dq[hf][r]);
            }
          }
        }
      } else if (BIAS_GRAD && wid >= QTB / 16) {
        // Bias grad via per-diagonal sums over the dS tile in ds_lds
        // (entries hold dlogits*scale; invisible positions are exact
        // zeros). One lane per diagonal d = qcol - key: elements
        // (qrow, key) with qb + qrow - kbase - key == d. Runs on the
        // waves otherwise idle during the dQ phase.
        const int ndiag = QTB + KTB - 1;
        const int lane_global = (wid - QTB / 16) * WAVE_SIZE + lane;
        const int nworkers = (NWB - QTB / 16) * WAVE_SIZE;
        const float inv_scale = 1.f / scale;
        for (int di = lane_global; di < ndiag; di += nworkers) {
          // diagonal offset within the tile: qrow - key = di - (KTB-1)
          const int doff = di - (KTB - 1);
          float sum = 0.f;
          const int q_lo = max(0, doff);
          const int q_hi = min(QTB - 1, KTB - 1 + doff);
          for (int qrow = q_lo; qrow <= q_hi; ++qrow) {
            const int key = qrow - doff;
            sum += bf16_bits_to_float(*reinterpret_cast<unsigned short*>(
                ds_lds + qrow * (KTB * 2) + swz(qrow, key * 2)));
          }
          if (sum != 0.f) {
            int d = (qb + doff) - kbase;  // global q - k distance
            d = d < -bias_clip ? -bias_clip
                               : (d > bias_clip ? bias_clip : d);
            atomicAdd(&dbias_s[d + bias_clip], sum * inv_scale);
          }
        }
      }
    }

    // Flush dbias for this head.
    if (BIAS_GRAD) {
      __syncthreads();
      for (int i = threadIdx.x; i < nbias; i += NWB * WAVE_SIZE) {
        if (dbias_s[i] != 0.f)
          atomicAdd(dbias + (long)n * nbias + i, dbias_s[i]);
      }
    }

    // dK/dV for this head's contribution: with GQA we accumulate across
    // heads; write after the head loop. To keep single-write semantics we
    // only write on the last head, so keep accumulating into global via
    // registers: here we add to global bf16 once per head via fp32 staging
    // in registers across heads instead. Simpler: accumulate across heads
    // in registers by NOT resetting acc_dk/acc_dv — but they are reset per
    // head above; so write-accumulate here:
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int key = k0 + g * 4 + r;
      if (key >= S) continue;  // acc is 0 for klen<=key<S: writes zeros
      long off = (((long)b * S + key) * NKV + nkv) * H;
#pragma unroll
      for (int hf = 0; hf < HF; ++hf) {
        if (group == 1) {
          dk[off + hf * 16 + cl] = float_to_bf16_bits(acc_dk[hf][r]);
          dv[off + hf * 16 + cl] = float_to_bf16_bits(acc_dv[hf][r]);
        } else {
          // GQA: multiple heads accumulate; use fp32 atomics into dq_acc?
          // dk/dv are bf16 outputs; for group>1 the host passes fp32
          // buffers aliased via dk/dv pointers being fp32. Handled on the
          // host by allocating fp32 and a separate cast. Here: atomicAdd.
          atomicAdd(reinterpret_cast<float*>(dk) + off + hf * 16 + cl,
                    acc_dk[hf][r]);
          atomicAdd(reinterpret_cast<float*>(dv) + off + hf * 16 + cl,
                    acc_dv[hf][r]);
        }
      }
    }
    __syncthreads();
  }
}

}  // namespace

// ---------------------------------------------------------------------------
// Host wrappers
// ---------------------------------------------------------------------------
static void check_btnh(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda() && t.is_contiguous() && t.dim() == 4 &&
                  t.scalar_type() == torch::kBFloat16,
              name, " must be contiguous bf16 [B,T,N,H]");
}

std::vector<torch::Tensor> fa_fwd(torch::Tensor q, torch::Tensor k,
                                  torch::Tensor v,
                                  c10::optional<torch::Tensor> klen,
                                  c10::optional<torch::Tensor> bias,
                                  c10::optional<torch::Tensor> qseg,
                                  c10::optional<torch::Tensor> kseg,
                                  int64_t win_l, int64_t win_r,
                                  int64_t bias_clip, double scale,
                                  int64_t chunk_size, int64_t left_chunks) {
  check_btnh(q, "q");
  check_btnh(k, "k");
  check_btnh(v, "v");
  const int B = q.size(0), T = q.size(1), N = q.size(2), H = q.size(3);
  const int S = k.size(1), NKV = k.size(2);
  TORCH_CHECK(H == 64 || H == 128, "H must be 64 or 128, got ", H);
  TORCH_CHECK(N % NKV == 0, "GQA requires N % NKV == 0");
  auto o = torch::empty_like(q);
  auto lse = torch::empty({B, N, T}, q.options().dtype(torch::kFloat32));
  auto stream = at::cuda::getCurrentCUDAStream();
  dim3 grid((T + FQT - 1) / FQT, N, B);
  const int* klp = klen.has_value() ? klen->data_ptr<int>() : nullptr;
  const unsigned short* bp =
      bias.has_value() ? (const unsigned short*)bias->data_ptr() : nullptr;
  const int* qsp = qseg.has_value() ? qseg->data_ptr<int>() : nullptr;
  const int* ksp = kseg.has_value() ? kseg->data_ptr<int>() : nullptr;
  // Big KV tiles amortize staging barriers, but waste work on short
  // sequences (masked overhang): pick by S.
  const int ktf = S >= 512 ? 128 : 64;
  size_t shmem =
      (size_t)ktf * H * 2 + (size_t)H * ktf * 2 + FWD_NW * 16 * ktf * 2;
#define FA_FWD(HH)                                                          \
  if (ktf == 128)                                                           \
    hipLaunchKernelGGL((fa_fwd_kernel<HH, 128, false>), grid,               \
                       dim3(FWD_BLOCK),                                     \
                       shmem, stream,                                       \
                     (const unsigned short*)q.data_ptr(),                   \
                     (const unsigned short*)k.data_ptr(),                   \
                     (const unsigned short*)v.data_ptr(), klp, bp, qsp,     \
                     ksp, (unsigned short*)o.data_ptr(),                    \
                     lse.data_ptr<float>(), B, T, S, N, NKV, (int)win_l,    \
                     (int)win_r, (int)bias_clip, (float)scale, (int)chunk_size, (int)left_chunks)
#define FA_FWD64(HH)                                                        \
  hipLaunchKernelGGL((fa_fwd_kernel<HH, 64, false>), grid,                  \
                     dim3(FWD_BLOCK),                                       \
                     shmem, stream, (const unsigned short*)q.data_ptr(),    \
                     (const unsigned short*)k.data_ptr(),                   \
                     (const unsigned short*)v.data_ptr(), klp, bp, qsp,     \
                     ksp, (unsigned short*)o.data_ptr(),                    \
                     lse.data_ptr<float>(), B, T, S, N, NKV, (int)win_l,    \
                     (int)win_r, (int)bias_clip, (float)scale, (int)chunk_size, (int)left_chunks)
#define FA_FWD_SEG(HH)                                                      \
  if (ktf == 128)                                                           \
    hipLaunchKernelGGL((fa_fwd_kernel<HH, 128, true>), grid,                \
                       dim3(FWD_BLOCK), shmem, stream,                      \
                       (const unsigned short*)q.data_ptr(),                 \
                       (const unsigned short*)k.data_ptr(),                 \
                       (const unsigned short*)v.data_ptr(), klp, bp, qsp,   \
                       ksp, (unsigned short*)o.data_ptr(),                  \
                       lse.data_ptr<float>(), B, T, S, N, NKV, (int)win_l,  \
                       (int)win_r, (int)bias_clip, (float)scale, (int)chunk_size, (int)left_chunks);           \
  else                                                                      \
    hipLaunchKernelGGL((fa_fwd_kernel<HH, 64, true>), grid,                 \
                       dim3(FWD_BLOCK), shmem, stream,                      \
                       (const unsigned short*)q.data_ptr(),                 \
                       (const unsigned short*)k.data_ptr(),                 \
                       (const unsigned short*)v.data_ptr(), klp, bp, qsp,   \
                       ksp, (unsigned short*)o.data_ptr(),                  \
                       lse.data_ptr<float>(), B, T, S, N, NKV, (int)win_l,  \
                       (int)win_r, (int)bias_clip, (float)scale, (int)chunk_size, (int)left_chunks)
  if (qsp != nullptr) {
    if (H == 64) {
      FA_FWD_SEG(64);
    } else {
      FA_FWD_SEG(128);
    }
  } else if (H == 64) {
    FA_FWD(64);
    else FA_FWD64(64);
  } else {
    FA_FWD(128);
    else FA_FWD64(128);
  }
#undef FA_FWD
#undef FA_FWD64
#undef FA_FWD_SEG
  return {o, lse};
}

std::vector<torch::Tensor> fa_bwd(torch::Tensor dout, torch::Tensor q,
                                  torch::Tensor k, torch::Tensor v,
                                  torch::Tensor o, torch::Tensor lse,
                                  c10::optional<torch::Tensor> klen,
                                  c10::optional<torch::Tensor> bias,
                                  c10::optional<torch::Tensor> qseg,
                                  c10::optional<torch::Tensor> kseg,
                                  bool bias_grad, int64_t win_l,
                                  int64_t win_r, int64_t bias_clip,
                                  double scale, int64_t chunk_size,
                                  int64_t left_chunks) {
  check_btnh(q, "q");
  const int B = q.size(0), T = q.size(1), N = q.size(2), H = q.size(3);
  const int S = k.size(1), NKV = k.size(2);
  const int group = N / NKV;
  auto stream = at::cuda::getCurrentCUDAStream();

  auto delta = torch::empty({B, N, T}, q.options().dtype(torch::kFloat32));
  {
    long rows = (long)B * T * N;
    dim3 grid((rows + NW - 1) / NW);
    if (H == 64)
      hipLaunchKernelGGL((fa_bwd_delta<64>), grid, dim3(BLOCK), 0, stream,
                         (const unsigned short*)dout.data_ptr(),
                         (const unsigned short*)o.data_ptr(),
                         delta.data_ptr<float>(), B, T, N);
    else
      hipLaunchKernelGGL((fa_bwd_delta<128>), grid, dim3(BLOCK), 0, stream,
                         (const unsigned short*)dout.data_ptr(),
                         (const unsigned short*)o.data_ptr(),
                         delta.data_ptr<float>(), B, T, N);
  }

  auto dq_acc = torch::zeros_like(q, q.options().dtype(torch::kFloat32));
  torch::Tensor dk_t, dv_t;
  if (group == 1) {
    dk_t = torch::empty_like(k);
    dv_t = torch::empty_like(v);
  } else {
    dk_t = torch::zeros_like(k, k.options().dtype(torch::kFloat32));
    dv_t = torch::zeros_like(v, v.options().dtype(torch::kFloat32));
  }
  const int nbias = 2 * (int)bias_clip + 1;
  torch::Tensor dbias_t = torch::zeros(
      {bias.has_value() ? N : 0, bias.has_value() ? nbias : 0},
      q.options().dtype(torch::kFloat32));

  const int* klp = klen.has_value() ? klen->data_ptr<int>() : nullptr;
  const unsigned short* bp =
      bias.has_value() ? (const unsigned short*)bias->data_ptr() : nullptr;
  const int* qsp = qseg.has_value() ? qseg->data_ptr<int>() : nullptr;
  const int* ksp = kseg.has_value() ? kseg->data_ptr<int>() : nullptr;
  bool bg = bias.has_value() && bias_grad;

  // Phase-skip bitmask for perf attribution only (results invalid when
  // nonzero): 1 = skip dQ math+atomics, 2 = skip transposed q/do
  // staging, 4 = skip dS-full store.
  static int skip_phases = []() {
    const char* e = getenv("LINGVO_FA_BWD_SKIP");
    return e ? atoi(e) : 0;
  }();
  // 128-key tiles with 8 waves (q-tile staging amortized 2x). H=128
  // uses 144KB LDS -> 1 block/CU; staging savings outweigh occupancy.
  const int ktb = 128;
  const int nwb = 8;
  // q-tile 64. A 32-row variant (2 blocks/CU via ~64 KB LDS) measured
  // SLOWER (bwd 0.73 -> 0.81 ms at the bench shape): doubled staging
  // sync passes outweigh the occupancy gain. QTB stays a template knob.
  const int qtb = 64;
  size_t shmem = (size_t)qtb * H * 2 * 2     // q_lds + do_lds
                 + (size_t)H * 128 * 2       // qt_lds + dot_lds (128B pitch)
                 + (size_t)H * ktb * 2       // kt_lds
                 + (size_t)qtb * ktb * 2     // ds_lds
                 + (size_t)nwb * 16 * 128    // a_lds (128B pitch)
                 + 2 * qtb * sizeof(float) + (bg ? nbias * sizeof(float) : 0);
  dim3 grid((S + ktb - 1) / ktb, NKV, B);
#define FA_BWD(HH, BG)                                                       \
  hipLaunchKernelGGL(                                                        \
      (fa_bwd_kernel<HH, BG, 128, 8, 64>), grid, dim3(8 * WAVE_SIZE), shmem, \
      stream,                                                                \
      (const unsigned short*)dout.data_ptr(),                                \
      (const unsigned short*)q.data_ptr(),                                   \
      (const unsigned short*)k.data_ptr(),                                   \
      (const unsigned short*)v.data_ptr(), lse.data_ptr<float>(),            \
      delta.data_ptr<float>(), klp, bp, qsp, ksp,                            \
      dq_acc.data_ptr<float>(),                                              \
      (unsigned short*)dk_t.data_ptr(), (unsigned short*)dv_t.data_ptr(),    \
      dbias_t.numel() ? dbias_t.data_ptr<float>() : nullptr, B, T, S, N,     \
      NKV, (int)win_l, (int)win_r, (int)bias_clip, (float)scale, skip_phases, (int)chunk_size, (int)left_chunks)
  if (H == 64) {
    if (bg) FA_BWD(64, true); else FA_BWD(64, false);
  } else {
    if (bg) FA_BWD(128, true); else FA_BWD(128, false);
  }
#undef FA_BWD

  auto dq = dq_acc.to(torch::kBFloat16);
  auto dk = group == 1 ? dk_t : dk_t.to(torch::kBFloat16);
  auto dv = group == 1 ? dv_t : dv_t.to(torch::kBFloat16);
  return {dq, dk, dv, dbias_t};
}
