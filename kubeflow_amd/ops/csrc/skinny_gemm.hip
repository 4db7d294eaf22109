// skinny_gemm.hip — decode-batch linear: C[M,N] = A[M,K] @ W[N,K]^T for
// M <= 16 (the serving decode step's GEMV-shaped GEMMs).
//
// hipBLASLt runs these weight-streaming shapes at ~30-50% of HBM BW at
// M=16 (profiles/r02: decode-step GEMMs dominate the 6.3 ms step whose
// weight-read floor is ~2.5 ms). This kernel streams W exactly once at
// full coalescing: block = 4 waves SHARING one 16-column N-tile with an
// in-block K-split, each wave issuing mfma_f32_16x16x32_bf16 over its K
// range (A operand rows = the M batch rows, zero-padded to 16), then a
// 4-way LDS reduction. Lanes {l, l+16, l+32, l+48} read consecutive
// 8-element chunks of the same W row, so each wave instruction covers
// contiguous 64 B per row — full line utilization while W streams.
//
// Grid = N/16 blocks; K % 32 == 0; bf16 in/out, fp32 accumulate.

#include "kf_common.h"

typedef __bf16 kf_bf16x8s __attribute__((ext_vector_type(8)));
typedef float kf_f32x4s __attribute__((ext_vector_type(4)));

// Fused-RMSNorm support: when rmsg != null, A rows are normalized
// in-flight — the block computes each A row's rstd once (it reads the
// whole row over the chunk loop anyway; A is LLC-hot) and scales
// fragments by rstd[row] * gamma[k]. Removes the two standalone rmsnorm
// launches per decode layer (profiles/r02_decode_anatomy.md).
typedef unsigned short kf_u16x8q __attribute__((ext_vector_type(8)));

__device__ __forceinline__ float kf_row_rstd(const unsigned short* row,
                                             int64_t K, float eps) {
  // all 64 lanes of the calling wave stride one row cooperatively
  float acc = 0.f;
  const int lane = threadIdx.x & (KF_WAVE - 1);
  for (int64_t i = (int64_t)lane * 8; i < K; i += KF_WAVE * 8) {
    const kf_u16x8q x = *reinterpret_cast<const kf_u16x8q*>(row + i);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float f = kf_bf16_to_f32(x[j]);
      acc += f * f;
    }
  }
  for (int off = 32; off; off >>= 1) acc += __shfl_xor(acc, off, KF_WAVE);
  return rsqrtf(acc / (float)K + eps);
}

// Fused-SwiGLU support (the w2 projection): when swiglu != 0, A is the
// RAW w13 output [M, 2K] (gate rows [0,K), up rows [K,2K)) and fragments
// are silu(gate)*up computed in-flight — removes the standalone swiglu
// kernel + the y round-trip from the decode layer. Pure elementwise on
// the A side (no extra row pass, unlike the rms fusion which measured
// negative).
__device__ __forceinline__ kf_bf16x8s kf_swiglu8(kf_bf16x8s g,
                                                 kf_bf16x8s u) {
  union {
    kf_bf16x8s v;
    unsigned short us[8];
  } gi, ui, o;
  gi.v = g;
  ui.v = u;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const float gf = kf_bf16_to_f32(gi.us[j]);
    const float uf = kf_bf16_to_f32(ui.us[j]);
    o.us[j] = kf_f32_to_bf16(gf / (1.f + __expf(-gf)) * uf);
  }
  return o.v;
}

__device__ __forceinline__ kf_bf16x8s kf_rms_scale8(
    kf_bf16x8s a, float rstd, const float* __restrict__ g) {
  union {
    kf_bf16x8s v;
    unsigned short u[8];
  } in, out;
  in.v = a;
#pragma unroll
  for (int j = 0; j < 8; ++j)
    out.u[j] = kf_f32_to_bf16(kf_bf16_to_f32(in.u[j]) * rstd * g[j]);
  return out.v;
}

#define SK_NT 16      // N columns per block
// K-split ways (waves/block) is a template knob: small-N shapes (wo:
// N=4096 -> 256 blocks == 1 block/CU == 1 wave/SIMD at 4 waves) are
// occupancy-starved and want 8; profiled per shape in
// profiles/r02_skinny_gemm.md.

template <int SKW>
__global__ __launch_bounds__(SKW * 64, 2) void kf_skinny_gemm_kernel(
    unsigned short* __restrict__ c, const unsigned short* __restrict__ a,
    const unsigned short* __restrict__ w,
    const unsigned short* __restrict__ res, int M, int64_t N, int64_t K,
    int64_t lda, int64_t ldw, int64_t ldc) {
  __shared__ float red[SKW][SK_NT][SK_NT];  // per-wave C tiles

  const int64_t n0 = (int64_t)blockIdx.x * SK_NT;
  const int tid = threadIdx.x;
  const int wv = tid / KF_WAVE;
  const int lane = tid & (KF_WAVE - 1);
  const int l15 = lane & 15;
  const int hi4 = lane >> 4;  // 0..3: k-subchunk within the 32-k step

  // this wave's K range (each wave strides by SK_WAVES*32 for coalesced
  // row-chunk progression shared with its sibling lanes)
  kf_f32x4s acc = kf_f32x4s{0.f, 0.f, 0.f, 0.f};
  const unsigned short* wrow = w + (n0 + l15) * ldw;
  const bool arow_ok = l15 < M;
  const unsigned short* arow = a + (arow_ok ? l15 : 0) * lda;
  const kf_bf16x8s zero8 = kf_bf16x8s{0, 0, 0, 0, 0, 0, 0, 0};
  // 8-deep unrolled stream: ~256 B of W per wave in flight so the HBM
  // latency pipelines (a single outstanding load-pair left the wave
  // latency-bound at ~45% of blaslt on the big shapes)
  const int64_t step = (int64_t)SKW * 32;
  int64_t k = (int64_t)wv * 32;
  for (; k + 7 * step + 32 <= K; k += 8 * step) {
    kf_bf16x8s afs[8], wfs[8];
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      const int64_t ku = k + u * step + hi4 * 8;
      afs[u] = arow_ok
          ? *reinterpret_cast<const kf_bf16x8s*>(arow + ku) : zero8;
      wfs[u] = *reinterpret_cast<const kf_bf16x8s*>(wrow + ku);
    }
#pragma unroll
    for (int u = 0; u < 8; ++u)
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afs[u], wfs[u], acc,
                                                    0, 0, 0);
  }
  for (; k < K; k += step) {
    kf_bf16x8s af = arow_ok
        ? *reinterpret_cast<const kf_bf16x8s*>(arow + k + hi4 * 8)
        : zero8;
    kf_bf16x8s wf =
        *reinterpret_cast<const kf_bf16x8s*>(wrow + k + hi4 * 8);
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, wf, acc, 0, 0, 0);
  }
  // C/D layout (guide §3, m89): col = lane&15, row = (lane>>4)*4 + j
#pragma unroll
  for (int j = 0; j < 4; ++j) red[wv][hi4 * 4 + j][l15] = acc[j];
  __syncthreads();
  // first waves reduce + write: thread (row, col) pairs
  if (tid < SK_NT * SK_NT) {
    const int row = tid / SK_NT, col = tid % SK_NT;
    if (row < M && n0 + col < N) {
      float s = 0.f;
#pragma unroll
      for (int ww = 0; ww < SKW; ++ww) s += red[ww][row][col];
      if (res) s += kf_bf16_to_f32(res[row * ldc + n0 + col]);
      c[row * ldc + n0 + col] = kf_f32_to_bf16(s);
    }
  }
}

// ---------------------------------------------------------------------
// LDS-staged variant. The direct kernel's W loads put 16 rows x 64 B per
// instruction on the wire — half of every 128 B line — capping it at
// ~3 TB/s. Here the block's 8 waves stage W chunks cooperatively with
// FULL-ROW 1 KB runs (wave wv loads rows {2wv, 2wv+1}, one b128 per lane
// per row), then read their MFMA fragments from LDS. Padded row stride
// (520 elems) keeps both LDS sides at worst 2-way bank-conflicted.
// Double-buffered: one __syncthreads per 512-k chunk; the next chunk's
// global loads issue before the barrier so HBM latency overlaps MFMA.

#define SKL_W 8
#define SKL_KC 512                      // staged K elems (1 KB per row)
#define SKL_STRIDE (SKL_KC + 8)        // +8 elems: 2-way max conflicts

// MT = M-tile (16 or 32 batch rows); W traffic is identical, the wider
// tile just adds a second A fragment + accumulator (decode buckets > 16
// otherwise fell back to hipBLASLt).
template <int MT, bool SWIGLU>
__global__ __launch_bounds__(SKL_W * 64, 2) void kf_skinny_lds_kernel(
    unsigned short* __restrict__ c, const unsigned short* __restrict__ a,
    const unsigned short* __restrict__ w,
    const unsigned short* __restrict__ res,
    const float* __restrict__ rmsg, float rms_eps, int M,
    int64_t N, int64_t K, int64_t lda, int64_t ldw, int64_t ldc) {
  __shared__ unsigned short wbuf[2][SK_NT][SKL_STRIDE];
  __shared__ float red[SKL_W][MT][SK_NT];
  __shared__ float rstd_lds[MT];

  const int64_t n0 = (int64_t)blockIdx.x * SK_NT;
  const int tid = threadIdx.x;
  const int wv = tid / KF_WAVE;
  const int lane = tid & (KF_WAVE - 1);
  const int l15 = lane & 15;
  const int hi4 = lane >> 4;

  const int NMT = MT / 16;  // A tiles (1 or 2)
  kf_f32x4s acc[NMT];
  bool arow_ok[NMT];
  const unsigned short* arow[NMT];
#pragma unroll
  for (int t = 0; t < NMT; ++t) {
    acc[t] = kf_f32x4s{0.f, 0.f, 0.f, 0.f};
    arow_ok[t] = l15 + 16 * t < M;
    arow[t] = a + (arow_ok[t] ? l15 + 16 * t : 0) * lda;
  }
  const kf_bf16x8s zero8 = kf_bf16x8s{0, 0, 0, 0, 0, 0, 0, 0};
  float rstd[NMT];
  if (rmsg) {
    // wave wv computes rstd for rows {wv*MT/8 .. } (MT/8 rows per wave)
    for (int r = wv * (MT / SKL_W); r < (wv + 1) * (MT / SKL_W); ++r) {
      const float v = kf_row_rstd(a + (r < M ? r : 0) * lda, K, rms_eps);
      if (lane == 0) rstd_lds[r] = v;
    }
    __syncthreads();
#pragma unroll
    for (int t = 0; t < NMT; ++t) rstd[t] = rstd_lds[l15 + 16 * t];
  }
  // writer: wave wv stages rows {2wv, 2wv+1}, lane covers elems
  // [lane*8, lane*8+8) of each 512-elem row slice
  const unsigned short* wr0 = w + (n0 + 2 * wv) * ldw + lane * 8;
  const unsigned short* wr1 = wr0 + ldw;

  const int64_t nch = K / SKL_KC;
  const int ke0 = wv * 64 + hi4 * 8;  // this wave's k slice (s=0; s=1 at +32)
  kf_bf16x8s st0 = *reinterpret_cast<const kf_bf16x8s*>(wr0);
  kf_bf16x8s st1 = *reinterpret_cast<const kf_bf16x8s*>(wr1);
  kf_bf16x8s af0[NMT], af1[NMT];
  kf_bf16x8s uf0[SWIGLU ? NMT : 1], uf1[SWIGLU ? NMT : 1];
#pragma unroll
  for (int t = 0; t < NMT; ++t) {
    af0[t] = arow_ok[t]
        ? *reinterpret_cast<const kf_bf16x8s*>(arow[t] + ke0) : zero8;
    af1[t] = arow_ok[t]
        ? *reinterpret_cast<const kf_bf16x8s*>(arow[t] + ke0 + 32) : zero8;
    if constexpr (SWIGLU) {
      uf0[t] = arow_ok[t]
          ? *reinterpret_cast<const kf_bf16x8s*>(arow[t] + K + ke0)
          : zero8;
      uf1[t] = arow_ok[t]
          ? *reinterpret_cast<const kf_bf16x8s*>(arow[t] + K + ke0 + 32)
          : zero8;
    }
  }
  *reinterpret_cast<kf_bf16x8s*>(&wbuf[0][2 * wv][lane * 8]) = st0;
  *reinterpret_cast<kf_bf16x8s*>(&wbuf[0][2 * wv + 1][lane * 8]) = st1;
  for (int64_t ch = 0; ch < nch; ++ch) {
    kf_bf16x8s a0[NMT], a1[NMT];
#pragma unroll
    for (int t = 0; t < NMT; ++t) {
      if constexpr (SWIGLU) {
        a0[t] = kf_swiglu8(af0[t], uf0[t]);
        a1[t] = kf_swiglu8(af1[t], uf1[t]);
      } else {
        a0[t] = af0[t];
        a1[t] = af1[t];
      }
    }
    if (ch + 1 < nch) {
      st0 = *reinterpret_cast<const kf_bf16x8s*>(wr0 + (ch + 1) * SKL_KC);
      st1 = *reinterpret_cast<const kf_bf16x8s*>(wr1 + (ch + 1) * SKL_KC);
#pragma unroll
      for (int t = 0; t < NMT; ++t)
        if (arow_ok[t]) {
          af0[t] = *reinterpret_cast<const kf_bf16x8s*>(
              arow[t] + (ch + 1) * SKL_KC + ke0);
          af1[t] = *reinterpret_cast<const kf_bf16x8s*>(
              arow[t] + (ch + 1) * SKL_KC + ke0 + 32);
          if constexpr (SWIGLU) {
            uf0[t] = *reinterpret_cast<const kf_bf16x8s*>(
                arow[t] + K + (ch + 1) * SKL_KC + ke0);
            uf1[t] = *reinterpret_cast<const kf_bf16x8s*>(
                arow[t] + K + (ch + 1) * SKL_KC + ke0 + 32);
          }
        }
    }
    // one barrier per chunk: makes buffer ch&1's writes visible AND
    // guarantees last iteration's readers of buffer (ch+1)&1 are done
    __syncthreads();
    kf_bf16x8s wf0 =
        *reinterpret_cast<const kf_bf16x8s*>(&wbuf[ch & 1][l15][ke0]);
    kf_bf16x8s wf1 =
        *reinterpret_cast<const kf_bf16x8s*>(&wbuf[ch & 1][l15][ke0 + 32]);
    if (rmsg) {
      const float* g0 = rmsg + ch * SKL_KC + ke0;
#pragma unroll
      for (int t = 0; t < NMT; ++t) {
        a0[t] = kf_rms_scale8(a0[t], rstd[t], g0);
        a1[t] = kf_rms_scale8(a1[t], rstd[t], g0 + 32);
      }
    }
#pragma unroll
    for (int t = 0; t < NMT; ++t) {
      acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0[t], wf0, acc[t],
                                                       0, 0, 0);
      acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1[t], wf1, acc[t],
                                                       0, 0, 0);
    }
    if (ch + 1 < nch) {
      const int b = (int)((ch + 1) & 1);
      *reinterpret_cast<kf_bf16x8s*>(&wbuf[b][2 * wv][lane * 8]) = st0;
      *reinterpret_cast<kf_bf16x8s*>(&wbuf[b][2 * wv + 1][lane * 8]) = st1;
    }
  }
#pragma unroll
  for (int t = 0; t < NMT; ++t)
#pragma unroll
    for (int j = 0; j < 4; ++j)
      red[wv][16 * t + hi4 * 4 + j][l15] = acc[t][j];
  __syncthreads();
  if (tid < MT * SK_NT) {
    const int row = tid / SK_NT, col = tid % SK_NT;
    if (row < M && n0 + col < N) {
      float s = 0.f;
#pragma unroll
      for (int ww = 0; ww < SKL_W; ++ww) s += red[ww][row][col];
      if (res) s += kf_bf16_to_f32(res[row * ldc + n0 + col]);
      c[row * ldc + n0 + col] = kf_f32_to_bf16(s);
    }
  }
}

// res (nullable, bf16, C layout): fused residual epilogue C = res + A@W^T
// — removes the separate elementwise add after the wo / w2 projections.
// ---------------------------------------------------------------------
// W8A16 quantized variant: W stored as OCP e4m3 fp8 with one fp32 scale
// per output row (absmax/448). The decode step is weight-BW-bound, so
// halving W bytes ≈ halves GEMM time; fragments convert fp8 -> f32 ->
// bf16 in-register (v_cvt_pk_f32_fp8 + v_cvt_pk_bf16_f32 — exact, since
// e4m3's 3 mantissa bits embed in bf16's 8) and MFMA stays bf16, so the
// activations keep full precision. Same LDS staging/shape rules as the
// bf16 kernel: K % 512 == 0, N % 16 == 0, M <= 32.

#define SKQ_KC 1024               // fp8 K elems per staged chunk (1 KB/row)
#define SKQ_STRIDE (SKQ_KC + 16)   // fp8 row slice bytes + pad

__device__ __forceinline__ kf_bf16x8s kf_fp8x8_to_bf16x8(
    const unsigned char* p8) {
  const unsigned int* pp = reinterpret_cast<const unsigned int*>(p8);
  const unsigned int lo = pp[0], hi = pp[1];
  typedef float kf_f32x2q __attribute__((ext_vector_type(2)));
  kf_f32x2q f0 = __builtin_amdgcn_cvt_pk_f32_fp8(lo, false);
  kf_f32x2q f1 = __builtin_amdgcn_cvt_pk_f32_fp8(lo, true);
  kf_f32x2q f2 = __builtin_amdgcn_cvt_pk_f32_fp8(hi, false);
  kf_f32x2q f3 = __builtin_amdgcn_cvt_pk_f32_fp8(hi, true);
  unsigned int r0, r1, r2, r3;
  asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r0) : "v"(f0[0]), "v"(f0[1]));
  asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r1) : "v"(f1[0]), "v"(f1[1]));
  asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r2) : "v"(f2[0]), "v"(f2[1]));
  asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r3) : "v"(f3[0]), "v"(f3[1]));
  union {
    unsigned int u[4];
    kf_bf16x8s v;
  } out;
  out.u[0] = r0;
  out.u[1] = r1;
  out.u[2] = r2;
  out.u[3] = r3;
  return out.v;
}

template <int MT, bool SWIGLU>
__global__ __launch_bounds__(SKL_W * 64, 2) void kf_skinny_q8_kernel(
    unsigned short* __restrict__ c, const unsigned short* __restrict__ a,
    const unsigned char* __restrict__ w8, const float* __restrict__ wscale,
    const unsigned short* __restrict__ res,
    const float* __restrict__ rmsg, float rms_eps, int M,
    int64_t N, int64_t K, int64_t lda, int64_t ldw, int64_t ldc) {
  __shared__ unsigned char wbuf8[2][SK_NT][SKQ_STRIDE];
  __shared__ float red[SKL_W][MT][SK_NT];
  __shared__ float rstd_lds[MT];

  const int64_t n0 = (int64_t)blockIdx.x * SK_NT;
  const int tid = threadIdx.x;
  const int wv = tid / KF_WAVE;
  const int lane = tid & (KF_WAVE - 1);
  const int l15 = lane & 15;
  const int hi4 = lane >> 4;

  const int NMT = MT / 16;
  kf_f32x4s acc[NMT];
  bool arow_ok[NMT];
  const unsigned short* arow[NMT];
#pragma unroll
  for (int t = 0; t < NMT; ++t) {
    acc[t] = kf_f32x4s{0.f, 0.f, 0.f, 0.f};
    arow_ok[t] = l15 + 16 * t < M;
    arow[t] = a + (arow_ok[t] ? l15 + 16 * t : 0) * lda;
  }
  const kf_bf16x8s zero8 = kf_bf16x8s{0, 0, 0, 0, 0, 0, 0, 0};
  float rstd[NMT];
  if (rmsg) {
    for (int r = wv * (MT / SKL_W); r < (wv + 1) * (MT / SKL_W); ++r) {
      const float vv = kf_row_rstd(a + (r < M ? r : 0) * lda, K, rms_eps);
      if (lane == 0) rstd_lds[r] = vv;
    }
    __syncthreads();
#pragma unroll
    for (int t = 0; t < NMT; ++t) rstd[t] = rstd_lds[l15 + 16 * t];
  }
  // 16 B per lane per row: a full 1 KB (1024 fp8) row slice per wave
  // instruction — same load width as the bf16 kernel, twice the K
  typedef unsigned int kf_u32x4q __attribute__((ext_vector_type(4)));
  const unsigned char* wr0 = w8 + (n0 + 2 * wv) * ldw + lane * 16;
  const unsigned char* wr1 = wr0 + ldw;

  const int64_t nch = K / SKQ_KC;
  const int ke0 = wv * 128 + hi4 * 8;  // k slices at +0,+32,+64,+96
  kf_u32x4q st0 = *reinterpret_cast<const kf_u32x4q*>(wr0);
  kf_u32x4q st1 = *reinterpret_cast<const kf_u32x4q*>(wr1);
  kf_bf16x8s af[4][NMT];
  kf_bf16x8s uf[4][SWIGLU ? NMT : 1];
#pragma unroll
  for (int si = 0; si < 4; ++si)
#pragma unroll
    for (int t = 0; t < NMT; ++t) {
      af[si][t] = arow_ok[t]
          ? *reinterpret_cast<const kf_bf16x8s*>(arow[t] + ke0 + 32 * si)
          : zero8;
      if constexpr (SWIGLU)
        uf[si][t] = arow_ok[t]
            ? *reinterpret_cast<const kf_bf16x8s*>(
                  arow[t] + K + ke0 + 32 * si)
            : zero8;
    }
  *reinterpret_cast<kf_u32x4q*>(&wbuf8[0][2 * wv][lane * 16]) = st0;
  *reinterpret_cast<kf_u32x4q*>(&wbuf8[0][2 * wv + 1][lane * 16]) = st1;
  for (int64_t ch = 0; ch < nch; ++ch) {
    kf_bf16x8s acur[4][NMT];
#pragma unroll
    for (int si = 0; si < 4; ++si)
#pragma unroll
      for (int t = 0; t < NMT; ++t) {
        if constexpr (SWIGLU)
          acur[si][t] = kf_swiglu8(af[si][t], uf[si][t]);
        else
          acur[si][t] = af[si][t];
      }
    if (ch + 1 < nch) {
      st0 = *reinterpret_cast<const kf_u32x4q*>(wr0 + (ch + 1) * SKQ_KC);
      st1 = *reinterpret_cast<const kf_u32x4q*>(wr1 + (ch + 1) * SKQ_KC);
#pragma unroll
      for (int si = 0; si < 4; ++si)
#pragma unroll
        for (int t = 0; t < NMT; ++t)
          if (arow_ok[t]) {
            af[si][t] = *reinterpret_cast<const kf_bf16x8s*>(
                arow[t] + (ch + 1) * SKQ_KC + ke0 + 32 * si);
            if constexpr (SWIGLU)
              uf[si][t] = *reinterpret_cast<const kf_bf16x8s*>(
                  arow[t] + K + (ch + 1) * SKQ_KC + ke0 + 32 * si);
          }
    }
    __syncthreads();
#pragma unroll
    for (int si = 0; si < 4; ++si) {
      kf_bf16x8s wf =
          kf_fp8x8_to_bf16x8(&wbuf8[ch & 1][l15][ke0 + 32 * si]);
      if (rmsg) {
        const float* g = rmsg + ch * SKQ_KC + ke0 + 32 * si;
#pragma unroll
        for (int t = 0; t < NMT; ++t)
          acur[si][t] = kf_rms_scale8(acur[si][t], rstd[t], g);
      }
#pragma unroll
      for (int t = 0; t < NMT; ++t)
        acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(acur[si][t], wf,
                                                         acc[t], 0, 0, 0);
    }
    if (ch + 1 < nch) {
      const int b = (int)((ch + 1) & 1);
      *reinterpret_cast<kf_u32x4q*>(&wbuf8[b][2 * wv][lane * 16]) = st0;
      *reinterpret_cast<kf_u32x4q*>(&wbuf8[b][2 * wv + 1][lane * 16]) =
          st1;
    }
  }
#pragma unroll
  for (int t = 0; t < NMT; ++t)
#pragma unroll
    for (int j = 0; j < 4; ++j)
      red[wv][16 * t + hi4 * 4 + j][l15] = acc[t][j];
  __syncthreads();
  if (tid < MT * SK_NT) {
    const int row = tid / SK_NT, col = tid % SK_NT;
    if (row < M && n0 + col < N) {
      float s = 0.f;
#pragma unroll
      for (int ww = 0; ww < SKL_W; ++ww) s += red[ww][row][col];
      s *= wscale[n0 + col];
      if (res) s += kf_bf16_to_f32(res[row * ldc + n0 + col]);
      c[row * ldc + n0 + col] = kf_f32_to_bf16(s);
    }
  }
}

KF_EXPORT int kf_skinny_gemm_q8(void* c, const void* a, const void* w8,
                                const float* wscale, const void* res,
                                const float* rmsg, float rms_eps,
                                int64_t swiglu, int64_t M, int64_t N,
                                int64_t K, int64_t lda, int64_t ldw,
                                int64_t ldc, void* stream) {
  if (M < 1 || M > 32 || K % SKQ_KC || N % SK_NT)
    return (int)hipErrorInvalidValue;
  if (lda == 0) lda = K;
  if (ldw == 0) ldw = K;
  if (ldc == 0) ldc = N;
  if (ldw % 8 || lda % 8) return (int)hipErrorInvalidValue;
  dim3 grid((unsigned)(N / SK_NT), 1, 1);
#define KF_SKQ_LAUNCH(MT, SW)                                            \
  hipLaunchKernelGGL((kf_skinny_q8_kernel<MT, SW>), grid,                \
                     dim3(SKL_W * 64), 0, (hipStream_t)stream,           \
                     (unsigned short*)c, (const unsigned short*)a,       \
                     (const unsigned char*)w8, wscale,                   \
                     (const unsigned short*)res, rmsg, rms_eps, (int)M,  \
                     N, K, lda, ldw, ldc)
  if (M > 16) {
    if (swiglu) KF_SKQ_LAUNCH(32, true);
    else KF_SKQ_LAUNCH(32, false);
  } else {
    if (swiglu) KF_SKQ_LAUNCH(16, true);
    else KF_SKQ_LAUNCH(16, false);
  }
  return (int)hipGetLastError();
}

// rmsg (nullable, fp32 [K]): fused input RMSNorm — C = res +
// rmsnorm(A; rmsg, rms_eps) @ W^T (LDS-staged kernels only).
// swiglu != 0: A is the raw [M, 2K] w13 output; fragments are
// silu(A[:, :K]) * A[:, K:] computed in-flight (LDS kernels only).
KF_EXPORT int kf_skinny_gemm(void* c, const void* a, const void* w,
                             const void* res, const float* rmsg,
                             float rms_eps, int64_t swiglu, int64_t M,
                             int64_t N, int64_t K, int64_t lda,
                             int64_t ldw, int64_t ldc, void* stream) {
  if (M < 1 || M > 32 || K % 32 || N % SK_NT) return (int)hipErrorInvalidValue;
  if (lda == 0) lda = K;
  if (ldw == 0) ldw = K;
  if (ldc == 0) ldc = N;
  const bool lds_ok = K % SKL_KC == 0 && ldw % 8 == 0 && lda % 8 == 0;
  if (M > 16 && !lds_ok) return (int)hipErrorInvalidValue;
  if ((rmsg || swiglu) && !lds_ok) return (int)hipErrorInvalidValue;
  dim3 grid((unsigned)(N / SK_NT), 1, 1);
#define KF_SKL_LAUNCH(MT, SW)                                            \
  hipLaunchKernelGGL((kf_skinny_lds_kernel<MT, SW>), grid,               \
                     dim3(SKL_W * 64), 0, (hipStream_t)stream,           \
                     (unsigned short*)c, (const unsigned short*)a,       \
                     (const unsigned short*)w,                           \
                     (const unsigned short*)res, rmsg, rms_eps, (int)M,  \
                     N, K, lda, ldw, ldc)
  if (lds_ok && M > 16) {
    if (swiglu) KF_SKL_LAUNCH(32, true);
    else KF_SKL_LAUNCH(32, false);
  } else if (lds_ok) {
    if (swiglu) KF_SKL_LAUNCH(16, true);
    else KF_SKL_LAUNCH(16, false);
  }
  // direct-load fallback: 8 waves when the grid can't fill the chip with
  // 4-wave blocks (<2 blocks/CU), 4 otherwise
  else if (N / SK_NT < 512)
    hipLaunchKernelGGL(kf_skinny_gemm_kernel<8>, grid, dim3(8 * 64), 0,
                       (hipStream_t)stream, (unsigned short*)c,
                       (const unsigned short*)a, (const unsigned short*)w,
                       (const unsigned short*)res, (int)M, N, K, lda, ldw,
                       ldc);
  else
    hipLaunchKernelGGL(kf_skinny_gemm_kernel<4>, grid, dim3(4 * 64), 0,
                       (hipStream_t)stream, (unsigned short*)c,
                       (const unsigned short*)a, (const unsigned short*)w,
                       (const unsigned short*)res, (int)M, N, K, lda, ldw,
                       ldc);
  return (int)hipGetLastError();
}
