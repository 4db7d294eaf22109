// attention_fwd5.hip — v5 causal flash-attention forward (bf16, GQA, gfx950).
//
// Structural successor to v4 (attention_fwd4.hip). v4's hardware ablation
// (scripts/attn_ablate.py + PMC) showed the kernel LDS-THROUGHPUT-bound:
// with 8 waves x 32 q-rows, every wave re-reads the whole K and V tile, so
// per-CU LDS traffic (~1.5x the MFMA pipe time) caps MFMA utilization at
// ~18% no matter how well softmax/staging overlap.
//
// v5 halves the LDS traffic per MFMA: 4 waves x 64 q-rows per wave at
// ONE wave per SIMD (1 block/CU, ~512-VGPR budget).
//
// MEASURED NEGATIVE RESULT (kept for the record): 253 TF vs v4's 445 on
// B4 Hq32 Hkv8 S4096 causal. The register content (~390 VGPR incl. AGPR
// aliasing; forcing 2 waves/SIMD spills 135) pins the kernel at 1
// wave/SIMD, and without a second wave the exposed LDS/MFMA latencies and
// softmax VALU cost more than the halved LDS traffic saves. v4 stays the
// default; this file documents the structure-vs-occupancy tradeoff.
//  * each K fragment (ds_read_b128) now feeds TWO QK^T MFMAs (two
//    independent accumulator chains, one per 32-q half — the chains also
//    interleave to fill the MFMA pipe);
//  * each V^T fragment pair (4x ds_read_b64_tr_b16) feeds FOUR PV MFMAs;
//  * per-CU LDS per 32-kv sub-block drops from 128 KiB to 64 KiB while
//    MFMA content stays — LDS moves below the MFMA pipe time.
// Everything else carries over from v4: double-buffered KVBLK=64 tiles
// with async-STAGE (T14), XOR-swizzled K rows, subtiled V for the tr_read
// hardware transpose (T10, mapping HW-verified by scripts/probe_tr16.py),
// exp2-domain defer-max softmax (T13), cvt_pk+permlane32_swap P repack
// (T12), diagonal-only masking, per-wave/per-half causal skip.
//
// Layout: bshd q [B,S,Hq,D], k/v [B,S,Hkv,D], o [B,S,Hq,D], lse [B,Hq,S]
// fp32. D == 128, S % 256 == 0. SURVEY.md §2.13 attention_fwd row.

#include "kf_common.h"

typedef __bf16 kf_bf16x8v5 __attribute__((ext_vector_type(8)));
typedef float kf_f32x16v5 __attribute__((ext_vector_type(16)));
typedef short kf_short4v5 __attribute__((ext_vector_type(4)));

#define A5_D 128
#define A5_QT 256      // q rows per block (4 waves x 64)
#define A5_KT 64       // kv rows per LDS tile (double-buffered)
#define A5_THREADS 256
#define A5_LOG2E 1.44269504f
#define A5_LN2 0.69314718f
#define A5_THR2 11.5415603f  // 8 * log2(e): defer-max threshold, exp2 units

__device__ __forceinline__ int kf_swz5(int row, int byte_in_row) {
  return row * (A5_D * 2) + (byte_in_row ^ ((row & 7) << 4));
}

__device__ __forceinline__ int kf_vsub5(int kv, int d) {
  return ((d >> 4) << 10) + ((kv >> 2) << 6) + ((kv & 3) << 4) + (d & 15);
}

__device__ __forceinline__ unsigned int kf_cvt_pk_bf16_v5(float lo, float hi) {
  unsigned int r;
  asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}

__device__ __forceinline__ float kf_exp2_v5(float x) {
  float r;
  asm volatile("v_exp_f32 %0, %1" : "=v"(r) : "v"(x));
  return r;
}

#define KF_TR16_V5(dst, addr, OFFLIT)                                   \
  asm volatile("ds_read_b64_tr_b16 %0, %1 offset:" OFFLIT               \
               : "=v"(dst) : "v"(addr))

__global__ __launch_bounds__(A5_THREADS, 1) void kf_attn_fwd5_kernel(
    unsigned short* __restrict__ o, float* __restrict__ lse,
    const unsigned short* __restrict__ q, const unsigned short* __restrict__ k,
    const unsigned short* __restrict__ v, int64_t B, int S, int Hq, int Hkv,
    int64_t qts, int64_t kts, float scale, int causal) {
  __shared__ unsigned char k_lds[2][A5_KT * A5_D * 2];  // 16 KiB x2, swizzled
  __shared__ unsigned char v_lds[2][A5_KT * A5_D * 2];  // 16 KiB x2, subtiled

  const int qt = blockIdx.x, hq = blockIdx.y;
  const int64_t b = blockIdx.z;
  const int hkv = hq / (Hq / Hkv);
  const int tid = threadIdx.x;
  const int w = tid / KF_WAVE;
  const int lane = tid & (KF_WAVE - 1);
  const int l31 = lane & 31;
  const int hi = lane >> 5;
  const float scale2 = scale * A5_LOG2E;

  // ---- persistent Q B-fragments for both 32-q halves ----
  kf_bf16x8v5 qfrag[2][8];
#pragma unroll
  for (int qh = 0; qh < 2; ++qh) {
    const int64_t qbase =
        (b * S + qt * A5_QT + w * 64 + qh * 32 + l31) * qts +
        (int64_t)hq * A5_D;
#pragma unroll
    for (int kk = 0; kk < 8; ++kk)
      qfrag[qh][kk] =
          *reinterpret_cast<const kf_bf16x8v5*>(q + qbase + kk * 16 + hi * 8);
  }

  kf_f32x16v5 oacc[4][2];  // [dt][qh]
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int qh = 0; qh < 2; ++qh) oacc[i][qh] = kf_f32x16v5{0.f};
  float m_run[2] = {-INFINITY, -INFINITY}, l_run[2] = {0.f, 0.f};
  const int qrow0 = qt * A5_QT + w * 64;         // wave's first q row
  const int qrow_g0 = qrow0 + l31;               // lane's q row, half 0
  const int qrow_g1 = qrow0 + 32 + l31;          // lane's q row, half 1

  const int last_kt =
      causal ? (qt * A5_QT + A5_QT - 1) / A5_KT : (S / A5_KT - 1);

  // staging: 256 threads x 4 vectors cover one 64x128 tile for K and V
  const unsigned short* kg0 = k + (b * S) * kts + (int64_t)hkv * A5_D;
  const unsigned short* vg0 = v + (b * S) * kts + (int64_t)hkv * A5_D;
  const int c8 = tid & 15;
  int srow[4];
#pragma unroll
  for (int j = 0; j < 4; ++j) srow[j] = (tid + A5_THREADS * j) >> 4;

  const int g = lane >> 4;
  const unsigned v_lane_off =
      (unsigned)(((g & 1) << 11) + ((g >> 1) << 8) + ((lane & 15) << 3));

  // ---- prologue: stage tile 0 into buffer 0 ----
  kf_short8 kst[4], vst[4];
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    kst[j] = *reinterpret_cast<const kf_short8*>(kg0 + srow[j] * kts + c8 * 8);
    vst[j] = *reinterpret_cast<const kf_short8*>(vg0 + srow[j] * kts + c8 * 8);
  }
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    *reinterpret_cast<kf_short8*>(k_lds[0] + kf_swz5(srow[j], c8 * 16)) =
        kst[j];
    *reinterpret_cast<kf_short8*>(v_lds[0] + 2 * kf_vsub5(srow[j], c8 * 8)) =
        vst[j];
  }
  __syncthreads();

  for (int kt = 0; kt <= last_kt; ++kt) {
    const int cur = kt & 1;
    const bool have_next = kt < last_kt;
    if (have_next) {  // async-STAGE issue (T14)
      const unsigned short* kg = kg0 + (int64_t)(kt + 1) * A5_KT * kts;
      const unsigned short* vg = vg0 + (int64_t)(kt + 1) * A5_KT * kts;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        kst[j] =
            *reinterpret_cast<const kf_short8*>(kg + srow[j] * kts + c8 * 8);
        vst[j] =
            *reinterpret_cast<const kf_short8*>(vg + srow[j] * kts + c8 * 8);
      }
    }

#pragma unroll
    for (int mt = 0; mt < 2; ++mt) {
      const int kv_lo = kt * A5_KT + mt * 32;
      // per-half causal activity (kv_lo and q bounds are 32-multiples)
      const bool act0 = !causal || kv_lo <= qrow0 + 31;
      const bool act1 = !causal || kv_lo <= qrow0 + 63;
      if (!act0 && !act1) continue;

      // ---- QK^T: one K-fragment read feeds both q-halves' chains ----
      kf_bf16x8v5 af[8];
#pragma unroll
      for (int kk = 0; kk < 8; ++kk)
        af[kk] = *reinterpret_cast<const kf_bf16x8v5*>(
            k_lds[cur] + kf_swz5(mt * 32 + l31, kk * 32 + hi * 16));
      kf_f32x16v5 st0 = kf_f32x16v5{0.f}, st1 = kf_f32x16v5{0.f};
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int kk = 0; kk < 8; ++kk) {
        if (act0)
          st0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af[kk], qfrag[0][kk],
                                                        st0, 0, 0, 0);
        if (act1)
          st1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af[kk], qfrag[1][kk],
                                                        st1, 0, 0, 0);
      }
      __builtin_amdgcn_s_setprio(0);

      // ---- softmax per half (exp2 domain, defer-max) -> pb repack ----
      kf_bf16x8v5 pb[2][2];  // [qh][step]
      const int kv0 = kv_lo + hi * 4;
#pragma unroll
      for (int qh = 0; qh < 2; ++qh) {
        if (!(qh ? act1 : act0)) continue;
        kf_f32x16v5& st = qh ? st1 : st0;
        const int qrow_g = qh ? qrow_g1 : qrow_g0;
        const bool need_mask =
            causal && kv_lo + 31 > qrow0 + qh * 32;  // diagonal only
        float mx = -INFINITY;
        if (need_mask) {
#pragma unroll
          for (int r = 0; r < 16; ++r) {
            const int kv = kv0 + (r & 3) + 8 * (r >> 2);
            float sv = st[r] * scale2;
            if (kv > qrow_g) sv = -INFINITY;
            st[r] = sv;
            mx = fmaxf(mx, sv);
          }
        } else {
#pragma unroll
          for (int r = 0; r < 16; ++r) {
            const float sv = st[r] * scale2;
            st[r] = sv;
            mx = fmaxf(mx, sv);
          }
        }
        mx = fmaxf(mx, __shfl_xor(mx, 32, KF_WAVE));
        if (!__all(mx <= m_run[qh] + A5_THR2)) {
          const float m_new = fmaxf(m_run[qh], mx);
          const float alpha = kf_exp2_v5(m_run[qh] - m_new);
#pragma unroll
          for (int i = 0; i < 4; ++i)
#pragma unroll
            for (int r = 0; r < 16; ++r) oacc[i][qh][r] *= alpha;
          l_run[qh] *= alpha;
          m_run[qh] = m_new;
        }
        float lsum = 0.f;
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const float pv = kf_exp2_v5(st[r] - m_run[qh]);
          st[r] = pv;
          lsum += pv;
        }
        lsum += __shfl_xor(lsum, 32, KF_WAVE);
        l_run[qh] += lsum;
#pragma unroll
        for (int step = 0; step < 2; ++step) {
          const int base = step * 8;
          unsigned int w0 = kf_cvt_pk_bf16_v5(st[base + 0], st[base + 1]);
          unsigned int w1 = kf_cvt_pk_bf16_v5(st[base + 2], st[base + 3]);
          unsigned int w2 = kf_cvt_pk_bf16_v5(st[base + 4], st[base + 5]);
          unsigned int w3 = kf_cvt_pk_bf16_v5(st[base + 6], st[base + 7]);
          auto s02 = __builtin_amdgcn_permlane32_swap(w0, w2, false, false);
          auto s13 = __builtin_amdgcn_permlane32_swap(w1, w3, false, false);
          unsigned int u[4] = {(unsigned)s02[0], (unsigned)s13[0],
                               (unsigned)s02[1], (unsigned)s13[1]};
          pb[qh][step] = *reinterpret_cast<kf_bf16x8v5*>(u);
        }
      }

      // ---- PV: each V^T fragment pair feeds up to 4 MFMAs (2 halves) ----
      const unsigned vbase =
          (unsigned)(size_t)(v_lds[cur]) + v_lane_off + (mt << 10);
      kf_short4v5 t[2][4];
      KF_TR16_V5(t[0][0], vbase, "0");
      KF_TR16_V5(t[0][1], vbase, "128");
      KF_TR16_V5(t[0][2], vbase, "512");
      KF_TR16_V5(t[0][3], vbase, "640");
#pragma unroll
      for (int dt = 0; dt < 4; ++dt) {
        if (dt < 3) {
          const unsigned va = vbase + ((dt + 1) << 12);
          KF_TR16_V5(t[(dt + 1) & 1][0], va, "0");
          KF_TR16_V5(t[(dt + 1) & 1][1], va, "128");
          KF_TR16_V5(t[(dt + 1) & 1][2], va, "512");
          KF_TR16_V5(t[(dt + 1) & 1][3], va, "640");
          asm volatile("s_waitcnt lgkmcnt(4)" ::: "memory");
        } else {
          asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        }
        __builtin_amdgcn_sched_barrier(0);  // guide rule 18
        kf_short8 f0 = __builtin_shufflevector(t[dt & 1][0], t[dt & 1][1],
                                               0, 1, 2, 3, 4, 5, 6, 7);
        kf_short8 f1 = __builtin_shufflevector(t[dt & 1][2], t[dt & 1][3],
                                               0, 1, 2, 3, 4, 5, 6, 7);
        __builtin_amdgcn_s_setprio(1);
        if (act0) {
          oacc[dt][0] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              *reinterpret_cast<kf_bf16x8v5*>(&f0), pb[0][0], oacc[dt][0],
              0, 0, 0);
          oacc[dt][0] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              *reinterpret_cast<kf_bf16x8v5*>(&f1), pb[0][1], oacc[dt][0],
              0, 0, 0);
        }
        if (act1) {
          oacc[dt][1] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              *reinterpret_cast<kf_bf16x8v5*>(&f0), pb[1][0], oacc[dt][1],
              0, 0, 0);
          oacc[dt][1] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              *reinterpret_cast<kf_bf16x8v5*>(&f1), pb[1][1], oacc[dt][1],
              0, 0, 0);
        }
        __builtin_amdgcn_s_setprio(0);
      }
    }

    if (have_next) {  // async-STAGE write (vmcnt auto-inserted here)
      const int nxt = cur ^ 1;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        *reinterpret_cast<kf_short8*>(k_lds[nxt] + kf_swz5(srow[j], c8 * 16)) =
            kst[j];
        *reinterpret_cast<kf_short8*>(v_lds[nxt] +
                                      2 * kf_vsub5(srow[j], c8 * 8)) = vst[j];
      }
    }
    __syncthreads();
  }

  // ---- epilogue ----
#pragma unroll
  for (int qh = 0; qh < 2; ++qh) {
    const float inv_l = l_run[qh] > 0.f ? 1.f / l_run[qh] : 0.f;
    const int qrow_g = qh ? qrow_g1 : qrow_g0;
    const int64_t obase = ((b * S + qrow_g) * (int64_t)Hq + hq) * A5_D;
#pragma unroll
    for (int dt = 0; dt < 4; ++dt)
#pragma unroll
      for (int rq = 0; rq < 4; ++rq) {
        const int d0 = dt * 32 + 8 * rq + 4 * hi;
        unsigned short q4[4];
#pragma unroll
        for (int j = 0; j < 4; ++j)
          q4[j] = kf_f32_to_bf16(oacc[dt][qh][rq * 4 + j] * inv_l);
        *reinterpret_cast<kf_short4v5*>(o + obase + d0) =
            *reinterpret_cast<kf_short4v5*>(q4);
      }
    if (hi == 0)
      lse[(b * Hq + hq) * (int64_t)S + qrow_g] =
          m_run[qh] * A5_LN2 + __logf(l_run[qh]);
  }
}

KF_EXPORT int kf_attn_fwd5(void* o, float* lse, const void* q, const void* k,
                           const void* v, int64_t B, int64_t S, int64_t Hq,
                           int64_t Hkv, int64_t D, int64_t qts, int64_t kts,
                           float scale, int causal, void* stream) {
  if (D != A5_D || S % A5_QT || Hq % Hkv) return (int)hipErrorInvalidValue;
  if (qts == 0) qts = Hq * A5_D;
  if (kts == 0) kts = Hkv * A5_D;
  dim3 grid((unsigned)(S / A5_QT), (unsigned)Hq, (unsigned)B);
  hipLaunchKernelGGL(kf_attn_fwd5_kernel, grid, dim3(A5_THREADS), 0,
                     (hipStream_t)stream, (unsigned short*)o, lse,
                     (const unsigned short*)q, (const unsigned short*)k,
                     (const unsigned short*)v, B, (int)S, (int)Hq, (int)Hkv,
                     qts, kts, scale, causal);
  return (int)hipGetLastError();
}
