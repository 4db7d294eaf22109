// attention_fwd.hip — causal flash-attention forward (bf16, GQA) for CDNA4.
//
// "attention_fwd HIP kernel: causal flash attention, bf16,
// GQA 32q/8kv; grid (batch×heads×seq-tiles), MFMA 16x16x32 bf16, LDS-tiled
// KV blocks". Layout is bshd: q [B,S,Hq,D], k/v [B,S,Hkv,D], o [B,S,Hq,D],
// lse [B,Hq,S] fp32 (saved for backward). D == 128, S % 64 == 0 (the Python
// wrapper pads causal sequences).
//
// v2 structure (per CDNA4 guide §B + §5.5 technique catalog):
//  * block = 256 threads = 4 waves; each block owns one (b, hq, 128-row
//    q-tile); each wave owns 32 q rows (2 MFMA row-tiles) — doubles the
//    MFMA work per staged K/V byte vs a 64-row tile.
//  * K/V stream through LDS in 64-row tiles: K row-major [64][128], V
//    transpose-staged [128][64]; both XOR-swizzled (byte ^= (row&7)<<4,
//    guide G4 — row-major D=128 tiles are a 16-way bank conflict).
//  * only lgkmcnt is drained at the P round-trip (inline asm + guide rule
//    18 sched_barrier(0)); cross-tile register prefetch was tried and
//    REVERTED: it pushed the kernel to 124 B/lane scratch spill at the
//    2-waves/SIMD budget, which costs more than the latency it hides.
//  * s_setprio(1) around the MFMA clusters (T5).
//  * online softmax entirely in registers/cross-lane shuffles; P makes one
//    swizzled LDS round-trip per wave to reorient acc -> A-fragment.

#include "kf_common.h"

typedef __bf16 kf_bf16x8 __attribute__((ext_vector_type(8)));
typedef float kf_f32x4 __attribute__((ext_vector_type(4)));

#define AT_D 128
#define AT_QT 128     // q rows per block
#define AT_KT 64      // kv rows per LDS tile
#define AT_RT 2       // 16-row MFMA row-tiles per wave (32 q rows)
#define AT_THREADS 256

__device__ __forceinline__ int kf_swz(int row, int byte_in_row, int row_bytes) {
  return row * row_bytes + (byte_in_row ^ ((row & 7) << 4));
}

__global__ __launch_bounds__(AT_THREADS, 2) void kf_attn_fwd_kernel(
    unsigned short* __restrict__ o, float* __restrict__ lse,
    const unsigned short* __restrict__ q, const unsigned short* __restrict__ k,
    const unsigned short* __restrict__ v, int64_t B, int S, int Hq, int Hkv,
    int64_t qts, int64_t kts, float scale, int causal) {
  __shared__ unsigned char k_lds[AT_KT * AT_D * 2];        // [64][128] swz
  __shared__ unsigned char vt_lds[AT_D * AT_KT * 2];       // [128][64] swz
  __shared__ unsigned char p_lds[4][32 * AT_KT * 2];       // per-wave [32][64]

  const int qt = blockIdx.x, hq = blockIdx.y;
  const int64_t b = blockIdx.z;
  const int hkv = hq / (Hq / Hkv);
  const int tid = threadIdx.x;
  const int w = tid / KF_WAVE;
  const int lane = tid & (KF_WAVE - 1);
  const int l16 = lane & 15;
  const int lg = lane >> 4;


  // ---- Q fragments in registers: rt row-tiles × 4 K-chunks ----
  kf_bf16x8 qfrag[AT_RT][4];
#pragma unroll
  for (int rt = 0; rt < AT_RT; ++rt) {
    const int64_t qbase =
        (b * S + qt * AT_QT + w * 32 + rt * 16 + l16) * qts +
        (int64_t)hq * AT_D;
#pragma unroll
    for (int kk = 0; kk < 4; ++kk)
      qfrag[rt][kk] =
          *reinterpret_cast<const kf_bf16x8*>(q + qbase + kk * 32 + lg * 8);
  }

  kf_f32x4 oacc[AT_RT][8];
  float m_run[AT_RT][4], l_run[AT_RT][4];
#pragma unroll
  for (int rt = 0; rt < AT_RT; ++rt) {
#pragma unroll
    for (int i = 0; i < 8; ++i) oacc[rt][i] = kf_f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      m_run[rt][r] = -INFINITY;
      l_run[rt][r] = 0.f;
    }
  }

  const int last_kt =
      causal ? (qt * AT_QT + AT_QT - 1) / AT_KT : (S / AT_KT - 1);

  for (int kt = 0; kt <= last_kt; ++kt) {
    // ---- stage K (row-major) and V (transposed) into LDS ----
    {
      const unsigned short* kg =
          k + (b * S + kt * AT_KT) * kts + (int64_t)hkv * AT_D;
      const unsigned short* vg =
          v + (b * S + kt * AT_KT) * kts + (int64_t)hkv * AT_D;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int vi = tid + AT_THREADS * j;
        const int r = vi >> 4, c8 = vi & 15;
        kf_short8 kv8 =
            *reinterpret_cast<const kf_short8*>(kg + r * kts + c8 * 8);
        *reinterpret_cast<kf_short8*>(k_lds + kf_swz(r, c8 * 16, AT_D * 2)) =
            kv8;
        kf_short8 vv8 =
            *reinterpret_cast<const kf_short8*>(vg + r * kts + c8 * 8);
#pragma unroll
        for (int jj = 0; jj < 8; ++jj) {
          const int el = (jj + tid) & 7;  // stagger to spread banks
          const int dd = c8 * 8 + el;
          *reinterpret_cast<unsigned short*>(
              vt_lds + kf_swz(dd, r * 2, AT_KT * 2)) =
              (unsigned short)vv8[el];
        }
      }
    }
    __syncthreads();

    // ---- per row-tile: S = scale·QK^T, online softmax, P, PV ----
#pragma unroll
    for (int rt = 0; rt < AT_RT; ++rt) {
      kf_f32x4 sacc[4];
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        sacc[nt] = kf_f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int kk = 0; kk < 4; ++kk) {
          kf_bf16x8 bfrag = *reinterpret_cast<const kf_bf16x8*>(
              k_lds + kf_swz(nt * 16 + l16, kk * 64 + lg * 16, AT_D * 2));
          sacc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              qfrag[rt][kk], bfrag, sacc[nt], 0, 0, 0);
        }
      }
      __builtin_amdgcn_s_setprio(0);

      const int qrow0 = qt * AT_QT + w * 32 + rt * 16 + lg * 4;
      float tile_max[4] = {-INFINITY, -INFINITY, -INFINITY, -INFINITY};
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        const int kcol = kt * AT_KT + nt * 16 + l16;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float sv = sacc[nt][r] * scale;
          if (causal && kcol > qrow0 + r) sv = -INFINITY;
          sacc[nt][r] = sv;
          tile_max[r] = fmaxf(tile_max[r], sv);
        }
      }
      float alpha[4], psum[4];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
#pragma unroll
        for (int off = 8; off > 0; off >>= 1)
          tile_max[r] =
              fmaxf(tile_max[r], __shfl_xor(tile_max[r], off, KF_WAVE));
        const float m_new = fmaxf(m_run[rt][r], tile_max[r]);
        alpha[r] =
            (m_run[rt][r] == -INFINITY) ? 0.f : __expf(m_run[rt][r] - m_new);
        m_run[rt][r] = m_new;
        psum[r] = 0.f;
      }
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float pv = (sacc[nt][r] == -INFINITY)
                         ? 0.f
                         : __expf(sacc[nt][r] - m_run[rt][r]);
          sacc[nt][r] = pv;
          psum[r] += pv;
          *reinterpret_cast<unsigned short*>(
              p_lds[w] + kf_swz(rt * 16 + lg * 4 + r, (nt * 16 + l16) * 2,
                                AT_KT * 2)) = kf_f32_to_bf16(pv);
        }
      }
#pragma unroll
      for (int r = 0; r < 4; ++r) {
#pragma unroll
        for (int off = 8; off > 0; off >>= 1)
          psum[r] += __shfl_xor(psum[r], off, KF_WAVE);
        l_run[rt][r] = l_run[rt][r] * alpha[r] + psum[r];
      }
#pragma unroll
      for (int nt = 0; nt < 8; ++nt)
#pragma unroll
        for (int r = 0; r < 4; ++r) oacc[rt][nt][r] *= alpha[r];

      // drain only LDS counters (P writes by other lanes of this wave);
      // vmcnt (the prefetch) stays in flight. Guide rule 18: fence the
      // scheduler right after the asm wait.
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_sched_barrier(0);

      kf_bf16x8 pfrag[2];
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
        pfrag[kk] = *reinterpret_cast<const kf_bf16x8*>(
            p_lds[w] + kf_swz(rt * 16 + l16, kk * 64 + lg * 16, AT_KT * 2));
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int nt = 0; nt < 8; ++nt) {
#pragma unroll
        for (int kk = 0; kk < 2; ++kk) {
          kf_bf16x8 vfrag = *reinterpret_cast<const kf_bf16x8*>(
              vt_lds + kf_swz(nt * 16 + l16, kk * 64 + lg * 16, AT_KT * 2));
          oacc[rt][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              pfrag[kk], vfrag, oacc[rt][nt], 0, 0, 0);
        }
      }
      __builtin_amdgcn_s_setprio(0);
    }
    __syncthreads();  // all waves done reading K/V before next overwrite
  }

  // ---- epilogue ----
#pragma unroll
  for (int rt = 0; rt < AT_RT; ++rt) {
    const int qrow0 = qt * AT_QT + w * 32 + rt * 16 + lg * 4;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const float inv_l = (l_run[rt][r] > 0.f) ? 1.f / l_run[rt][r] : 0.f;
      const int64_t obase = ((b * S + qrow0 + r) * (int64_t)Hq + hq) * AT_D;
#pragma unroll
      for (int nt = 0; nt < 8; ++nt)
        o[obase + nt * 16 + l16] = kf_f32_to_bf16(oacc[rt][nt][r] * inv_l);
      if (l16 == 0)
        lse[(b * Hq + hq) * (int64_t)S + qrow0 + r] =
            m_run[rt][r] + __logf(l_run[rt][r]);
    }
  }
}


// ---------------------------------------------------------------------------
// v3: 8-wave swapped-QK^T kernel (S % 256 == 0 — the training shapes).
//
// Per the CDNA4 guide's verified attention ladder (§B "8-warp 32×32"):
//  * block = 512 threads = 8 waves; each block owns a 256-row q-tile, each
//    wave 32 q rows; 32×32×16 MFMA.
//  * SWAPPED QK^T: computes S^T = mfma(K, Q) so each lane's accumulator
//    column is ONE q row (col = lane&31) — the online-softmax max/sum are
//    register reductions plus a single shfl_xor(32), no LDS round-trip and
//    no lgkmcnt drain on the softmax path.
//  * P^T -> PV B-fragments in-register via v_cvt_pk_bf16_f32 +
//    permlane32_swap (guide T12): 8 cvt_pk + 4 swaps per 32-kv tile.
//  * K row-major and V transpose-staged in XOR-swizzled LDS as in v2;
//    PV A-fragments (V^T) are b128 reads from Vt.
// C/D layout for mfma_f32_32x32x16_bf16: col = lane&31,
// row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)  [guide §3, m74/m101].
// ---------------------------------------------------------------------------

typedef float kf_f32x16 __attribute__((ext_vector_type(16)));

#define A8_QT 256  // q rows per block (4-wave/128-row blocks measured 193 TF)
#define A8_KT 128  // kv rows per LDS tile (64 -> 236 TF, 128 -> 296, 256 -> 240)
#define A8_THREADS 512

__device__ __forceinline__ unsigned int kf_cvt_pk_bf16(float lo, float hi) {
  unsigned int r;
  asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}

__global__ __launch_bounds__(A8_THREADS, 2) void kf_attn_fwd8_kernel(
    unsigned short* __restrict__ o, float* __restrict__ lse,
    const unsigned short* __restrict__ q, const unsigned short* __restrict__ k,
    const unsigned short* __restrict__ v, int64_t B, int S, int Hq, int Hkv,
    int64_t qts, int64_t kts, float scale, int causal) {
  __shared__ unsigned char k_lds[A8_KT * AT_D * 2];   // [A8_KT][128] swz
  __shared__ unsigned char vt_lds[AT_D * A8_KT * 2];  // [128][A8_KT] swz

  const int qt = blockIdx.x, hq = blockIdx.y;
  const int64_t b = blockIdx.z;
  const int hkv = hq / (Hq / Hkv);
  const int tid = threadIdx.x;
  const int w = tid / KF_WAVE;
  const int lane = tid & (KF_WAVE - 1);
  const int l31 = lane & 31;
  const int hi = lane >> 5;  // 0 for lanes 0-31, 1 for 32-63

  // ---- persistent Q B-fragments: B[k=d][n=q], lane holds q-col l31 ----
  kf_bf16x8 qfrag[8];
  {
    const int64_t qbase =
        (b * S + qt * A8_QT + w * 32 + l31) * qts + (int64_t)hq * AT_D;
#pragma unroll
    for (int kk = 0; kk < 8; ++kk)
      qfrag[kk] =
          *reinterpret_cast<const kf_bf16x8*>(q + qbase + kk * 16 + hi * 8);
  }

  kf_f32x16 oacc[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) oacc[i] = kf_f32x16{0.f};
  float m_run = -INFINITY, l_run = 0.f;
  const int qrow_g = qt * A8_QT + w * 32 + l31;

  const int last_kt =
      causal ? (qt * A8_QT + A8_QT - 1) / A8_KT : (S / A8_KT - 1);
  for (int kt = 0; kt <= last_kt; ++kt) {
    // ---- stage K row-major + V transposed (512 threads, 2 vecs each) ----
    {
      const unsigned short* kg =
          k + (b * S + kt * A8_KT) * kts + (int64_t)hkv * AT_D;
      const unsigned short* vg =
          v + (b * S + kt * A8_KT) * kts + (int64_t)hkv * AT_D;
#pragma unroll
      for (int j = 0; j < A8_KT * 16 / A8_THREADS; ++j) {
        const int vi = tid + A8_THREADS * j;
        const int r = vi >> 4, c8 = vi & 15;
        kf_short8 kv8 =
            *reinterpret_cast<const kf_short8*>(kg + r * kts + c8 * 8);
        *reinterpret_cast<kf_short8*>(k_lds + kf_swz(r, c8 * 16, AT_D * 2)) =
            kv8;
        kf_short8 vv8 =
            *reinterpret_cast<const kf_short8*>(vg + r * kts + c8 * 8);
#pragma unroll
        for (int jj = 0; jj < 8; ++jj) {
          const int el = (jj + tid) & 7;
          const int dd = c8 * 8 + el;
          *reinterpret_cast<unsigned short*>(
              vt_lds + kf_swz(dd, r * 2, A8_KT * 2)) = (unsigned short)vv8[el];
        }
      }
    }
    __syncthreads();

#pragma unroll 1  // dynamic: full unroll at KT=128 spilled 127 VGPRs
    for (int mt = 0; mt < A8_KT / 32; ++mt) {  // 32-kv M-tiles
      // ---- S^T = mfma(K, Q): rows kv, cols q ----
      kf_f32x16 st = kf_f32x16{0.f};
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int kk = 0; kk < 8; ++kk) {
        kf_bf16x8 afrag = *reinterpret_cast<const kf_bf16x8*>(
            k_lds + kf_swz(mt * 32 + l31, kk * 32 + hi * 16, AT_D * 2));
        st = __builtin_amdgcn_mfma_f32_32x32x16_bf16(afrag, qfrag[kk], st,
                                                     0, 0, 0);
      }
      __builtin_amdgcn_s_setprio(0);

      // ---- mask + online softmax (register-local; one shfl per reduce) --
      const int kv0 = kt * A8_KT + mt * 32 + hi * 4;
      float mx = -INFINITY;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int kv = kv0 + (r & 3) + 8 * (r >> 2);
        float sv = st[r] * scale;
        if (causal && kv > qrow_g) sv = -INFINITY;
        st[r] = sv;
        mx = fmaxf(mx, sv);
      }
      mx = fmaxf(mx, __shfl_xor(mx, 32, KF_WAVE));
      const float m_new = fmaxf(m_run, mx);
      const float alpha =
          (m_run == -INFINITY) ? 0.f : __expf(m_run - m_new);
      float lsum = 0.f;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const float pv =
            (st[r] == -INFINITY) ? 0.f : __expf(st[r] - m_new);
        st[r] = pv;
        lsum += pv;
      }
      lsum += __shfl_xor(lsum, 32, KF_WAVE);
      l_run = l_run * alpha + lsum;
      m_run = m_new;
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int r = 0; r < 16; ++r) oacc[i][r] *= alpha;

      // ---- P^T -> two B-fragments via cvt_pk + permlane32_swap ----
      // regs 0-7 cover kv-local {0..3, 8..11} (+4 for hi lanes);
      // regs 8-15 cover {16..19, 24..27} (+4 for hi lanes).
      kf_bf16x8 pb[2];
#pragma unroll
      for (int step = 0; step < 2; ++step) {
        const int base = step * 8;
        unsigned int w0 = kf_cvt_pk_bf16(st[base + 0], st[base + 1]);
        unsigned int w1 = kf_cvt_pk_bf16(st[base + 2], st[base + 3]);
        unsigned int w2 = kf_cvt_pk_bf16(st[base + 4], st[base + 5]);
        unsigned int w3 = kf_cvt_pk_bf16(st[base + 6], st[base + 7]);
        auto s02 = __builtin_amdgcn_permlane32_swap(w0, w2, false, false);
        auto s13 = __builtin_amdgcn_permlane32_swap(w1, w3, false, false);
        unsigned int u[4] = {(unsigned)s02[0], (unsigned)s13[0],
                             (unsigned)s02[1], (unsigned)s13[1]};
        pb[step] = *reinterpret_cast<kf_bf16x8*>(u);
      }

      // ---- O^T += V^T P^T over the two 16-kv steps ----
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int dt = 0; dt < 4; ++dt) {
#pragma unroll
        for (int step = 0; step < 2; ++step) {
          kf_bf16x8 vfrag = *reinterpret_cast<const kf_bf16x8*>(
              vt_lds + kf_swz(dt * 32 + l31,
                              mt * 64 + step * 32 + hi * 16, A8_KT * 2));
          oacc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vfrag, pb[step],
                                                             oacc[dt], 0, 0, 0);
        }
      }
      __builtin_amdgcn_s_setprio(0);
    }
    __syncthreads();
  }

  // ---- epilogue: O^T regs -> o[token][d]; lse per q row ----
  const float inv_l = l_run > 0.f ? 1.f / l_run : 0.f;
  const int64_t obase = ((b * S + qrow_g) * (int64_t)Hq + hq) * AT_D;
#pragma unroll
  for (int dt = 0; dt < 4; ++dt)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int d = dt * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
      o[obase + d] = kf_f32_to_bf16(oacc[dt][r] * inv_l);
    }
  if (hi == 0)
    lse[(b * Hq + hq) * (int64_t)S + qrow_g] = m_run + __logf(l_run);
}


// Newer kernels: v4 (attention_fwd4.hip, 8-wave async-STAGE + tr_read V)
// and v5 (attention_fwd5.hip, 4-wave x 64-q LDS-traffic-halved). Dispatch
// order: KF_ATTN_IMPL=5 (default) -> 4 -> 3 for bisection.
KF_EXPORT int kf_attn_fwd4(void* o, float* lse, const void* q, const void* k,
                           const void* v, int64_t B, int64_t S, int64_t Hq,
                           int64_t Hkv, int64_t D, int64_t qts, int64_t kts,
                           float scale, int causal, void* stream);
KF_EXPORT int kf_attn_fwd5(void* o, float* lse, const void* q, const void* k,
                           const void* v, int64_t B, int64_t S, int64_t Hq,
                           int64_t Hkv, int64_t D, int64_t qts, int64_t kts,
                           float scale, int causal, void* stream);

static int kf_attn_impl() {
  static int cached = -1;
  if (cached < 0) {
    const char* e = getenv("KF_ATTN_IMPL");
    cached = (e && e[0] >= '3' && e[0] <= '5') ? e[0] - '0' : 4;
    const char* v4 = getenv("KF_ATTN_V4");  // legacy bisection knob
    if (v4 && v4[0] == '0' && cached > 3) cached = 3;
  }
  return cached;
}

KF_EXPORT int kf_attn_fwd(void* o, float* lse, const void* q, const void* k,
                          const void* v, int64_t B, int64_t S, int64_t Hq,
                          int64_t Hkv, int64_t D, int64_t qts, int64_t kts,
                          float scale, int causal, void* stream) {
  if (D != AT_D || S % AT_QT || Hq % Hkv) return (int)hipErrorInvalidValue;
  if (qts == 0) qts = Hq * AT_D;
  if (kts == 0) kts = Hkv * AT_D;
  if (S % A8_QT == 0 && kf_attn_impl() == 5)
    return kf_attn_fwd5(o, lse, q, k, v, B, S, Hq, Hkv, D, qts, kts, scale,
                        causal, stream);
  if (S % A8_QT == 0 && kf_attn_impl() == 4)
    return kf_attn_fwd4(o, lse, q, k, v, B, S, Hq, Hkv, D, qts, kts, scale,
                        causal, stream);
  if (S % A8_QT == 0) {  // 8-wave swapped kernel for the training shapes
    dim3 grid((unsigned)(S / A8_QT), (unsigned)Hq, (unsigned)B);
    // (an async global_load_lds double-buffer variant was measured NULL
    // here — see docs/ROUND1.md negative results — and removed)
    hipLaunchKernelGGL(kf_attn_fwd8_kernel, grid, dim3(A8_THREADS), 0,
                       (hipStream_t)stream, (unsigned short*)o, lse,
                       (const unsigned short*)q, (const unsigned short*)k,
                       (const unsigned short*)v, B, (int)S, (int)Hq,
                       (int)Hkv, qts, kts, scale, causal);
    return (int)hipGetLastError();
  }
  dim3 grid((unsigned)(S / AT_QT), (unsigned)Hq, (unsigned)B);
  hipLaunchKernelGGL(kf_attn_fwd_kernel, grid, dim3(AT_THREADS), 0,
                     (hipStream_t)stream, (unsigned short*)o, lse,
                     (const unsigned short*)q, (const unsigned short*)k,
                     (const unsigned short*)v, B, (int)S, (int)Hq, (int)Hkv,
                     qts, kts, scale, causal);
  return (int)hipGetLastError();
}
