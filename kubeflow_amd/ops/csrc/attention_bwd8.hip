// attention_bwd8.hip — 8-wave swapped flash-attention backward (bf16, GQA).
//
// Round-2 rework applying the techniques HW-verified on the v4 forward
// (attention_fwd4.hip — ablation + PMC on MI355X):
//  * every transpose-scatter staging path (round 1: 8 scalar ds_writes per
//    staged vector; 30k LDS conflict-cycles/wave in dk8) is replaced by the
//    [d/16-slab][r/4-tile][4][16] SUBTILED layout: staged with plain b128
//    writes and consumed BOTH as row-chunk b128 reads (MFMA A/B row
//    operands) and via ds_read_b64_tr_b16 (hardware transpose, guide T10;
//    lane mapping verified by scripts/probe_tr16.py);
//  * all tiles are DOUBLE-BUFFERED with the async-STAGE split (T14): next
//    tile's global loads issue before compute, LDS writes after, one
//    barrier per tile;
//  * P recomputation runs in the exp2 domain (lse is staged pre-multiplied
//    by log2e; one v_exp_f32 per element);
//  * transpose MFMA feeds use a 1-deep tr_read prefetch with counted
//    lgkmcnt; diagonal sub-blocks only pay the causal mask; fully-masked
//    32-row sub-blocks are skipped per wave.
//
// pass dQ (block = b, hq, 256-row q-tile; wave = 32 q rows):
//   S^T = mfma(K, Q), dP^T = mfma(V, dO)      K/V subtiled in LDS
//   dS^T = P^T ∘ (dP^T − delta) · scale       (registers; P via exp2)
//   dQ^T += mfma(tr(K), exch(dS^T))
//
// pass dKV (block = b, hkv, 256-row kv-tile; wave = 32 kv rows; loops the
// GQA group's heads × 64-row q-tiles; split into dV and dK kernels —
// carrying both accumulators spilled 62 VGPRs in round 1):
//   S = mfma(Q, K), dP = mfma(dO, V)          Q/dO subtiled in LDS
//   dV += mfma(exch(P), tr(dO)) ; dK += mfma(exch(dS), tr(Q))

#include "kf_common.h"

typedef __bf16 kf_bf16x8 __attribute__((ext_vector_type(8)));
typedef float kf_f32x16 __attribute__((ext_vector_type(16)));
typedef short kf_short4b __attribute__((ext_vector_type(4)));

#define AB_D 128
#define AB_LOG2E 1.44269504f

// subtiled layout shared with the fwd kernel: element offset of X[r][d]
// (r = the streamed row axis, 64-row tiles; d = 0..127)
__device__ __forceinline__ int kf_vsub8(int r, int d) {
  return ((d >> 4) << 10) + ((r >> 2) << 6) + ((r & 3) << 4) + (d & 15);
}

// row-chunk b128 read address: elements (r, d0..d0+8), d0 % 8 == 0
__device__ __forceinline__ int kf_subrow8(int r, int byte_in_row) {
  return 2 * kf_vsub8(r, byte_in_row >> 1);
}

__device__ __forceinline__ unsigned int kf_cvt_pk_bf16b(float lo, float hi) {
  unsigned int r;
  asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}

__device__ __forceinline__ float kf_exp2b(float x) {
  float r;
  asm volatile("v_exp_f32 %0, %1" : "=v"(r) : "v"(x));
  return r;
}

#define KF_TR16_B(dst, addr, OFFLIT)                                    \
  asm volatile("ds_read_b64_tr_b16 %0, %1 offset:" OFFLIT               \
               : "=v"(dst) : "v"(addr))

// exchange an f32x16 acc half (8 regs from `base`) into one bf16x8 fragment
// whose k-chunks follow the (lane>>5)*8 layout (see fwd v3 derivation).
__device__ __forceinline__ kf_bf16x8 kf_exchange8(const kf_f32x16& a,
                                                  int base) {
  unsigned int w0 = kf_cvt_pk_bf16b(a[base + 0], a[base + 1]);
  unsigned int w1 = kf_cvt_pk_bf16b(a[base + 2], a[base + 3]);
  unsigned int w2 = kf_cvt_pk_bf16b(a[base + 4], a[base + 5]);
  unsigned int w3 = kf_cvt_pk_bf16b(a[base + 6], a[base + 7]);
  auto s02 = __builtin_amdgcn_permlane32_swap(w0, w2, false, false);
  auto s13 = __builtin_amdgcn_permlane32_swap(w1, w3, false, false);
  unsigned int u[4] = {(unsigned)s02[0], (unsigned)s13[0], (unsigned)s02[1],
                       (unsigned)s13[1]};
  return *reinterpret_cast<kf_bf16x8*>(u);
}

// tr_read per-lane base byte offset within one subtiled 64x128 buffer
__device__ __forceinline__ unsigned kf_tr_lane_off(int lane) {
  const int g = lane >> 4;
  return (unsigned)(((g & 1) << 11) + ((g >> 1) << 8) + ((lane & 15) << 3));
}

// accumulate acc[dt] += mfma(tr-frag(dt), pb[step]) over dt=0..3 with a
// 1-deep tr_read prefetch (identical loop to the fwd v4 pv_block).
#define KF_TR_ACC_LOOP(acc, vbase, pb0, pb1)                                 \
  {                                                                          \
    kf_short4b t_[2][4];                                                     \
    KF_TR16_B(t_[0][0], (vbase), "0");                                       \
    KF_TR16_B(t_[0][1], (vbase), "128");                                     \
    KF_TR16_B(t_[0][2], (vbase), "512");                                     \
    KF_TR16_B(t_[0][3], (vbase), "640");                                     \
    _Pragma("unroll") for (int dt = 0; dt < 4; ++dt) {                       \
      if (dt < 3) {                                                          \
        const unsigned va_ = (vbase) + ((dt + 1) << 12);                     \
        KF_TR16_B(t_[(dt + 1) & 1][0], va_, "0");                            \
        KF_TR16_B(t_[(dt + 1) & 1][1], va_, "128");                          \
        KF_TR16_B(t_[(dt + 1) & 1][2], va_, "512");                          \
        KF_TR16_B(t_[(dt + 1) & 1][3], va_, "640");                          \
        asm volatile("s_waitcnt lgkmcnt(4)" ::: "memory");                   \
      } else {                                                               \
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");                   \
      }                                                                      \
      __builtin_amdgcn_sched_barrier(0);                                     \
      kf_short8 f0_ = __builtin_shufflevector(t_[dt & 1][0], t_[dt & 1][1],  \
                                              0, 1, 2, 3, 4, 5, 6, 7);       \
      kf_short8 f1_ = __builtin_shufflevector(t_[dt & 1][2], t_[dt & 1][3],  \
                                              0, 1, 2, 3, 4, 5, 6, 7);       \
      __builtin_amdgcn_s_setprio(1);                                         \
      acc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(                     \
          *reinterpret_cast<kf_bf16x8*>(&f0_), (pb0), acc[dt], 0, 0, 0);     \
      acc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(                     \
          *reinterpret_cast<kf_bf16x8*>(&f1_), (pb1), acc[dt], 0, 0, 0);     \
      __builtin_amdgcn_s_setprio(0);                                         \
    }                                                                        \
  }

// variant with the tr-frag as the B operand (dV += mfma(pa, tr(dO)))
#define KF_TR_ACC_LOOP_B(acc, vbase, pa0, pa1)                               \
  {                                                                          \
    kf_short4b t_[2][4];                                                     \
    KF_TR16_B(t_[0][0], (vbase), "0");                                       \
    KF_TR16_B(t_[0][1], (vbase), "128");                                     \
    KF_TR16_B(t_[0][2], (vbase), "512");                                     \
    KF_TR16_B(t_[0][3], (vbase), "640");                                     \
    _Pragma("unroll") for (int dt = 0; dt < 4; ++dt) {                       \
      if (dt < 3) {                                                          \
        const unsigned va_ = (vbase) + ((dt + 1) << 12);                     \
        KF_TR16_B(t_[(dt + 1) & 1][0], va_, "0");                            \
        KF_TR16_B(t_[(dt + 1) & 1][1], va_, "128");                          \
        KF_TR16_B(t_[(dt + 1) & 1][2], va_, "512");                          \
        KF_TR16_B(t_[(dt + 1) & 1][3], va_, "640");                          \
        asm volatile("s_waitcnt lgkmcnt(4)" ::: "memory");                   \
      } else {                                                               \
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");                   \
      }                                                                      \
      __builtin_amdgcn_sched_barrier(0);                                     \
      kf_short8 f0_ = __builtin_shufflevector(t_[dt & 1][0], t_[dt & 1][1],  \
                                              0, 1, 2, 3, 4, 5, 6, 7);       \
      kf_short8 f1_ = __builtin_shufflevector(t_[dt & 1][2], t_[dt & 1][3],  \
                                              0, 1, 2, 3, 4, 5, 6, 7);       \
      __builtin_amdgcn_s_setprio(1);                                         \
      acc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(                     \
          (pa0), *reinterpret_cast<kf_bf16x8*>(&f0_), acc[dt], 0, 0, 0);     \
      acc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(                     \
          (pa1), *reinterpret_cast<kf_bf16x8*>(&f1_), acc[dt], 0, 0, 0);     \
      __builtin_amdgcn_s_setprio(0);                                         \
    }                                                                        \
  }

// ---------------------------------------------------------------- pass dQ --
#define DQ8_QT 256
#define DQ8_KT 64

__global__ __launch_bounds__(512, 2) void kf_attn_dq8_kernel(
    unsigned short* __restrict__ dq, const unsigned short* __restrict__ q,
    const unsigned short* __restrict__ k, const unsigned short* __restrict__ v,
    const unsigned short* __restrict__ dout, const float* __restrict__ lse,
    const float* __restrict__ delta, int64_t B, int S, int Hq, int Hkv,
    int64_t qts, int64_t kts, int64_t dqts, float scale, int causal) {
  __shared__ unsigned char k_lds[2][DQ8_KT * AB_D * 2];  // subtiled
  __shared__ unsigned char v_lds[2][DQ8_KT * AB_D * 2];  // subtiled

  // 1-D LPT grid: longest q-tiles (largest qt) dispatch first
  const int nqt = gridDim.x / (Hq * (int)B);
  const int qt = causal ? (nqt - 1 - blockIdx.x / (Hq * (int)B))
                        : (int)(blockIdx.x / (Hq * (int)B));
  const int rest = blockIdx.x % (Hq * (int)B);
  const int hq = rest % Hq;
  const int64_t b = rest / Hq;
  const int hkv = hq / (Hq / Hkv);
  const int tid = threadIdx.x;
  const int w = tid / KF_WAVE;
  const int lane = tid & 63, l31 = lane & 31, hi = lane >> 5;
  const int qrow_g = qt * DQ8_QT + w * 32 + l31;
  const int wave_qmax = qt * DQ8_QT + w * 32 + 31;
  const float scale2 = scale * AB_LOG2E;

  // persistent B-fragments of Q and dO for this lane's q column
  kf_bf16x8 qfrag[8], dofrag[8];
  {
    const int64_t qb = (b * S + qrow_g) * qts + (int64_t)hq * AB_D;
    const int64_t db = ((b * S + qrow_g) * (int64_t)Hq + hq) * AB_D;
#pragma unroll
    for (int kk = 0; kk < 8; ++kk) {
      qfrag[kk] =
          *reinterpret_cast<const kf_bf16x8*>(q + qb + kk * 16 + hi * 8);
      dofrag[kk] =
          *reinterpret_cast<const kf_bf16x8*>(dout + db + kk * 16 + hi * 8);
    }
  }
  const float lse2_q =
      lse[(b * Hq + hq) * (int64_t)S + qrow_g] * AB_LOG2E;
  const float dlt_q = delta[(b * Hq + hq) * (int64_t)S + qrow_g];

  kf_f32x16 dqacc[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) dqacc[i] = kf_f32x16{0.f};

  const unsigned short* kg0 = k + (b * S) * kts + (int64_t)hkv * AB_D;
  const unsigned short* vg0 = v + (b * S) * kts + (int64_t)hkv * AB_D;
  const int sr0 = tid >> 4, sr1 = (tid + 512) >> 4, c8 = tid & 15;
  const unsigned tr_off = kf_tr_lane_off(lane);

  const int last_kt =
      causal ? (qt * DQ8_QT + DQ8_QT - 1) / DQ8_KT : (S / DQ8_KT - 1);

  // prologue: stage tile 0
  kf_short8 kst0, kst1, vst0, vst1;
  kst0 = *reinterpret_cast<const kf_short8*>(kg0 + sr0 * kts + c8 * 8);
  kst1 = *reinterpret_cast<const kf_short8*>(kg0 + sr1 * kts + c8 * 8);
  vst0 = *reinterpret_cast<const kf_short8*>(vg0 + sr0 * kts + c8 * 8);
  vst1 = *reinterpret_cast<const kf_short8*>(vg0 + sr1 * kts + c8 * 8);
  *reinterpret_cast<kf_short8*>(k_lds[0] + 2 * kf_vsub8(sr0, c8 * 8)) = kst0;
  *reinterpret_cast<kf_short8*>(k_lds[0] + 2 * kf_vsub8(sr1, c8 * 8)) = kst1;
  *reinterpret_cast<kf_short8*>(v_lds[0] + 2 * kf_vsub8(sr0, c8 * 8)) = vst0;
  *reinterpret_cast<kf_short8*>(v_lds[0] + 2 * kf_vsub8(sr1, c8 * 8)) = vst1;
  __syncthreads();

  for (int kt = 0; kt <= last_kt; ++kt) {
    const int cur = kt & 1;
    const bool have_next = kt < last_kt;
    if (have_next) {
      const unsigned short* kg = kg0 + (int64_t)(kt + 1) * DQ8_KT * kts;
      const unsigned short* vg = vg0 + (int64_t)(kt + 1) * DQ8_KT * kts;
      kst0 = *reinterpret_cast<const kf_short8*>(kg + sr0 * kts + c8 * 8);
      kst1 = *reinterpret_cast<const kf_short8*>(kg + sr1 * kts + c8 * 8);
      vst0 = *reinterpret_cast<const kf_short8*>(vg + sr0 * kts + c8 * 8);
      vst1 = *reinterpret_cast<const kf_short8*>(vg + sr1 * kts + c8 * 8);
    }

#pragma unroll
    for (int mt = 0; mt < 2; ++mt) {
      const int kv_lo = kt * DQ8_KT + mt * 32;
      if (causal && kv_lo > wave_qmax) continue;  // wave-uniform skip
      kf_f32x16 st = kf_f32x16{0.f}, dpt = kf_f32x16{0.f};
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int kk = 0; kk < 8; ++kk) {
        kf_bf16x8 ka = *reinterpret_cast<const kf_bf16x8*>(
            k_lds[cur] + kf_subrow8(mt * 32 + l31, kk * 32 + hi * 16));
        st = __builtin_amdgcn_mfma_f32_32x32x16_bf16(ka, qfrag[kk], st, 0, 0,
                                                     0);
        kf_bf16x8 va = *reinterpret_cast<const kf_bf16x8*>(
            v_lds[cur] + kf_subrow8(mt * 32 + l31, kk * 32 + hi * 16));
        dpt = __builtin_amdgcn_mfma_f32_32x32x16_bf16(va, dofrag[kk], dpt, 0,
                                                      0, 0);
      }
      __builtin_amdgcn_s_setprio(0);

      const int kv0 = kv_lo + hi * 4;
      const bool need_mask = causal && kv_lo + 31 > qt * DQ8_QT + w * 32;
      if (need_mask) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int kv = kv0 + (r & 3) + 8 * (r >> 2);
          const float p = (kv > qrow_g)
                              ? 0.f
                              : kf_exp2b(st[r] * scale2 - lse2_q);
          st[r] = p * (dpt[r] - dlt_q) * scale;  // dS^T
        }
      } else {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const float p = kf_exp2b(st[r] * scale2 - lse2_q);
          st[r] = p * (dpt[r] - dlt_q) * scale;
        }
      }
      kf_bf16x8 pb0 = kf_exchange8(st, 0), pb1 = kf_exchange8(st, 8);
      const unsigned vbase =
          (unsigned)(size_t)(k_lds[cur]) + tr_off + (mt << 10);
      KF_TR_ACC_LOOP(dqacc, vbase, pb0, pb1);
    }

    if (have_next) {
      const int nxt = cur ^ 1;
      *reinterpret_cast<kf_short8*>(k_lds[nxt] + 2 * kf_vsub8(sr0, c8 * 8)) =
          kst0;
      *reinterpret_cast<kf_short8*>(k_lds[nxt] + 2 * kf_vsub8(sr1, c8 * 8)) =
          kst1;
      *reinterpret_cast<kf_short8*>(v_lds[nxt] + 2 * kf_vsub8(sr0, c8 * 8)) =
          vst0;
      *reinterpret_cast<kf_short8*>(v_lds[nxt] + 2 * kf_vsub8(sr1, c8 * 8)) =
          vst1;
    }
    __syncthreads();
  }

  // epilogue: dQ^T regs -> dq (4 consecutive d per quad -> b64 stores)
  const int64_t dqb = (b * S + qrow_g) * dqts + (int64_t)hq * AB_D;
#pragma unroll
  for (int dt = 0; dt < 4; ++dt)
#pragma unroll
    for (int rq = 0; rq < 4; ++rq) {
      const int d0 = dt * 32 + 8 * rq + 4 * hi;
      unsigned short q4[4];
#pragma unroll
      for (int j = 0; j < 4; ++j)
        q4[j] = kf_f32_to_bf16(dqacc[dt][rq * 4 + j]);
      *reinterpret_cast<kf_short4b*>(dq + dqb + d0) =
          *reinterpret_cast<kf_short4b*>(q4);
    }
}

// --------------------------------------------------------------- pass dKV --
// Split into dV and dK kernels: carrying both 64-register accumulators in
// one kernel spilled 62 VGPRs at the 2-waves/SIMD budget; recomputing S in
// a second kernel (+25% MFMA) is far cheaper than scratch traffic.
#define DKV8_KT 256   // kv rows per block (8 waves x 32)
#define DKV8_QT 64    // q rows per LDS tile

__global__ __launch_bounds__(512, 2) void kf_attn_dv8_kernel(
    unsigned short* __restrict__ dv, const unsigned short* __restrict__ q,
    const unsigned short* __restrict__ k,
    const unsigned short* __restrict__ dout, const float* __restrict__ lse,
    int64_t B, int S, int Hq, int Hkv, int64_t qts, int64_t kts,
    int64_t dkts, float scale, int causal) {
  __shared__ unsigned char q_lds[2][DKV8_QT * AB_D * 2];   // subtiled
  __shared__ unsigned char dot_lds[2][DKV8_QT * AB_D * 2]; // subtiled
  __shared__ float lse_s[2][DKV8_QT];                      // pre-mul log2e

  // 1-D grid decoded kt-major: causal work decreases with kt, so the
  // longest blocks (kt=0) dispatch first (LPT order — with only
  // S/256 x Hkv x B blocks the schedule tail otherwise sets the wall)
  const int kt = blockIdx.x / (Hkv * (int)B);
  const int rest = blockIdx.x % (Hkv * (int)B);
  const int hkv = rest % Hkv;
  const int64_t b = rest / Hkv;
  const int g = Hq / Hkv;
  const int tid = threadIdx.x;
  const int w = tid / KF_WAVE;
  const int lane = tid & 63, l31 = lane & 31, hi = lane >> 5;
  const int kvrow_g = kt * DKV8_KT + w * 32 + l31;
  const int wave_kv_min = kt * DKV8_KT + w * 32;
  const float scale2 = scale * AB_LOG2E;

  const unsigned short* kvb_k =
      k + (b * S + kvrow_g) * kts + (int64_t)hkv * AB_D;
  // persistent K B-fragments: loaded ONCE; without the explicit array the
  // compiler rematerialized these 8 global loads inside every sub-block
  // (regression caught by rocprof: dv8 2.97 ms vs round-1 2.16)
  kf_bf16x8 kbf[8];
#pragma unroll
  for (int kk = 0; kk < 8; ++kk)
    kbf[kk] =
        *reinterpret_cast<const kf_bf16x8*>(kvb_k + kk * 16 + hi * 8);

  kf_f32x16 dvacc[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) dvacc[i] = kf_f32x16{0.f};

  const int qt0 = causal ? (kt * DKV8_KT) / DKV8_QT : 0;
  const int nqt = S / DKV8_QT;
  const int per_head = nqt - qt0;
  const int total = g * per_head;
  const int sr0 = tid >> 4, sr1 = (tid + 512) >> 4, c8 = tid & 15;
  const unsigned tr_off = kf_tr_lane_off(lane);

  // async-STAGE split (T14): loads issue at the top of the iteration, LDS
  // writes land after compute has covered the HBM latency.
  kf_short8 sq0, sq1, sd0, sd1;
  float lse_ld = 0.f;
  auto stage_load = [&](int flat) {
    const int hq = hkv * g + flat / per_head;
    const int qt = qt0 + flat % per_head;
    const unsigned short* qg =
        q + (b * S + qt * DKV8_QT) * qts + (int64_t)hq * AB_D;
    const unsigned short* dog =
        dout + ((b * S + qt * DKV8_QT) * (int64_t)Hq + hq) * AB_D;
    if (tid < DKV8_QT)
      lse_ld = lse[(b * Hq + hq) * (int64_t)S + qt * DKV8_QT + tid];
    sq0 = *reinterpret_cast<const kf_short8*>(qg + sr0 * qts + c8 * 8);
    sq1 = *reinterpret_cast<const kf_short8*>(qg + sr1 * qts + c8 * 8);
    sd0 = *reinterpret_cast<const kf_short8*>(
        dog + sr0 * (int64_t)Hq * AB_D + c8 * 8);
    sd1 = *reinterpret_cast<const kf_short8*>(
        dog + sr1 * (int64_t)Hq * AB_D + c8 * 8);
  };
  auto stage_write = [&](int buf) {
    if (tid < DKV8_QT) lse_s[buf][tid] = lse_ld * AB_LOG2E;
    *reinterpret_cast<kf_short8*>(q_lds[buf] + 2 * kf_vsub8(sr0, c8 * 8)) =
        sq0;
    *reinterpret_cast<kf_short8*>(q_lds[buf] + 2 * kf_vsub8(sr1, c8 * 8)) =
        sq1;
    *reinterpret_cast<kf_short8*>(dot_lds[buf] + 2 * kf_vsub8(sr0, c8 * 8)) =
        sd0;
    *reinterpret_cast<kf_short8*>(dot_lds[buf] + 2 * kf_vsub8(sr1, c8 * 8)) =
        sd1;
  };

  stage_load(0);
  stage_write(0);
  __syncthreads();
  for (int flat = 0; flat < total; ++flat) {
    const int cur = flat & 1;
    const bool have_next = flat + 1 < total;
    if (have_next) stage_load(flat + 1);
    const int qt = qt0 + flat % per_head;

#pragma unroll
    for (int mt = 0; mt < 2; ++mt) {
      const int q_lo = qt * DKV8_QT + mt * 32;  // global q of sub-block
      // causal: contributes iff some qq >= kv in this wave
      if (causal && q_lo + 31 < wave_kv_min) continue;
      kf_f32x16 sacc = kf_f32x16{0.f};
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int kk = 0; kk < 8; ++kk) {
        kf_bf16x8 qa = *reinterpret_cast<const kf_bf16x8*>(
            q_lds[cur] + kf_subrow8(mt * 32 + l31, kk * 32 + hi * 16));
        sacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qa, kbf[kk], sacc, 0,
                                                       0, 0);
      }
      __builtin_amdgcn_s_setprio(0);

      const int q0 = mt * 32 + hi * 4;  // q-tile-local
      const bool need_mask = causal && q_lo <= kt * DKV8_KT + w * 32 + 31;
      if (need_mask) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int ql = q0 + (r & 3) + 8 * (r >> 2);
          const int qq = qt * DKV8_QT + ql;
          sacc[r] = (kvrow_g > qq)
                        ? 0.f
                        : kf_exp2b(sacc[r] * scale2 - lse_s[cur][ql]);
        }
      } else {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int ql = q0 + (r & 3) + 8 * (r >> 2);
          sacc[r] = kf_exp2b(sacc[r] * scale2 - lse_s[cur][ql]);
        }
      }
      kf_bf16x8 pa0 = kf_exchange8(sacc, 0), pa1 = kf_exchange8(sacc, 8);
      const unsigned vbase =
          (unsigned)(size_t)(dot_lds[cur]) + tr_off + (mt << 10);
      KF_TR_ACC_LOOP_B(dvacc, vbase, pa0, pa1);
    }
    if (have_next) stage_write(cur ^ 1);
    __syncthreads();
  }

  const int kvbase = kt * DKV8_KT + w * 32 + hi * 4;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int kvr = kvbase + (r & 3) + 8 * (r >> 2);
    const int64_t base = (b * S + kvr) * dkts + (int64_t)hkv * AB_D;
#pragma unroll
    for (int dt = 0; dt < 4; ++dt)
      dv[base + dt * 32 + l31] = kf_f32_to_bf16(dvacc[dt][r]);
  }
}

__global__ __launch_bounds__(512, 2) void kf_attn_dk8_kernel(
    unsigned short* __restrict__ dk, const unsigned short* __restrict__ q,
    const unsigned short* __restrict__ k, const unsigned short* __restrict__ v,
    const unsigned short* __restrict__ dout, const float* __restrict__ lse,
    const float* __restrict__ delta, int64_t B, int S, int Hq, int Hkv,
    int64_t qts, int64_t kts, int64_t dkts, float scale, int causal) {
  __shared__ unsigned char q_lds[2][DKV8_QT * AB_D * 2];   // subtiled
  __shared__ unsigned char do_lds[2][DKV8_QT * AB_D * 2];  // subtiled
  __shared__ float lse_s2[2][DKV8_QT], dlt_s2[2][DKV8_QT];

  const int kt = blockIdx.x / (Hkv * (int)B);  // LPT decode (see dv8)
  const int rest = blockIdx.x % (Hkv * (int)B);
  const int hkv = rest % Hkv;
  const int64_t b = rest / Hkv;
  const int g = Hq / Hkv;
  const int tid = threadIdx.x;
  const int w = tid / KF_WAVE;
  const int lane = tid & 63, l31 = lane & 31, hi = lane >> 5;
  const int kvrow_g = kt * DKV8_KT + w * 32 + l31;
  const int wave_kv_min = kt * DKV8_KT + w * 32;
  const float scale2 = scale * AB_LOG2E;

  const unsigned short* kvb_k =
      k + (b * S + kvrow_g) * kts + (int64_t)hkv * AB_D;
  const unsigned short* kvb_v =
      v + (b * S + kvrow_g) * kts + (int64_t)hkv * AB_D;
  kf_bf16x8 kbf[8], vbf[8];  // persistent B-fragments (see dv8 note)
#pragma unroll
  for (int kk = 0; kk < 8; ++kk) {
    kbf[kk] =
        *reinterpret_cast<const kf_bf16x8*>(kvb_k + kk * 16 + hi * 8);
    vbf[kk] =
        *reinterpret_cast<const kf_bf16x8*>(kvb_v + kk * 16 + hi * 8);
  }

  kf_f32x16 dkacc[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) dkacc[i] = kf_f32x16{0.f};

  const int qt0 = causal ? (kt * DKV8_KT) / DKV8_QT : 0;
  const int nqt = S / DKV8_QT;
  const int per_head = nqt - qt0;
  const int total = g * per_head;
  const int sr0 = tid >> 4, sr1 = (tid + 512) >> 4, c8 = tid & 15;
  const unsigned tr_off = kf_tr_lane_off(lane);

  kf_short8 sq0, sq1, sd0, sd1;
  float lse_ld = 0.f, dlt_ld = 0.f;
  auto stage_load = [&](int flat) {
    const int hq = hkv * g + flat / per_head;
    const int qt = qt0 + flat % per_head;
    const unsigned short* qg =
        q + (b * S + qt * DKV8_QT) * qts + (int64_t)hq * AB_D;
    const unsigned short* dog =
        dout + ((b * S + qt * DKV8_QT) * (int64_t)Hq + hq) * AB_D;
    if (tid < DKV8_QT)
      lse_ld = lse[(b * Hq + hq) * (int64_t)S + qt * DKV8_QT + tid];
    else if (tid < 2 * DKV8_QT)
      dlt_ld =
          delta[(b * Hq + hq) * (int64_t)S + qt * DKV8_QT + tid - DKV8_QT];
    sq0 = *reinterpret_cast<const kf_short8*>(qg + sr0 * qts + c8 * 8);
    sq1 = *reinterpret_cast<const kf_short8*>(qg + sr1 * qts + c8 * 8);
    sd0 = *reinterpret_cast<const kf_short8*>(
        dog + sr0 * (int64_t)Hq * AB_D + c8 * 8);
    sd1 = *reinterpret_cast<const kf_short8*>(
        dog + sr1 * (int64_t)Hq * AB_D + c8 * 8);
  };
  auto stage_write = [&](int buf) {
    if (tid < DKV8_QT) lse_s2[buf][tid] = lse_ld * AB_LOG2E;
    else if (tid < 2 * DKV8_QT) dlt_s2[buf][tid - DKV8_QT] = dlt_ld;
    *reinterpret_cast<kf_short8*>(q_lds[buf] + 2 * kf_vsub8(sr0, c8 * 8)) =
        sq0;
    *reinterpret_cast<kf_short8*>(q_lds[buf] + 2 * kf_vsub8(sr1, c8 * 8)) =
        sq1;
    *reinterpret_cast<kf_short8*>(do_lds[buf] + 2 * kf_vsub8(sr0, c8 * 8)) =
        sd0;
    *reinterpret_cast<kf_short8*>(do_lds[buf] + 2 * kf_vsub8(sr1, c8 * 8)) =
        sd1;
  };

  stage_load(0);
  stage_write(0);
  __syncthreads();
  for (int flat = 0; flat < total; ++flat) {
    const int cur = flat & 1;
    const bool have_next = flat + 1 < total;
    if (have_next) stage_load(flat + 1);
    const int qt = qt0 + flat % per_head;

#pragma unroll
    for (int mt = 0; mt < 2; ++mt) {
      const int q_lo = qt * DKV8_QT + mt * 32;
      if (causal && q_lo + 31 < wave_kv_min) continue;
      kf_f32x16 sacc = kf_f32x16{0.f}, dpacc = kf_f32x16{0.f};
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int kk = 0; kk < 8; ++kk) {
        kf_bf16x8 qa = *reinterpret_cast<const kf_bf16x8*>(
            q_lds[cur] + kf_subrow8(mt * 32 + l31, kk * 32 + hi * 16));
        sacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qa, kbf[kk], sacc, 0,
                                                       0, 0);
        kf_bf16x8 da = *reinterpret_cast<const kf_bf16x8*>(
            do_lds[cur] + kf_subrow8(mt * 32 + l31, kk * 32 + hi * 16));
        dpacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(da, vbf[kk], dpacc,
                                                        0, 0, 0);
      }
      __builtin_amdgcn_s_setprio(0);

      const int q0 = mt * 32 + hi * 4;
      const bool need_mask = causal && q_lo <= kt * DKV8_KT + w * 32 + 31;
      if (need_mask) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int ql = q0 + (r & 3) + 8 * (r >> 2);
          const int qq = qt * DKV8_QT + ql;
          const float p =
              (kvrow_g > qq)
                  ? 0.f
                  : kf_exp2b(sacc[r] * scale2 - lse_s2[cur][ql]);
          sacc[r] = p * (dpacc[r] - dlt_s2[cur][ql]) * scale;  // dS
        }
      } else {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int ql = q0 + (r & 3) + 8 * (r >> 2);
          const float p = kf_exp2b(sacc[r] * scale2 - lse_s2[cur][ql]);
          sacc[r] = p * (dpacc[r] - dlt_s2[cur][ql]) * scale;
        }
      }
      kf_bf16x8 ds0 = kf_exchange8(sacc, 0), ds1 = kf_exchange8(sacc, 8);
      const unsigned vbase =
          (unsigned)(size_t)(q_lds[cur]) + tr_off + (mt << 10);
      KF_TR_ACC_LOOP_B(dkacc, vbase, ds0, ds1);
    }
    if (have_next) stage_write(cur ^ 1);
    __syncthreads();
  }

  const int kvbase = kt * DKV8_KT + w * 32 + hi * 4;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int kvr = kvbase + (r & 3) + 8 * (r >> 2);
    const int64_t base = (b * S + kvr) * dkts + (int64_t)hkv * AB_D;
#pragma unroll
    for (int dt = 0; dt < 4; ++dt)
      dk[base + dt * 32 + l31] = kf_f32_to_bf16(dkacc[dt][r]);
  }
}

KF_EXPORT int kf_attn_bwd8_dq(void* dq, const void* q, const void* k,
                              const void* v, const void* dout,
                              const float* lse, const float* delta, int64_t B,
                              int64_t S, int64_t Hq, int64_t Hkv, int64_t qts,
                              int64_t kts, int64_t dqts, float scale,
                              int causal, void* stream) {
  dim3 gq((unsigned)((S / DQ8_QT) * Hq * B), 1, 1);
  hipLaunchKernelGGL(kf_attn_dq8_kernel, gq, dim3(512), 0,
                     (hipStream_t)stream, (unsigned short*)dq,
                     (const unsigned short*)q, (const unsigned short*)k,
                     (const unsigned short*)v, (const unsigned short*)dout,
                     lse, delta, B, (int)S, (int)Hq, (int)Hkv, qts, kts, dqts,
                     scale, causal);
  return (int)hipGetLastError();
}

KF_EXPORT int kf_attn_bwd8_dkv(void* dk, void* dv, const void* q,
                               const void* k, const void* v, const void* dout,
                               const float* lse, const float* delta,
                               int64_t B, int64_t S, int64_t Hq, int64_t Hkv,
                               int64_t qts, int64_t kts, int64_t dkts,
                               float scale, int causal, void* stream) {
  dim3 gkv((unsigned)((S / DKV8_KT) * Hkv * B), 1, 1);
  hipLaunchKernelGGL(kf_attn_dv8_kernel, gkv, dim3(512), 0,
                     (hipStream_t)stream, (unsigned short*)dv,
                     (const unsigned short*)q, (const unsigned short*)k,
                     (const unsigned short*)dout, lse, B, (int)S, (int)Hq,
                     (int)Hkv, qts, kts, dkts, scale, causal);
  int err = (int)hipGetLastError();
  if (err) return err;
  hipLaunchKernelGGL(kf_attn_dk8_kernel, gkv, dim3(512), 0,
                     (hipStream_t)stream, (unsigned short*)dk,
                     (const unsigned short*)q, (const unsigned short*)k,
                     (const unsigned short*)v, (const unsigned short*)dout,
                     lse, delta, B, (int)S, (int)Hq, (int)Hkv, qts, kts,
                     dkts, scale, causal);
  return (int)hipGetLastError();
}
