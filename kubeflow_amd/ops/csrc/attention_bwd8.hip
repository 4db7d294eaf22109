// attention_bwd8.hip — 8-wave swapped flash-attention backward (bf16, GQA).
//
// Applies the verified forward v3 structure (attention_fwd.hip: swapped
// MFMA so softmax-axis values are register-local, cvt_pk_bf16 +
// permlane32_swap to re-fragment P/dS without LDS round-trips) to the two
// backward passes. Requires S % 256 == 0 (the training shapes); smaller
// sequences fall back to the 4-wave kernels in attention_bwd.hip.
//
// pass dQ (block = b, hq, 256-row q-tile; wave = 32 q rows):
//   S^T = mfma(K, Q)      rows kv (regs), cols q (lane) — lse/delta are
//   dP^T = mfma(V, dO)    per-lane scalars
//   dS^T = P^T ∘ (dP^T − delta) · scale        (pure registers)
//   dQ^T += mfma(Kt, exch(dS^T))               (Kt transpose-staged in LDS)
//
// pass dKV (block = b, hkv, 256-row kv-tile; wave = 32 kv rows; loops the
// GQA group's heads × 64-row q-tiles):
//   S = mfma(Q, K)        rows q (regs), cols kv (lane); K/V B-fragments
//   dP = mfma(dO, V)      are persistent per-lane registers
//   dS = P ∘ (dP − delta) · scale              (lse/delta per-REG loads)
//   dV^ += mfma(exch(P), dOt) ; dK += mfma(exch(dS), Qt)

#include "kf_common.h"

typedef __bf16 kf_bf16x8 __attribute__((ext_vector_type(8)));
typedef float kf_f32x16 __attribute__((ext_vector_type(16)));

#define AB_D 128

__device__ __forceinline__ int kf_swz8(int row, int byte_in_row,
                                       int row_bytes) {
  return row * row_bytes + (byte_in_row ^ ((row & 7) << 4));
}

__device__ __forceinline__ unsigned int kf_cvt_pk_bf16b(float lo, float hi) {
  unsigned int r;
  asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}

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

// ---------------------------------------------------------------- pass dQ --
#define DQ8_QT 256
#define DQ8_KT 64   // kv rows per LDS tile (128 measured slower)

__global__ __launch_bounds__(512, 2) void kf_attn_dq8_kernel(
    unsigned short* __restrict__ dq, const unsigned short* __restrict__ q,
    const unsigned short* __restrict__ k, const unsigned short* __restrict__ v,
    const unsigned short* __restrict__ dout, const float* __restrict__ lse,
    const float* __restrict__ delta, int64_t B, int S, int Hq, int Hkv,
    int64_t qts, int64_t kts, int64_t dqts, float scale, int causal) {
  __shared__ unsigned char k_lds[DQ8_KT * AB_D * 2];   // [64][128] row-major
  __shared__ unsigned char v_lds[DQ8_KT * AB_D * 2];   // [64][128] row-major
  __shared__ unsigned char kt_lds[AB_D * DQ8_KT * 2];  // [128][64] transposed

  const int qt = blockIdx.x, hq = blockIdx.y;
  const int64_t b = blockIdx.z;
  const int hkv = hq / (Hq / Hkv);
  const int tid = threadIdx.x;
  const int w = tid / KF_WAVE;
  const int lane = tid & 63, l31 = lane & 31, hi = lane >> 5;
  const int qrow_g = qt * DQ8_QT + w * 32 + l31;

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
  const float lse_q = lse[(b * Hq + hq) * (int64_t)S + qrow_g];
  const float dlt_q = delta[(b * Hq + hq) * (int64_t)S + qrow_g];

  kf_f32x16 dqacc[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) dqacc[i] = kf_f32x16{0.f};

  const int last_kt =
      causal ? (qt * DQ8_QT + DQ8_QT - 1) / DQ8_KT : (S / DQ8_KT - 1);
  for (int kt = 0; kt <= last_kt; ++kt) {
    {  // stage K row-major + Kt transposed + V row-major
      const unsigned short* kg =
          k + (b * S + kt * DQ8_KT) * kts + (int64_t)hkv * AB_D;
      const unsigned short* vg =
          v + (b * S + kt * DQ8_KT) * kts + (int64_t)hkv * AB_D;
#pragma unroll
      for (int j = 0; j < DQ8_KT * 16 / 512; ++j) {
        const int vi = tid + 512 * j;
        const int r = vi >> 4, c8 = vi & 15;
        kf_short8 kv8 =
            *reinterpret_cast<const kf_short8*>(kg + r * kts + c8 * 8);
        *reinterpret_cast<kf_short8*>(k_lds + kf_swz8(r, c8 * 16, 256)) = kv8;
#pragma unroll
        for (int jj = 0; jj < 8; ++jj) {
          const int el = (jj + tid) & 7;
          const int dd = c8 * 8 + el;
          *reinterpret_cast<unsigned short*>(
              kt_lds + kf_swz8(dd, r * 2, DQ8_KT * 2)) =
              (unsigned short)kv8[el];
        }
        kf_short8 vv8 =
            *reinterpret_cast<const kf_short8*>(vg + r * kts + c8 * 8);
        *reinterpret_cast<kf_short8*>(v_lds + kf_swz8(r, c8 * 16, 256)) = vv8;
      }
    }
    __syncthreads();

#pragma unroll 1  // dynamic (full unroll spills at KT=128)
    for (int mt = 0; mt < DQ8_KT / 32; ++mt) {
      kf_f32x16 st = kf_f32x16{0.f}, dpt = kf_f32x16{0.f};
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int kk = 0; kk < 8; ++kk) {
        kf_bf16x8 ka = *reinterpret_cast<const kf_bf16x8*>(
            k_lds + kf_swz8(mt * 32 + l31, kk * 32 + hi * 16, 256));
        st = __builtin_amdgcn_mfma_f32_32x32x16_bf16(ka, qfrag[kk], st, 0, 0, 0);
        kf_bf16x8 va = *reinterpret_cast<const kf_bf16x8*>(
            v_lds + kf_swz8(mt * 32 + l31, kk * 32 + hi * 16, 256));
        dpt = __builtin_amdgcn_mfma_f32_32x32x16_bf16(va, dofrag[kk], dpt, 0,
                                                      0, 0);
      }
      __builtin_amdgcn_s_setprio(0);

      const int kv0 = kt * DQ8_KT + mt * 32 + hi * 4;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int kv = kv0 + (r & 3) + 8 * (r >> 2);
        const float p = (causal && kv > qrow_g)
                            ? 0.f
                            : __expf(st[r] * scale - lse_q);
        st[r] = p * (dpt[r] - dlt_q) * scale;  // dS^T
      }
      kf_bf16x8 pb[2] = {kf_exchange8(st, 0), kf_exchange8(st, 8)};
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int dt = 0; dt < 4; ++dt) {
#pragma unroll
        for (int step = 0; step < 2; ++step) {
          kf_bf16x8 kta = *reinterpret_cast<const kf_bf16x8*>(
              kt_lds + kf_swz8(dt * 32 + l31,
                               mt * 64 + step * 32 + hi * 16, DQ8_KT * 2));
          dqacc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kta, pb[step],
                                                              dqacc[dt], 0, 0,
                                                              0);
        }
      }
      __builtin_amdgcn_s_setprio(0);
    }
    __syncthreads();
  }

  // epilogue: dQ^T regs -> dq (strided)
  const int64_t dqb = (b * S + qrow_g) * dqts + (int64_t)hq * AB_D;
#pragma unroll
  for (int dt = 0; dt < 4; ++dt)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int d = dt * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
      dq[dqb + d] = kf_f32_to_bf16(dqacc[dt][r]);
    }
}

// --------------------------------------------------------------- pass dKV --
// Split into dV and dK kernels: carrying both 64-register accumulators in
// one kernel spilled 62 VGPRs at the 2-waves/SIMD budget; recomputing S in
// a second kernel (+25% MFMA) is far cheaper than scratch traffic.
#define DKV8_KT 256   // kv rows per block (8 waves x 32)
#define DV8_QT 64     // dv8 q-tile (128 measured slower)
#define DK8_QT 64     // dk8 q-tile (128 spilled 29 VGPRs -> regressed)

__global__ __launch_bounds__(512, 2) void kf_attn_dv8_kernel(
    unsigned short* __restrict__ dv, const unsigned short* __restrict__ q,
    const unsigned short* __restrict__ k,
    const unsigned short* __restrict__ dout, const float* __restrict__ lse,
    int64_t B, int S, int Hq, int Hkv, int64_t qts, int64_t kts,
    int64_t dkts, float scale, int causal) {
  __shared__ unsigned char q_lds[DV8_QT * AB_D * 2];    // [64][128]
  __shared__ unsigned char dot_lds[AB_D * DV8_QT * 2];  // [128][64]
  __shared__ float lse_s[DV8_QT];

  const int kt = blockIdx.x, hkv = blockIdx.y;
  const int64_t b = blockIdx.z;
  const int g = Hq / Hkv;
  const int tid = threadIdx.x;
  const int w = tid / KF_WAVE;
  const int lane = tid & 63, l31 = lane & 31, hi = lane >> 5;
  const int kvrow_g = kt * DKV8_KT + w * 32 + l31;

  const unsigned short* kvb_k =
      k + (b * S + kvrow_g) * kts + (int64_t)hkv * AB_D;

  kf_f32x16 dvacc[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) dvacc[i] = kf_f32x16{0.f};

  const int qt0 = causal ? (kt * DKV8_KT) / DV8_QT : 0;
  const int nqt = S / DV8_QT;
  for (int hg = 0; hg < g; ++hg) {
    const int hq = hkv * g + hg;
    for (int qt = qt0; qt < nqt; ++qt) {
      {  // stage Q row-major + dO transposed + the q-tile's lse row
        const unsigned short* qg =
            q + (b * S + qt * DV8_QT) * qts + (int64_t)hq * AB_D;
        const unsigned short* dog =
            dout + ((b * S + qt * DV8_QT) * (int64_t)Hq + hq) * AB_D;
        if (tid < DV8_QT)
          lse_s[tid] =
              lse[(b * Hq + hq) * (int64_t)S + qt * DV8_QT + tid];
#pragma unroll
        for (int j = 0; j < DV8_QT * 16 / 512; ++j) {
          const int vi = tid + 512 * j;
          const int r = vi >> 4, c8 = vi & 15;
          kf_short8 q8 =
              *reinterpret_cast<const kf_short8*>(qg + r * qts + c8 * 8);
          *reinterpret_cast<kf_short8*>(q_lds + kf_swz8(r, c8 * 16, 256)) = q8;
          kf_short8 do8 = *reinterpret_cast<const kf_short8*>(
              dog + r * (int64_t)Hq * AB_D + c8 * 8);
#pragma unroll
          for (int jj = 0; jj < 8; ++jj) {
            const int el = (jj + tid) & 7;
            const int dd = c8 * 8 + el;
            *reinterpret_cast<unsigned short*>(
                dot_lds + kf_swz8(dd, r * 2, DV8_QT * 2)) =
                (unsigned short)do8[el];
          }
        }
      }
      __syncthreads();

#pragma unroll 1
      for (int mt = 0; mt < DV8_QT / 32; ++mt) {
        kf_f32x16 sacc = kf_f32x16{0.f};
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int kk = 0; kk < 8; ++kk) {
          kf_bf16x8 qa = *reinterpret_cast<const kf_bf16x8*>(
              q_lds + kf_swz8(mt * 32 + l31, kk * 32 + hi * 16, 256));
          kf_bf16x8 kb = *reinterpret_cast<const kf_bf16x8*>(
              kvb_k + kk * 16 + hi * 8);
          sacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qa, kb, sacc, 0, 0, 0);
        }
        __builtin_amdgcn_s_setprio(0);

        const int q0 = mt * 32 + hi * 4;  // q-tile-local
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int ql = q0 + (r & 3) + 8 * (r >> 2);
          const int qq = qt * DV8_QT + ql;
          sacc[r] = (causal && kvrow_g > qq)
                        ? 0.f
                        : __expf(sacc[r] * scale - lse_s[ql]);
        }
        kf_bf16x8 pa[2] = {kf_exchange8(sacc, 0), kf_exchange8(sacc, 8)};
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int dt = 0; dt < 4; ++dt) {
#pragma unroll
          for (int step = 0; step < 2; ++step) {
            kf_bf16x8 dob = *reinterpret_cast<const kf_bf16x8*>(
                dot_lds + kf_swz8(dt * 32 + l31,
                                  mt * 64 + step * 32 + hi * 16,
                                  DV8_QT * 2));
            dvacc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                pa[step], dob, dvacc[dt], 0, 0, 0);
          }
        }
        __builtin_amdgcn_s_setprio(0);
      }
      __syncthreads();
    }
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
  __shared__ unsigned char q_lds[DK8_QT * AB_D * 2];    // [64][128]
  __shared__ unsigned char do_lds[DK8_QT * AB_D * 2];   // [64][128]
  __shared__ unsigned char qt_lds[AB_D * DK8_QT * 2];   // [128][64]
  __shared__ float lse_s2[DK8_QT], dlt_s2[DK8_QT];

  const int kt = blockIdx.x, hkv = blockIdx.y;
  const int64_t b = blockIdx.z;
  const int g = Hq / Hkv;
  const int tid = threadIdx.x;
  const int w = tid / KF_WAVE;
  const int lane = tid & 63, l31 = lane & 31, hi = lane >> 5;
  const int kvrow_g = kt * DKV8_KT + w * 32 + l31;

  const unsigned short* kvb_k =
      k + (b * S + kvrow_g) * kts + (int64_t)hkv * AB_D;
  const unsigned short* kvb_v =
      v + (b * S + kvrow_g) * kts + (int64_t)hkv * AB_D;

  kf_f32x16 dkacc[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) dkacc[i] = kf_f32x16{0.f};

  const int qt0 = causal ? (kt * DKV8_KT) / DK8_QT : 0;
  const int nqt = S / DK8_QT;
  for (int hg = 0; hg < g; ++hg) {
    const int hq = hkv * g + hg;
    for (int qt = qt0; qt < nqt; ++qt) {
      {  // stage Q (row-major + transposed), dO row-major, lse/delta rows
        const unsigned short* qg =
            q + (b * S + qt * DK8_QT) * qts + (int64_t)hq * AB_D;
        const unsigned short* dog =
            dout + ((b * S + qt * DK8_QT) * (int64_t)Hq + hq) * AB_D;
        if (tid < DK8_QT)
          lse_s2[tid] =
              lse[(b * Hq + hq) * (int64_t)S + qt * DK8_QT + tid];
        else if (tid < 2 * DK8_QT)
          dlt_s2[tid - DK8_QT] =
              delta[(b * Hq + hq) * (int64_t)S + qt * DK8_QT + tid -
                    DK8_QT];
#pragma unroll
        for (int j = 0; j < DK8_QT * 16 / 512; ++j) {
          const int vi = tid + 512 * j;
          const int r = vi >> 4, c8 = vi & 15;
          kf_short8 q8 =
              *reinterpret_cast<const kf_short8*>(qg + r * qts + c8 * 8);
          *reinterpret_cast<kf_short8*>(q_lds + kf_swz8(r, c8 * 16, 256)) = q8;
          kf_short8 do8 = *reinterpret_cast<const kf_short8*>(
              dog + r * (int64_t)Hq * AB_D + c8 * 8);
          *reinterpret_cast<kf_short8*>(do_lds + kf_swz8(r, c8 * 16, 256)) =
              do8;
#pragma unroll
          for (int jj = 0; jj < 8; ++jj) {
            const int el = (jj + tid) & 7;
            const int dd = c8 * 8 + el;
            *reinterpret_cast<unsigned short*>(
                qt_lds + kf_swz8(dd, r * 2, DK8_QT * 2)) =
                (unsigned short)q8[el];
          }
        }
      }
      __syncthreads();

#pragma unroll 1
      for (int mt = 0; mt < DK8_QT / 32; ++mt) {
        kf_f32x16 sacc = kf_f32x16{0.f}, dpacc = kf_f32x16{0.f};
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int kk = 0; kk < 8; ++kk) {
          kf_bf16x8 qa = *reinterpret_cast<const kf_bf16x8*>(
              q_lds + kf_swz8(mt * 32 + l31, kk * 32 + hi * 16, 256));
          kf_bf16x8 kb = *reinterpret_cast<const kf_bf16x8*>(
              kvb_k + kk * 16 + hi * 8);
          sacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qa, kb, sacc, 0, 0, 0);
          kf_bf16x8 da = *reinterpret_cast<const kf_bf16x8*>(
              do_lds + kf_swz8(mt * 32 + l31, kk * 32 + hi * 16, 256));
          kf_bf16x8 vb = *reinterpret_cast<const kf_bf16x8*>(
              kvb_v + kk * 16 + hi * 8);
          dpacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(da, vb, dpacc, 0, 0,
                                                          0);
        }
        __builtin_amdgcn_s_setprio(0);

        const int q0 = mt * 32 + hi * 4;  // q-tile-local
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int ql = q0 + (r & 3) + 8 * (r >> 2);
          const int qq = qt * DK8_QT + ql;
          const float p = (causal && kvrow_g > qq)
                              ? 0.f
                              : __expf(sacc[r] * scale - lse_s2[ql]);
          sacc[r] = p * (dpacc[r] - dlt_s2[ql]) * scale;  // dS
        }
        kf_bf16x8 dsa[2] = {kf_exchange8(sacc, 0), kf_exchange8(sacc, 8)};
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int dt = 0; dt < 4; ++dt) {
#pragma unroll
          for (int step = 0; step < 2; ++step) {
            kf_bf16x8 qb = *reinterpret_cast<const kf_bf16x8*>(
                qt_lds + kf_swz8(dt * 32 + l31,
                                 mt * 64 + step * 32 + hi * 16,
                                 DK8_QT * 2));
            dkacc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                dsa[step], qb, dkacc[dt], 0, 0, 0);
          }
        }
        __builtin_amdgcn_s_setprio(0);
      }
      __syncthreads();
    }
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
  dim3 gq((unsigned)(S / DQ8_QT), (unsigned)Hq, (unsigned)B);
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
  dim3 gkv((unsigned)(S / DKV8_KT), (unsigned)Hkv, (unsigned)B);
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
