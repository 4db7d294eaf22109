// attention_bwd.hip — causal flash-attention backward (bf16, GQA), CDNA4.
//
// SURVEY.md §2.13 attention_bwd. Two-pass design (no atomics, deterministic):
//   pass 0 (preprocess): delta[b,h,s] = rowsum(dO ∘ O)
//   pass 1 (dQ):  block per (b,hq,q-tile):  dQ = scale · dS K,
//                 dS = P ∘ (dO Vᵀ − delta), P = exp(scale·S − lse)
//   pass 2 (dKV): block per (b,hkv,kv-tile), looping the GQA group's q-heads
//                 and q-tiles:  dV += Pᵀ dO,  dK += scale · dSᵀ Q
//
// All MFMA B-operands read contiguous-in-K from LDS tiles; tiles that are
// consumed column-wise (K in dQ's dS·K step; Q and dO in the dKV pass) are
// transpose-staged ([D][rows]) at load time. All LDS tiles use the XOR
// bank swizzle (guide §6 G4), with the shift adapted to the row byte width.
// lse layout [B,Hq,S] fp32 comes from kf_attn_fwd.

#include "kf_common.h"

typedef __bf16 kf_bf16x8 __attribute__((ext_vector_type(8)));
typedef float kf_f32x4 __attribute__((ext_vector_type(4)));

#define AB_D 128

// Swizzled byte offset; xor_mask picks how many row bits fold in (row_bytes
// 128 -> 7, row_bytes 64 -> 3).
__device__ __forceinline__ int kf_swz2(int row, int byte_in_row, int row_bytes,
                                       int xor_mask) {
  return row * row_bytes + (byte_in_row ^ ((row & xor_mask) << 4));
}

// ---------------------------------------------------------------- pass 0 --
__global__ __launch_bounds__(256) void kf_attn_delta2_kernel(
    float* __restrict__ delta, const unsigned short* __restrict__ dout,
    const unsigned short* __restrict__ o, int64_t B, int S, int Hq) {
  const int64_t nrows = B * (int64_t)S * Hq;
  const int wid = threadIdx.x / KF_WAVE;
  const int lane = threadIdx.x & (KF_WAVE - 1);
  for (int64_t row = blockIdx.x * 4 + wid; row < nrows; row += gridDim.x * 4) {
    const unsigned short* dor = dout + row * AB_D;
    const unsigned short* orow = o + row * AB_D;
    kf_short4 a = *reinterpret_cast<const kf_short4*>(dor + lane * 2);
    kf_short4 bv = *reinterpret_cast<const kf_short4*>(orow + lane * 2);
    float s = kf_bf16_to_f32((unsigned short)a[0]) *
                  kf_bf16_to_f32((unsigned short)bv[0]) +
              kf_bf16_to_f32((unsigned short)a[1]) *
                  kf_bf16_to_f32((unsigned short)bv[1]);
    s = kf_wave_sum(s);
    if (lane == 0) {
      const int h = (int)(row % Hq);
      const int64_t bs = row / Hq;
      const int64_t b = bs / S;
      const int st = (int)(bs % S);
      delta[(b * Hq + h) * (int64_t)S + st] = s;
    }
  }
}

// ---------------------------------------------------------------- pass 1 --
// dQ: block = 256 threads / 4 waves, one (b, hq, 64-row q-tile).
#define DQ_KT 64

__global__ __launch_bounds__(256) void kf_attn_dq_kernel(
    unsigned short* __restrict__ dq, const unsigned short* __restrict__ q,
    const unsigned short* __restrict__ k, const unsigned short* __restrict__ v,
    const unsigned short* __restrict__ dout, const float* __restrict__ lse,
    const float* __restrict__ delta, int64_t B, int S, int Hq, int Hkv,
    int64_t qts, int64_t kts, int64_t dqts, float scale, int causal) {
  __shared__ unsigned char k_lds[DQ_KT * AB_D * 2];    // [64][128] row-major
  __shared__ unsigned char v_lds[DQ_KT * AB_D * 2];    // [64][128] row-major
  __shared__ unsigned char kt_lds[AB_D * DQ_KT * 2];   // [128][64] transposed
  __shared__ unsigned char ds_lds[4][16 * DQ_KT * 2];  // per-wave [16][64]

  const int qt = blockIdx.x, hq = blockIdx.y;
  const int64_t b = blockIdx.z;
  const int hkv = hq / (Hq / Hkv);
  const int tid = threadIdx.x;
  const int w = tid / KF_WAVE, lane = tid & 63, l16 = lane & 15, lg = lane >> 4;
  // Q (strided) and dO (contiguous bshd) fragments in registers
  kf_bf16x8 qfrag[4], dofrag[4];
  {
    const int64_t tok = b * S + qt * 64 + w * 16 + l16;
    const int64_t qb = tok * qts + (int64_t)hq * AB_D;
    const int64_t db = (tok * (int64_t)Hq + hq) * AB_D;
#pragma unroll
    for (int kk = 0; kk < 4; ++kk) {
      qfrag[kk] = *reinterpret_cast<const kf_bf16x8*>(q + qb + kk * 32 + lg * 8);
      dofrag[kk] =
          *reinterpret_cast<const kf_bf16x8*>(dout + db + kk * 32 + lg * 8);
    }
  }
  // per-row lse/delta for the 4 rows this lane's acc regs cover
  float lse_r[4], dlt_r[4];
  {
    const int64_t base = (b * Hq + hq) * (int64_t)S + qt * 64 + w * 16 + lg * 4;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      lse_r[r] = lse[base + r];
      dlt_r[r] = delta[base + r];
    }
  }

  kf_f32x4 dqacc[8];
#pragma unroll
  for (int i = 0; i < 8; ++i) dqacc[i] = kf_f32x4{0.f, 0.f, 0.f, 0.f};

  const int last_kt = causal ? qt : (S / DQ_KT - 1);
  for (int kt = 0; kt <= last_kt; ++kt) {
    __syncthreads();
    {  // stage K (row-major + transposed) and V (row-major)
      const unsigned short* kg = k + (b * S + kt * DQ_KT) * kts + hkv * AB_D;
      const unsigned short* vg = v + (b * S + kt * DQ_KT) * kts + hkv * AB_D;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int vi = tid + 256 * j;
        const int r = vi >> 4, c8 = vi & 15;
        kf_short8 kv8 =
            *reinterpret_cast<const kf_short8*>(kg + r * kts + c8 * 8);
        *reinterpret_cast<kf_short8*>(k_lds + kf_swz2(r, c8 * 16, 256, 7)) = kv8;
#pragma unroll
        for (int jj = 0; jj < 8; ++jj) {
          const int el = (jj + tid) & 7;
          const int dd = c8 * 8 + el;
          *reinterpret_cast<unsigned short*>(
              kt_lds + kf_swz2(dd, r * 2, 128, 7)) = (unsigned short)kv8[el];
        }
        kf_short8 vv8 =
            *reinterpret_cast<const kf_short8*>(vg + r * kts + c8 * 8);
        *reinterpret_cast<kf_short8*>(v_lds + kf_swz2(r, c8 * 16, 256, 7)) = vv8;
      }
    }
    __syncthreads();

    // S = Q K^T, dP = dO V^T  (acc row = q (lg*4+r), col = kv (nt*16+l16))
    kf_f32x4 sacc[4], dpacc[4];
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      sacc[nt] = kf_f32x4{0.f, 0.f, 0.f, 0.f};
      dpacc[nt] = kf_f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kk = 0; kk < 4; ++kk) {
        kf_bf16x8 kb = *reinterpret_cast<const kf_bf16x8*>(
            k_lds + kf_swz2(nt * 16 + l16, kk * 64 + lg * 16, 256, 7));
        sacc[nt] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfrag[kk], kb, sacc[nt], 0, 0, 0);
        kf_bf16x8 vb = *reinterpret_cast<const kf_bf16x8*>(
            v_lds + kf_swz2(nt * 16 + l16, kk * 64 + lg * 16, 256, 7));
        dpacc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dofrag[kk], vb,
                                                            dpacc[nt], 0, 0, 0);
      }
    }

    // dS = P ∘ (dP − delta) · scale  -> bf16 -> per-wave LDS buf
    const int qrow0 = qt * 64 + w * 16 + lg * 4;
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      const int kcol = kt * DQ_KT + nt * 16 + l16;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float p = (causal && kcol > qrow0 + r)
                      ? 0.f
                      : __expf(sacc[nt][r] * scale - lse_r[r]);
        float ds = p * (dpacc[nt][r] - dlt_r[r]) * scale;
        *reinterpret_cast<unsigned short*>(
            ds_lds[w] + kf_swz2(lg * 4 + r, (nt * 16 + l16) * 2, 128, 7)) =
            kf_f32_to_bf16(ds);
      }
    }
    __builtin_amdgcn_s_waitcnt(0);

    // dQ += dS · K   (A = dS [q][kv], B = Kt [d][kv] read col-wise)
    kf_bf16x8 dsfrag[2];
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
      dsfrag[kk] = *reinterpret_cast<const kf_bf16x8*>(
          ds_lds[w] + kf_swz2(l16, kk * 64 + lg * 16, 128, 7));
#pragma unroll
    for (int nt = 0; nt < 8; ++nt) {
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        kf_bf16x8 ktb = *reinterpret_cast<const kf_bf16x8*>(
            kt_lds + kf_swz2(nt * 16 + l16, kk * 64 + lg * 16, 128, 7));
        dqacc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dsfrag[kk], ktb,
                                                            dqacc[nt], 0, 0, 0);
      }
    }
  }

  // epilogue (strided output: dq may live inside a fused dqkv buffer)
  const int qrow0 = qt * 64 + w * 16 + lg * 4;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int64_t base = (b * S + qrow0 + r) * dqts + (int64_t)hq * AB_D;
#pragma unroll
    for (int nt = 0; nt < 8; ++nt)
      dq[base + nt * 16 + l16] = kf_f32_to_bf16(dqacc[nt][r]);
  }
}

// ---------------------------------------------------------------- pass 2 --
// dK/dV: block = 256 threads / 4 waves, one (b, hkv, 64-row kv-tile);
// iterates the GQA group's q-heads × 32-row q-tiles.
#define DKV_QT 32

__global__ __launch_bounds__(256) void kf_attn_dkv_kernel(
    unsigned short* __restrict__ dk, unsigned short* __restrict__ dv,
    const unsigned short* __restrict__ q, const unsigned short* __restrict__ k,
    const unsigned short* __restrict__ v,
    const unsigned short* __restrict__ dout, const float* __restrict__ lse,
    const float* __restrict__ delta, int64_t B, int S, int Hq, int Hkv,
    int64_t qts, int64_t kts, int64_t dkts, float scale, int causal) {
  __shared__ unsigned char q_lds[DKV_QT * AB_D * 2];    // [32][128]
  __shared__ unsigned char qt_lds[AB_D * DKV_QT * 2];   // [128][32]
  __shared__ unsigned char do_lds[DKV_QT * AB_D * 2];   // [32][128]
  __shared__ unsigned char dot_lds[AB_D * DKV_QT * 2];  // [128][32]
  __shared__ unsigned char pt_lds[4][16 * DKV_QT * 2];  // per-wave [16][32]
  __shared__ unsigned char dst_lds[4][16 * DKV_QT * 2];

  const int kt = blockIdx.x, hkv = blockIdx.y;
  const int64_t b = blockIdx.z;
  const int g = Hq / Hkv;
  const int tid = threadIdx.x;
  const int w = tid / KF_WAVE, lane = tid & 63, l16 = lane & 15, lg = lane >> 4;
  const int64_t dstride = (int64_t)Hq * AB_D;  // dout is contiguous bshd

  // K and V fragments in registers (A-operands, 16 kv rows per wave)
  kf_bf16x8 kfrag[4], vfrag[4];
  {
    const int64_t base =
        (b * S + kt * 64 + w * 16 + l16) * kts + (int64_t)hkv * AB_D;
#pragma unroll
    for (int kk = 0; kk < 4; ++kk) {
      kfrag[kk] = *reinterpret_cast<const kf_bf16x8*>(k + base + kk * 32 + lg * 8);
      vfrag[kk] = *reinterpret_cast<const kf_bf16x8*>(v + base + kk * 32 + lg * 8);
    }
  }

  kf_f32x4 dkacc[8], dvacc[8];
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    dkacc[i] = kf_f32x4{0.f, 0.f, 0.f, 0.f};
    dvacc[i] = kf_f32x4{0.f, 0.f, 0.f, 0.f};
  }

  const int qt0 = causal ? (kt * 64) / DKV_QT : 0;  // first q-tile that can see kt
  const int nqt = S / DKV_QT;
  for (int hg = 0; hg < g; ++hg) {
    const int hq = hkv * g + hg;
    for (int qt = qt0; qt < nqt; ++qt) {
      __syncthreads();
      {  // stage Q, dO in both orientations
        const unsigned short* qg =
            q + (b * S + qt * DKV_QT) * qts + (int64_t)hq * AB_D;
        const unsigned short* dog =
            dout + ((b * S + qt * DKV_QT) * (int64_t)Hq + hq) * AB_D;
#pragma unroll
        for (int j = 0; j < 2; ++j) {
          const int vi = tid + 256 * j;     // 0..511 ; 32 rows × 16 chunks
          const int r = vi >> 4, c8 = vi & 15;
          kf_short8 q8 =
              *reinterpret_cast<const kf_short8*>(qg + r * qts + c8 * 8);
          *reinterpret_cast<kf_short8*>(q_lds + kf_swz2(r, c8 * 16, 256, 7)) = q8;
          kf_short8 do8 =
              *reinterpret_cast<const kf_short8*>(dog + r * dstride + c8 * 8);
          *reinterpret_cast<kf_short8*>(do_lds + kf_swz2(r, c8 * 16, 256, 7)) =
              do8;
#pragma unroll
          for (int jj = 0; jj < 8; ++jj) {
            const int el = (jj + tid) & 7;
            const int dd = c8 * 8 + el;
            *reinterpret_cast<unsigned short*>(
                qt_lds + kf_swz2(dd, r * 2, 64, 3)) = (unsigned short)q8[el];
            *reinterpret_cast<unsigned short*>(
                dot_lds + kf_swz2(dd, r * 2, 64, 3)) = (unsigned short)do8[el];
          }
        }
      }
      __syncthreads();

      // S^T = K Q^T, dP^T = V dO^T  (acc row = kv, col = q)
      kf_f32x4 st[2], dpt[2];
#pragma unroll
      for (int nt = 0; nt < 2; ++nt) {
        st[nt] = kf_f32x4{0.f, 0.f, 0.f, 0.f};
        dpt[nt] = kf_f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int kk = 0; kk < 4; ++kk) {
          kf_bf16x8 qb = *reinterpret_cast<const kf_bf16x8*>(
              q_lds + kf_swz2(nt * 16 + l16, kk * 64 + lg * 16, 256, 7));
          st[nt] =
              __builtin_amdgcn_mfma_f32_16x16x32_bf16(kfrag[kk], qb, st[nt], 0, 0, 0);
          kf_bf16x8 dob = *reinterpret_cast<const kf_bf16x8*>(
              do_lds + kf_swz2(nt * 16 + l16, kk * 64 + lg * 16, 256, 7));
          dpt[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(vfrag[kk], dob,
                                                            dpt[nt], 0, 0, 0);
        }
      }

      // P^T, dS^T -> per-wave LDS bufs
      const int krow0 = kt * 64 + w * 16 + lg * 4;
#pragma unroll
      for (int nt = 0; nt < 2; ++nt) {
        const int qcol = qt * DKV_QT + nt * 16 + l16;
        const float lse_c = lse[(b * Hq + hq) * (int64_t)S + qcol];
        const float dlt_c = delta[(b * Hq + hq) * (int64_t)S + qcol];
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float p = (causal && (krow0 + r) > qcol)
                        ? 0.f
                        : __expf(st[nt][r] * scale - lse_c);
          float ds = p * (dpt[nt][r] - dlt_c) * scale;
          const int prow = lg * 4 + r;
          *reinterpret_cast<unsigned short*>(
              pt_lds[w] + kf_swz2(prow, (nt * 16 + l16) * 2, 64, 3)) =
              kf_f32_to_bf16(p);
          *reinterpret_cast<unsigned short*>(
              dst_lds[w] + kf_swz2(prow, (nt * 16 + l16) * 2, 64, 3)) =
              kf_f32_to_bf16(ds);
        }
      }
      __builtin_amdgcn_s_waitcnt(0);

      // dV += P^T dO (B = dOt), dK += dS^T Q (B = Qt); K-dim = q (32) = 1 step
      kf_bf16x8 ptf = *reinterpret_cast<const kf_bf16x8*>(
          pt_lds[w] + kf_swz2(l16, lg * 16, 64, 3));
      kf_bf16x8 dstf = *reinterpret_cast<const kf_bf16x8*>(
          dst_lds[w] + kf_swz2(l16, lg * 16, 64, 3));
#pragma unroll
      for (int nt = 0; nt < 8; ++nt) {
        kf_bf16x8 dob = *reinterpret_cast<const kf_bf16x8*>(
            dot_lds + kf_swz2(nt * 16 + l16, lg * 16, 64, 3));
        dvacc[nt] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(ptf, dob, dvacc[nt], 0, 0, 0);
        kf_bf16x8 qb = *reinterpret_cast<const kf_bf16x8*>(
            qt_lds + kf_swz2(nt * 16 + l16, lg * 16, 64, 3));
        dkacc[nt] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(dstf, qb, dkacc[nt], 0, 0, 0);
      }
    }
  }

  // epilogue: write dK, dV (strided: may live inside a fused dqkv buffer)
  const int krow0 = kt * 64 + w * 16 + lg * 4;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int64_t base = (b * S + krow0 + r) * dkts + (int64_t)hkv * AB_D;
#pragma unroll
    for (int nt = 0; nt < 8; ++nt) {
      dk[base + nt * 16 + l16] = kf_f32_to_bf16(dkacc[nt][r]);
      dv[base + nt * 16 + l16] = kf_f32_to_bf16(dvacc[nt][r]);
    }
  }
}

// 8-wave swapped backward kernels (attention_bwd8.hip)
extern "C" int kf_attn_bwd8_dq(void*, const void*, const void*, const void*,
                               const void*, const float*, const float*,
                               int64_t, int64_t, int64_t, int64_t, int64_t,
                               int64_t, int64_t, float, int, void*);
extern "C" int kf_attn_bwd8_dkv(void*, void*, const void*, const void*,
                                const void*, const void*, const float*,
                                const float*, int64_t, int64_t, int64_t,
                                int64_t, int64_t, int64_t, int64_t, float,
                                int, void*);

KF_EXPORT int kf_attn_bwd(void* dq, void* dk, void* dv, const void* dout,
                          const void* q, const void* k, const void* v,
                          const void* o, const float* lse, float* delta,
                          int64_t B, int64_t S, int64_t Hq, int64_t Hkv,
                          int64_t D, int64_t qts, int64_t kts, int64_t dqts,
                          int64_t dkts, float scale, int causal,
                          void* stream) {
  if (D != AB_D || S % 64 || Hq % Hkv) return (int)hipErrorInvalidValue;
  if (qts == 0) qts = Hq * AB_D;
  if (kts == 0) kts = Hkv * AB_D;
  if (dqts == 0) dqts = Hq * AB_D;
  if (dkts == 0) dkts = Hkv * AB_D;
  hipLaunchKernelGGL(kf_attn_delta2_kernel,
                     dim3(kf_grid_for(B * S * Hq, 4)), dim3(256), 0,
                     (hipStream_t)stream, delta, (const unsigned short*)dout,
                     (const unsigned short*)o, B, (int)S, (int)Hq);
  int err = (int)hipGetLastError();
  if (err) return err;
  if (S % 256 == 0) {  // 8-wave swapped path for the training shapes
    err = kf_attn_bwd8_dq(dq, q, k, v, dout, lse, delta, B, S, Hq, Hkv, qts,
                          kts, dqts, scale, causal, stream);
    if (err) return err;
    return kf_attn_bwd8_dkv(dk, dv, q, k, v, dout, lse, delta, B, S, Hq, Hkv,
                            qts, kts, dkts, scale, causal, stream);
  }
  dim3 gq((unsigned)(S / 64), (unsigned)Hq, (unsigned)B);
  hipLaunchKernelGGL(kf_attn_dq_kernel, gq, dim3(256), 0, (hipStream_t)stream,
                     (unsigned short*)dq, (const unsigned short*)q,
                     (const unsigned short*)k, (const unsigned short*)v,
                     (const unsigned short*)dout, lse, delta, B, (int)S,
                     (int)Hq, (int)Hkv, qts, kts, dqts, scale, causal);
  err = (int)hipGetLastError();
  if (err) return err;
  dim3 gkv((unsigned)(S / 64), (unsigned)Hkv, (unsigned)B);
  hipLaunchKernelGGL(kf_attn_dkv_kernel, gkv, dim3(256), 0,
                     (hipStream_t)stream, (unsigned short*)dk,
                     (unsigned short*)dv, (const unsigned short*)q,
                     (const unsigned short*)k, (const unsigned short*)v,
                     (const unsigned short*)dout, lse, delta, B, (int)S,
                     (int)Hq, (int)Hkv, qts, kts, dkts, scale, causal);
  return (int)hipGetLastError();
}
