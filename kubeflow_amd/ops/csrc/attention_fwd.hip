// attention_fwd.hip — causal flash-attention forward (bf16, GQA) for CDNA4.
//
// "attention_fwd HIP kernel: causal flash attention, bf16,
// GQA 32q/8kv; grid (batch×heads×seq-tiles), MFMA 16x16x32 bf16, LDS-tiled
// KV blocks". Layout is bshd: q [B,S,Hq,D], k/v [B,S,Hkv,D], o [B,S,Hq,D],
// lse [B,Hq,S] fp32 (saved for backward). D == 128.
//
// Structure (per CDNA4 guide §B):
//  * block = 256 threads = 4 waves; each block owns one (b, hq, 64-row
//    q-tile); each wave owns 16 q rows. Q fragments live in registers.
//  * K/V are streamed in 64-row tiles through LDS. K is stored row-major
//    [64][128]; V is transpose-staged as Vt [128][64] so the PV MFMA
//    B-operand reads are contiguous ds_read_b128. Both tiles use the
//    XOR bank-swizzle byte ^= ((row&7)<<4) (guide §6 G4: row-major D=128
//    tiles are a 16-way bank conflict otherwise).
//  * QK^T and PV use mfma_f32_16x16x32_bf16; the online-softmax running
//    (m, l) state and the O accumulator stay in registers; P makes one
//    swizzled LDS round-trip per wave to reorient acc-layout -> A-fragment
//    layout.

#include "kf_common.h"

typedef __bf16 kf_bf16x8 __attribute__((ext_vector_type(8)));
typedef float kf_f32x4 __attribute__((ext_vector_type(4)));

#define AT_D 128
#define AT_QT 64      // q rows per block
#define AT_KT 64      // kv rows per LDS tile
#define AT_WAVES 4    // 16 q rows per wave
#define AT_THREADS (AT_WAVES * KF_WAVE)

// Swizzled byte offset inside a row-major [rows][cols*2B] LDS tile.
__device__ __forceinline__ int kf_swz(int row, int byte_in_row, int row_bytes) {
  return row * row_bytes + (byte_in_row ^ ((row & 7) << 4));
}

__global__ __launch_bounds__(AT_THREADS) void kf_attn_fwd_kernel(
    unsigned short* __restrict__ o, float* __restrict__ lse,
    const unsigned short* __restrict__ q, const unsigned short* __restrict__ k,
    const unsigned short* __restrict__ v, int64_t B, int S, int Hq, int Hkv,
    float scale, int causal) {
  __shared__ unsigned char k_lds[AT_KT * AT_D * 2];        // [64][128] bf16 swz
  __shared__ unsigned char vt_lds[AT_D * AT_KT * 2];       // [128][64] bf16 swz
  __shared__ unsigned char p_lds[AT_WAVES][16 * AT_KT * 2];  // per-wave [16][64]

  const int qt = blockIdx.x, hq = blockIdx.y;
  const int64_t b = blockIdx.z;
  const int hkv = hq / (Hq / Hkv);
  const int tid = threadIdx.x;
  const int w = tid / KF_WAVE;
  const int lane = tid & (KF_WAVE - 1);
  const int l16 = lane & 15;      // col index inside a 16-wide MFMA tile
  const int lg = lane >> 4;       // 16-lane group 0..3

  const int64_t qstride = (int64_t)Hq * AT_D;   // tokens stride in q/o
  const int64_t kstride = (int64_t)Hkv * AT_D;

  // ---- load Q fragments (A-operand): row l16, k-chunk kk*32 + lg*8 ----
  kf_bf16x8 qfrag[4];
  {
    const int64_t qbase =
        ((b * S + qt * AT_QT + w * 16 + l16) * (int64_t)Hq + hq) * AT_D;
#pragma unroll
    for (int kk = 0; kk < 4; ++kk)
      qfrag[kk] = *reinterpret_cast<const kf_bf16x8*>(q + qbase + kk * 32 + lg * 8);
  }

  kf_f32x4 oacc[8];
#pragma unroll
  for (int i = 0; i < 8; ++i) oacc[i] = kf_f32x4{0.f, 0.f, 0.f, 0.f};
  float m_run[4] = {-INFINITY, -INFINITY, -INFINITY, -INFINITY};
  float l_run[4] = {0.f, 0.f, 0.f, 0.f};

  const int last_kt = causal ? qt : ((S + AT_KT - 1) / AT_KT - 1);
  for (int kt = 0; kt <= last_kt; ++kt) {
    // ---- stage K (row-major) and V (transposed) into LDS, swizzled ----
    __syncthreads();
    {
      const unsigned short* kg =
          k + ((b * S + kt * AT_KT) * (int64_t)Hkv + hkv) * AT_D;
      const unsigned short* vg =
          v + ((b * S + kt * AT_KT) * (int64_t)Hkv + hkv) * AT_D;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int vi = tid + AT_THREADS * j;   // 0..1023
        const int r = vi >> 4;                 // kv row
        const int c8 = vi & 15;                // 8-col chunk
        kf_short8 kv8 =
            *reinterpret_cast<const kf_short8*>(kg + r * kstride + c8 * 8);
        *reinterpret_cast<kf_short8*>(
            k_lds + kf_swz(r, c8 * 16, AT_D * 2)) = kv8;
        kf_short8 vv8 =
            *reinterpret_cast<const kf_short8*>(vg + r * kstride + c8 * 8);
#pragma unroll
        for (int jj = 0; jj < 8; ++jj) {
          const int el = (jj + tid) & 7;       // stagger to spread banks
          const int dd = c8 * 8 + el;
          *reinterpret_cast<unsigned short*>(
              vt_lds + kf_swz(dd, r * 2, AT_KT * 2)) = (unsigned short)vv8[el];
        }
      }
    }
    __syncthreads();

    // ---- S = scale * Q K^T  (4 n-tiles of 16 kv cols) ----
    kf_f32x4 sacc[4];
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      sacc[nt] = kf_f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kk = 0; kk < 4; ++kk) {
        kf_bf16x8 bfrag = *reinterpret_cast<const kf_bf16x8*>(
            k_lds + kf_swz(nt * 16 + l16, kk * 64 + lg * 16, AT_D * 2));
        sacc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            qfrag[kk], bfrag, sacc[nt], 0, 0, 0);
      }
    }

    // ---- causal mask + online softmax update ----
    const int qrow0 = qt * AT_QT + w * 16 + lg * 4;  // + r
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
#pragma unroll
    for (int r = 0; r < 4; ++r) {
#pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        tile_max[r] = fmaxf(tile_max[r], __shfl_xor(tile_max[r], off, KF_WAVE));
    }
    float alpha[4], psum[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const float m_new = fmaxf(m_run[r], tile_max[r]);
      alpha[r] = (m_run[r] == -INFINITY) ? 0.f : __expf(m_run[r] - m_new);
      m_run[r] = m_new;
      psum[r] = 0.f;
    }
    // P = exp(S - m), accumulate row sums, write P to this wave's LDS buf
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float pv = (sacc[nt][r] == -INFINITY)
                       ? 0.f
                       : __expf(sacc[nt][r] - m_run[r]);
        sacc[nt][r] = pv;
        psum[r] += pv;
        const int prow = lg * 4 + r;
        *reinterpret_cast<unsigned short*>(
            p_lds[w] + kf_swz(prow, (nt * 16 + l16) * 2, AT_KT * 2)) =
            kf_f32_to_bf16(pv);
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
#pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        psum[r] += __shfl_xor(psum[r], off, KF_WAVE);
      l_run[r] = l_run[r] * alpha[r] + psum[r];
    }
    // rescale O by alpha
#pragma unroll
    for (int nt = 0; nt < 8; ++nt)
#pragma unroll
      for (int r = 0; r < 4; ++r) oacc[nt][r] *= alpha[r];

    // wave-local LDS ordering: P writes above are read below by the SAME
    // wave only; s_waitcnt lgkmcnt is compiler-inserted on the dependence,
    // but the addresses differ per lane — use a wave-visible fence.
    __builtin_amdgcn_s_waitcnt(0);  // drain lgkm for this wave

    // ---- O += P V  (8 d-tiles × K-loop over 64 kv in 2 steps) ----
    kf_bf16x8 pfrag[2];
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
      pfrag[kk] = *reinterpret_cast<const kf_bf16x8*>(
          p_lds[w] + kf_swz(l16, kk * 64 + lg * 16, AT_KT * 2));
#pragma unroll
    for (int nt = 0; nt < 8; ++nt) {
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        kf_bf16x8 vfrag = *reinterpret_cast<const kf_bf16x8*>(
            vt_lds + kf_swz(nt * 16 + l16, kk * 64 + lg * 16, AT_KT * 2));
        oacc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pfrag[kk], vfrag,
                                                           oacc[nt], 0, 0, 0);
      }
    }
  }

  // ---- epilogue: O /= l, write O (bf16) and lse (fp32) ----
  const int qrow0 = qt * AT_QT + w * 16 + lg * 4;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const float inv_l = (l_run[r] > 0.f) ? 1.f / l_run[r] : 0.f;
    const int64_t obase = ((b * S + qrow0 + r) * (int64_t)Hq + hq) * AT_D;
#pragma unroll
    for (int nt = 0; nt < 8; ++nt)
      o[obase + nt * 16 + l16] = kf_f32_to_bf16(oacc[nt][r] * inv_l);
    if (l16 == 0)
      lse[(b * Hq + hq) * (int64_t)S + qrow0 + r] =
          m_run[r] + __logf(l_run[r]);
  }
}

KF_EXPORT int kf_attn_fwd(void* o, float* lse, const void* q, const void* k,
                          const void* v, int64_t B, int64_t S, int64_t Hq,
                          int64_t Hkv, int64_t D, float scale, int causal,
                          void* stream) {
  if (D != AT_D || S % AT_QT || Hq % Hkv) return (int)hipErrorInvalidValue;
  dim3 grid((unsigned)(S / AT_QT), (unsigned)Hq, (unsigned)B);
  hipLaunchKernelGGL(kf_attn_fwd_kernel, grid, dim3(AT_THREADS), 0,
                     (hipStream_t)stream, (unsigned short*)o, lse,
                     (const unsigned short*)q, (const unsigned short*)k,
                     (const unsigned short*)v, B, (int)S, (int)Hq, (int)Hkv,
                     scale, causal);
  return (int)hipGetLastError();
}
