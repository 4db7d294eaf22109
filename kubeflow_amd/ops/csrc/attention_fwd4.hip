// attention_fwd4.hip — v4 causal flash-attention forward (bf16, GQA, gfx950).
//
// Round-2 rework of the 8-wave swapped-QK^T kernel (attention_fwd.hip v3,
// 296 TF causal) applying the CDNA4 guide's full verified technique stack
// (§B 8-warp ladder, §5.5 T3/T10/T13/T14) plus a VALU-overlap pipeline:
//  * KVBLK=64 K/V tiles DOUBLE-BUFFERED in LDS (64 KiB total), with the
//    async-STAGE split: next tile's global loads issue at the top of the
//    iteration and land during compute; LDS writes + one barrier per tile
//    (T3 minimum 2-phase recipe + T14) — replaces the serial
//    sync/load/scatter/sync staging of v3.
//  * V is stored in a [d/16-slab][kv/4-tile][4][16] subtiled layout and the
//    PV A-operand (V^T) is read with ds_read_b64_tr_b16, the gfx950
//    hardware transpose read (T10) — replaces v3's 8-scalar-ds_write
//    transpose-scatter per staged vector. Lane mapping HW-verified by
//    scripts/probe_tr16.py (each lane reads 8 B at its own address; each
//    4-lane subgroup supplies one 16-elem tile row; lane gets column l&15).
//  * both 32-kv sub-blocks' QK^T MFMA clusters issue back-to-back so the
//    first block's softmax VALU overlaps the second's MFMA tail; PV
//    tr_reads are 1-deep prefetched with counted lgkmcnt.
//  * softmax runs in the exp2 domain (one v_exp_f32 per element, scale and
//    log2e folded into one multiply), with defer-max (T13, THR=8/ln2) and
//    causal masking only on diagonal sub-blocks (wave-uniform template
//    branch); fully-masked sub-blocks are skipped per wave.
//  * swapped QK^T (S^T = mfma(K, Q)) with fully in-register softmax and
//    cvt_pk_bf16 + permlane32_swap P repack, K XOR-swizzled LDS, setprio
//    around MFMA clusters — carried over from v3.
//
// Layout: bshd q [B,S,Hq,D], k/v [B,S,Hkv,D], o [B,S,Hq,D], lse [B,Hq,S]
// fp32. D == 128, S % 256 == 0 (wrapper pads). Reference behavior anchor:
// SURVEY.md §2.13 attention_fwd row.

#include "kf_common.h"

typedef __bf16 kf_bf16x8v4 __attribute__((ext_vector_type(8)));
typedef float kf_f32x16v4 __attribute__((ext_vector_type(16)));
typedef short kf_short4v4 __attribute__((ext_vector_type(4)));

#define A4_D 128
#define A4_QT 256      // q rows per block (8 waves x 32)
// kv rows per LDS tile (double-buffered). 64 -> 64 KiB LDS; 128 -> 128 KiB
// (still 1 block/CU at this register budget) with HALF the barriers.
// (128 measured 656 TF vs 64's 674 on B4 Hq32 S4096 causal — the halved
// barrier count does not pay for the deeper staging; keep 64)
#ifndef A4_KT
#define A4_KT 64
#endif
#define A4_SLAB (A4_KT / 4 * 64)   // elems per d16-slab in the V layout
#define A4_THREADS 512
#define A4_LOG2E 1.44269504f
#define A4_LN2 0.69314718f
#define A4_RESCALE_THR2 11.5415603f  // 8 * log2(e): defer-max in exp2 units

// row-major K tile swizzle (guide G4): byte ^= (row&7)<<4
__device__ __forceinline__ int kf_swz4(int row, int byte_in_row) {
  return row * (A4_D * 2) + (byte_in_row ^ ((row & 7) << 4));
}

// V subtile layout for ds_read_b64_tr_b16 (T10): element offset of V[kv][d]
// = slab(d>>4)*1024 + (kv>>2)*64 + (kv&3)*16 + (d&15). Each [4][16] subtile
// is 128 contiguous bytes; a 16-lane group's tr_read covers one subtile and
// delivers column (lane&15): lane l receives V^T[d=base_d+(l&15)][kv=base+j].
__device__ __forceinline__ int kf_vsub4(int kv, int d) {
  return (d >> 4) * A4_SLAB + ((kv >> 2) << 6) + ((kv & 3) << 4) + (d & 15);
}

__device__ __forceinline__ unsigned int kf_cvt_pk_bf16_v4(float lo, float hi) {
  unsigned int r;
  asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}

__device__ __forceinline__ float kf_exp2(float x) {
  float r;
  asm volatile("v_exp_f32 %0, %1" : "=v"(r) : "v"(x));
  return r;
}

#define KF_TR16(dst, addr, OFFLIT)                                      \
  asm volatile("ds_read_b64_tr_b16 %0, %1 offset:" OFFLIT               \
               : "=v"(dst) : "v"(addr))

// ABL ablation bits (guide m164 methodology — ablate before optimizing):
// 1 = skip softmax math (pb straight from scaled st), 2 = skip PV MFMA
// (pb kept live), 4 = skip next-tile staging (compute on stale LDS).
template <int ABL>
__global__ __launch_bounds__(A4_THREADS, 2) void kf_attn_fwd4_kernel(
    unsigned short* __restrict__ o, float* __restrict__ lse,
    const unsigned short* __restrict__ q, const unsigned short* __restrict__ k,
    const unsigned short* __restrict__ v, int64_t B, int S, int Hq, int Hkv,
    int64_t qts, int64_t kts, float scale, int causal, int skv, int qoff) {
  // rectangular-causal extension (chunked prefill): q rows are globally
  // offset by qoff and attend kv in [0, min(qoff+qrow, skv-1)]; the plain
  // causal square is the skv=S, qoff=0 special case.
  __shared__ unsigned char k_lds[2][A4_KT * A4_D * 2];  // 16 KiB x2, swizzled
  __shared__ unsigned char v_lds[2][A4_KT * A4_D * 2];  // 16 KiB x2, subtiled

  // 1-D grid, qt-major DESCENDING: causal work grows with qt, so the
  // longest q-tiles dispatch first (LPT order)
  const int nqt = gridDim.x / (Hq * (int)B);
  const int qt = causal ? (nqt - 1 - blockIdx.x / (Hq * (int)B))
                        : (int)(blockIdx.x / (Hq * (int)B));
  const int rest = blockIdx.x % (Hq * (int)B);
  const int hq = rest % Hq;
  const int64_t b = rest / Hq;
  const int hkv = hq / (Hq / Hkv);
  const int tid = threadIdx.x;
  const int w = tid / KF_WAVE;
  const int lane = tid & (KF_WAVE - 1);
  const int l31 = lane & 31;
  const int hi = lane >> 5;
  const float scale2 = scale * A4_LOG2E;  // QK^T scaled into exp2 domain

  // ---- persistent Q B-fragments: lane holds q-col l31 ----
  kf_bf16x8v4 qfrag[8];
  {
    const int64_t qbase =
        (b * S + qt * A4_QT + w * 32 + l31) * qts + (int64_t)hq * A4_D;
#pragma unroll
    for (int kk = 0; kk < 8; ++kk)
      qfrag[kk] =
          *reinterpret_cast<const kf_bf16x8v4*>(q + qbase + kk * 16 + hi * 8);
  }

  kf_f32x16v4 oacc[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) oacc[i] = kf_f32x16v4{0.f};
  float m_run = -INFINITY, l_run = 0.f;  // m_run in exp2 units
  const int qrow_g = qt * A4_QT + w * 32 + l31;
  const int wave_qmin = qt * A4_QT + w * 32;
  const int wave_qmax = wave_qmin + 31;
  // lane's highest visible kv row (pad q rows clamp to skv-1)
  const int kv_limit = causal ? min(qrow_g + qoff, skv - 1) : skv - 1;

  const int kv_tiles = (skv + A4_KT - 1) / A4_KT;
  int last_kt = kv_tiles - 1;
  if (causal) {
    const int c = (qoff + qt * A4_QT + A4_QT - 1) / A4_KT;
    if (c < last_kt) last_kt = c;
  }

  // staging: 512 threads x 2 vectors cover one 64x128 tile for K and V.
  const unsigned short* kg0 = k + (b * S) * kts + (int64_t)hkv * A4_D;
  const unsigned short* vg0 = v + (b * S) * kts + (int64_t)hkv * A4_D;
  const int c8 = tid & 15;
  int srow[A4_KT / 32];                            // staged rows per thread
#pragma unroll
  for (int j = 0; j < A4_KT / 32; ++j)
    srow[j] = (tid + A4_THREADS * j) >> 4;

  // per-lane tr_read base byte address inside a V buffer: group g covers
  // slab (g&1), kv-subtile 2*(g>>1), column lane&15 (see kf_vsub4).
  const int g = lane >> 4;
  const unsigned v_lane_off =
      (unsigned)(((g & 1) * A4_SLAB * 2) + ((g >> 1) << 8) +
                 ((lane & 15) << 3));

  // ---- S^T = mfma(K, Q) over one 32-kv sub-block. All 8 K-fragment
  // ds_reads issue back-to-back BEFORE the MFMA chain so the ~120-cycle
  // LDS latency pipelines instead of serializing per MFMA (the same
  // issue-all-then-compute discipline as the guide's GEMM ladder). ----
  auto qk_block = [&](int cur, int mt) {
    kf_bf16x8v4 af[8];
#pragma unroll
    for (int kk = 0; kk < 8; ++kk)
      af[kk] = *reinterpret_cast<const kf_bf16x8v4*>(
          k_lds[cur] + kf_swz4(mt * 32 + l31, kk * 32 + hi * 16));
    kf_f32x16v4 st = kf_f32x16v4{0.f};
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 8; ++kk)
      st = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af[kk], qfrag[kk], st,
                                                   0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    return st;
  };

  // ---- defer-max online softmax (exp2 domain) + P repack (T12/T13) ----
  auto sm_block = [&](kf_f32x16v4& st, int kv_lo, bool need_mask,
                      kf_bf16x8v4* pb) {
    if (ABL & 1) {  // NOSM: scale-only, straight to repack (keeps st live)
#pragma unroll
      for (int r = 0; r < 16; ++r) st[r] *= scale2;
      l_run += st[0];
#pragma unroll
      for (int step = 0; step < 2; ++step) {
        const int base = step * 8;
        unsigned int w0 = kf_cvt_pk_bf16_v4(st[base + 0], st[base + 1]);
        unsigned int w1 = kf_cvt_pk_bf16_v4(st[base + 2], st[base + 3]);
        unsigned int w2 = kf_cvt_pk_bf16_v4(st[base + 4], st[base + 5]);
        unsigned int w3 = kf_cvt_pk_bf16_v4(st[base + 6], st[base + 7]);
        auto s02 = __builtin_amdgcn_permlane32_swap(w0, w2, false, false);
        auto s13 = __builtin_amdgcn_permlane32_swap(w1, w3, false, false);
        unsigned int u[4] = {(unsigned)s02[0], (unsigned)s13[0],
                             (unsigned)s02[1], (unsigned)s13[1]};
        pb[step] = *reinterpret_cast<kf_bf16x8v4*>(u);
      }
      return;
    }
    const int kv0 = kv_lo + hi * 4;
    float mx = -INFINITY;
    if (need_mask) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int kv = kv0 + (r & 3) + 8 * (r >> 2);
        float sv = st[r] * scale2;
        if (kv > kv_limit) sv = -INFINITY;
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
    if (!__all(mx <= m_run + A4_RESCALE_THR2)) {
      const float m_new = fmaxf(m_run, mx);
      const float alpha = kf_exp2(m_run - m_new);  // 0 on first block
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int r = 0; r < 16; ++r) oacc[i][r] *= alpha;
      l_run *= alpha;
      m_run = m_new;
    }
    float lsum = 0.f;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const float pv = kf_exp2(st[r] - m_run);  // masked -inf -> 0
      st[r] = pv;
      lsum += pv;
    }
    lsum += __shfl_xor(lsum, 32, KF_WAVE);
    l_run += lsum;
#pragma unroll
    for (int step = 0; step < 2; ++step) {
      const int base = step * 8;
      unsigned int w0 = kf_cvt_pk_bf16_v4(st[base + 0], st[base + 1]);
      unsigned int w1 = kf_cvt_pk_bf16_v4(st[base + 2], st[base + 3]);
      unsigned int w2 = kf_cvt_pk_bf16_v4(st[base + 4], st[base + 5]);
      unsigned int w3 = kf_cvt_pk_bf16_v4(st[base + 6], st[base + 7]);
      auto s02 = __builtin_amdgcn_permlane32_swap(w0, w2, false, false);
      auto s13 = __builtin_amdgcn_permlane32_swap(w1, w3, false, false);
      unsigned int u[4] = {(unsigned)s02[0], (unsigned)s13[0],
                           (unsigned)s02[1], (unsigned)s13[1]};
      pb[step] = *reinterpret_cast<kf_bf16x8v4*>(u);
    }
  };

  // ---- O^T += V^T P^T with 1-deep tr_read prefetch ----
  auto pv_block = [&](const kf_bf16x8v4* pb, unsigned vbase) {
    if (ABL & 2) {  // NOPV: keep pb live, skip tr_reads + MFMAs
      asm volatile("" ::"v"(*(const unsigned*)&pb[0]),
                   "v"(*(const unsigned*)&pb[1]), "v"(vbase));
      oacc[0][0] += (float)pb[0][0];
      return;
    }
    kf_short4v4 t[2][4];
    KF_TR16(t[0][0], vbase, "0");
    KF_TR16(t[0][1], vbase, "128");
    KF_TR16(t[0][2], vbase, "512");
    KF_TR16(t[0][3], vbase, "640");
#pragma unroll
    for (int dt = 0; dt < 4; ++dt) {
      if (dt < 3) {
        const unsigned va = vbase + (unsigned)((dt + 1) * A4_SLAB * 4);
        KF_TR16(t[(dt + 1) & 1][0], va, "0");
        KF_TR16(t[(dt + 1) & 1][1], va, "128");
        KF_TR16(t[(dt + 1) & 1][2], va, "512");
        KF_TR16(t[(dt + 1) & 1][3], va, "640");
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
      oacc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
          *reinterpret_cast<kf_bf16x8v4*>(&f0), pb[0], oacc[dt], 0, 0, 0);
      oacc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
          *reinterpret_cast<kf_bf16x8v4*>(&f1), pb[1], oacc[dt], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
  };

  // ---- prologue: stage tile 0 into buffer 0 ----
  kf_short8 kst[A4_KT / 32], vst[A4_KT / 32];
#pragma unroll
  for (int j = 0; j < A4_KT / 32; ++j) {
    kst[j] = *reinterpret_cast<const kf_short8*>(kg0 + srow[j] * kts + c8 * 8);
    vst[j] = *reinterpret_cast<const kf_short8*>(vg0 + srow[j] * kts + c8 * 8);
  }
#pragma unroll
  for (int j = 0; j < A4_KT / 32; ++j) {
    *reinterpret_cast<kf_short8*>(k_lds[0] + kf_swz4(srow[j], c8 * 16)) =
        kst[j];
    *reinterpret_cast<kf_short8*>(v_lds[0] + 2 * kf_vsub4(srow[j], c8 * 8)) =
        vst[j];
  }
  __syncthreads();

  for (int kt = 0; kt <= last_kt; ++kt) {
    const int cur = kt & 1;
    // ---- async-STAGE: issue next tile's global loads now; they land
    // under this tile's compute (T14) ----
    const bool have_next = (ABL & 4) ? false : kt < last_kt;
    if (have_next) {
      const unsigned short* kg = kg0 + (int64_t)(kt + 1) * A4_KT * kts;
      const unsigned short* vg = vg0 + (int64_t)(kt + 1) * A4_KT * kts;
#pragma unroll
      for (int j = 0; j < A4_KT / 32; ++j) {
        kst[j] =
            *reinterpret_cast<const kf_short8*>(kg + srow[j] * kts + c8 * 8);
        vst[j] =
            *reinterpret_cast<const kf_short8*>(vg + srow[j] * kts + c8 * 8);
      }
    }

    // ---- compute: per 64-kv pair, both QK^T clusters issue first so
    // sub-block A's softmax VALU overlaps sub-block B's MFMA tail ----
    const unsigned vbase = (unsigned)(size_t)(v_lds[cur]) + v_lane_off;
#pragma unroll
    for (int pair = 0; pair < A4_KT / 64; ++pair) {
      const int kv_lo0 = kt * A4_KT + pair * 64;
      const int kv_lo1 = kv_lo0 + 32;
      const bool do0 = (!causal || kv_lo0 <= wave_qmax + qoff) &&
                       kv_lo0 < skv;
      const bool do1 = (!causal || kv_lo1 <= wave_qmax + qoff) &&
                       kv_lo1 < skv;
      if (!do0 && !do1) continue;
      kf_f32x16v4 st0, st1;
      if (do0) st0 = qk_block(cur, pair * 2);
      if (do1) st1 = qk_block(cur, pair * 2 + 1);
      kf_bf16x8v4 pb[2];
      if (do0) {
        sm_block(st0, kv_lo0,
                 (causal && kv_lo0 + 31 > wave_qmin + qoff) ||
                     kv_lo0 + 31 >= skv, pb);
        pv_block(pb, vbase + (unsigned)(pair << 11));
      }
      // write the staged tile into the other buffer BETWEEN the two
      // sub-blocks: the vmcnt wait on the staged registers lands after
      // QK0/SM0/PV0 covered the HBM latency, and the ds_writes overlap
      // sub-block 1's MFMAs instead of sitting in the serial tail
      if (pair == 0 && have_next) {
        const int nxt = cur ^ 1;
#pragma unroll
        for (int j = 0; j < A4_KT / 32; ++j) {
          *reinterpret_cast<kf_short8*>(k_lds[nxt] +
                                        kf_swz4(srow[j], c8 * 16)) = kst[j];
          *reinterpret_cast<kf_short8*>(v_lds[nxt] +
                                        2 * kf_vsub4(srow[j], c8 * 8)) =
              vst[j];
        }
      }
      if (do1) {
        sm_block(st1, kv_lo1,
                 (causal && kv_lo1 + 31 > wave_qmin + qoff) ||
                     kv_lo1 + 31 >= skv, pb);
        pv_block(pb, vbase + (unsigned)(pair << 11) + 1024);
      }
    }
    // causal skip can bypass the pair-0 body entirely: make sure the
    // staged tile still lands before the barrier
    if (have_next && causal && kt * A4_KT > wave_qmax + qoff) {
      const int nxt = cur ^ 1;
#pragma unroll
      for (int j = 0; j < A4_KT / 32; ++j) {
        *reinterpret_cast<kf_short8*>(k_lds[nxt] + kf_swz4(srow[j], c8 * 16)) =
            kst[j];
        *reinterpret_cast<kf_short8*>(v_lds[nxt] +
                                      2 * kf_vsub4(srow[j], c8 * 8)) = vst[j];
      }
    }
    __syncthreads();  // readers of [cur] done + writes to [nxt] visible
  }

  // ---- epilogue: O^T regs -> o[token][d]; lse per q row ----
  const float inv_l = l_run > 0.f ? 1.f / l_run : 0.f;
  const int64_t obase = ((b * S + qrow_g) * (int64_t)Hq + hq) * A4_D;
#pragma unroll
  for (int dt = 0; dt < 4; ++dt)
#pragma unroll
    for (int rq = 0; rq < 4; ++rq) {  // 4 consecutive-d quads per dt
      const int d0 = dt * 32 + 8 * rq + 4 * hi;
      unsigned short q4[4];
#pragma unroll
      for (int j = 0; j < 4; ++j)
        q4[j] = kf_f32_to_bf16(oacc[dt][rq * 4 + j] * inv_l);
      *reinterpret_cast<kf_short4v4*>(o + obase + d0) =
          *reinterpret_cast<kf_short4v4*>(q4);
    }
  if (hi == 0)
    lse[(b * Hq + hq) * (int64_t)S + qrow_g] =
        m_run * A4_LN2 + __logf(l_run);
}

template <int ABL>
static int kf_attn_fwd4_launch(void* o, float* lse, const void* q,
                               const void* k, const void* v, int64_t B,
                               int64_t S, int64_t Hq, int64_t Hkv, int64_t D,
                               int64_t qts, int64_t kts, float scale,
                               int causal, void* stream) {
  if (D != A4_D || S % A4_QT || Hq % Hkv) return (int)hipErrorInvalidValue;
  if (qts == 0) qts = Hq * A4_D;
  if (kts == 0) kts = Hkv * A4_D;
  dim3 grid((unsigned)((S / A4_QT) * Hq * B), 1, 1);
  hipLaunchKernelGGL(kf_attn_fwd4_kernel<ABL>, grid, dim3(A4_THREADS), 0,
                     (hipStream_t)stream, (unsigned short*)o, lse,
                     (const unsigned short*)q, (const unsigned short*)k,
                     (const unsigned short*)v, B, (int)S, (int)Hq, (int)Hkv,
                     qts, kts, scale, causal, (int)S, 0);
  return (int)hipGetLastError();
}

// Rectangular-causal forward for chunked prefill: Sq query rows (padded to
// a 256 multiple) at global offset qoff attend skv cached kv rows. The
// kv buffers must extend to ceil(skv/64)*64 rows (the KV-cache slab does).
KF_EXPORT int kf_attn_fwd4_rect(void* o, float* lse, const void* q,
                                const void* k, const void* v, int64_t B,
                                int64_t Sq, int64_t Skv, int64_t Hq,
                                int64_t Hkv, int64_t D, int64_t qts,
                                int64_t kts, float scale, int64_t qoff,
                                void* stream) {
  if (D != A4_D || Sq % A4_QT || Hq % Hkv) return (int)hipErrorInvalidValue;
  if (qts == 0) qts = Hq * A4_D;
  if (kts == 0) kts = Hkv * A4_D;
  dim3 grid((unsigned)((Sq / A4_QT) * Hq * B), 1, 1);
  hipLaunchKernelGGL(kf_attn_fwd4_kernel<0>, grid, dim3(A4_THREADS), 0,
                     (hipStream_t)stream, (unsigned short*)o, lse,
                     (const unsigned short*)q, (const unsigned short*)k,
                     (const unsigned short*)v, B, (int)Sq, (int)Hq,
                     (int)Hkv, qts, kts, scale, 1, (int)Skv, (int)qoff);
  return (int)hipGetLastError();
}

KF_EXPORT int kf_attn_fwd4(void* o, float* lse, const void* q, const void* k,
                           const void* v, int64_t B, int64_t S, int64_t Hq,
                           int64_t Hkv, int64_t D, int64_t qts, int64_t kts,
                           float scale, int causal, void* stream) {
  return kf_attn_fwd4_launch<0>(o, lse, q, k, v, B, S, Hq, Hkv, D, qts, kts,
                                scale, causal, stream);
}

// Ablation entry: mode bits as above (scratch results are NOT valid
// attention outputs for mode != 0).
KF_EXPORT int kf_attn_fwd4_abl(int mode, void* o, float* lse, const void* q,
                               const void* k, const void* v, int64_t B,
                               int64_t S, int64_t Hq, int64_t Hkv, int64_t D,
                               int64_t qts, int64_t kts, float scale,
                               int causal, void* stream) {
  switch (mode) {
    case 1: return kf_attn_fwd4_launch<1>(o, lse, q, k, v, B, S, Hq, Hkv, D,
                                          qts, kts, scale, causal, stream);
    case 2: return kf_attn_fwd4_launch<2>(o, lse, q, k, v, B, S, Hq, Hkv, D,
                                          qts, kts, scale, causal, stream);
    case 3: return kf_attn_fwd4_launch<3>(o, lse, q, k, v, B, S, Hq, Hkv, D,
                                          qts, kts, scale, causal, stream);
    case 4: return kf_attn_fwd4_launch<4>(o, lse, q, k, v, B, S, Hq, Hkv, D,
                                          qts, kts, scale, causal, stream);
    default:
      return kf_attn_fwd4_launch<0>(o, lse, q, k, v, B, S, Hq, Hkv, D, qts,
                                    kts, scale, causal, stream);
  }
}

// ---------------------------------------------------------------------------
// ds_read_b64_tr_b16 semantics probe: fills LDS with element indices and
// dumps what each lane receives for three addressing modes. Used once on
// hardware to pin the lane->element mapping the v4 kernel assumes
// (guide T10, m156/m162); kept for regression if the layout ever changes.
// HW result (scripts/probe_tr16.py, MI355X): with addr_l = base + 8*l each
// lane receives column (l&15) of the [4][16] row-major tile covered by its
// 16-lane group's addresses; uniform addressing degenerates to lds[l&3].
// ---------------------------------------------------------------------------
__global__ void kf_tr16_probe_kernel(short* out, const short* in, int mode) {
  __shared__ short lds[1024];
  const int lane = threadIdx.x & 63;
  for (int i = threadIdx.x; i < 1024; i += 64) lds[i] = in[i];
  __syncthreads();
  unsigned base = (unsigned)(size_t)(&lds[0]);
  unsigned addr;
  if (mode == 0) addr = base + 8u * lane;          // 4 tiles, one per group
  else if (mode == 1) addr = base;                 // uniform
  else addr = base + 8u * (lane & 15);             // same tile all groups
  kf_short4v4 r0, r1;
  KF_TR16(r0, addr, "0");
  KF_TR16(r1, addr, "128");
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_sched_barrier(0);
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    out[lane * 8 + j] = r0[j];
    out[lane * 8 + 4 + j] = r1[j];
  }
}

KF_EXPORT int kf_tr16_probe(void* out, const void* in, int mode,
                            void* stream) {
  hipLaunchKernelGGL(kf_tr16_probe_kernel, dim3(1), dim3(64), 0,
                     (hipStream_t)stream, (short*)out, (const short*)in,
                     mode);
  return (int)hipGetLastError();
}
