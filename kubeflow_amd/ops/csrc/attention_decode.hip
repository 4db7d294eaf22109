// attention_decode.hip — single-token GQA attention over the KV cache.
//
// The serving-side decode hot loop (SURVEY.md §2.13 "decode = GEMV-like"):
// memory-bound streaming of the KV cache. One block per (sequence, kv-head);
// the GQA group's q-heads (one per wave) share each staged K/V tile, so the
// cache is read ONCE per kv-head regardless of group size — the main
// bandwidth lever for GQA decode.
//
//   q      [N, Hq, 128] bf16   (one new token per active sequence)
//   kcache [SLOTS, SMAX, Hkv, 128] bf16 (same layout for vcache)
//   slots  [N] int32 cache slot per sequence; lens [N] tokens to attend over
//   out    [N, Hq, 128] bf16
//
// Per 64-row cache tile: all waves cooperatively stage K,V into swizzled
// LDS; each wave owns one q-head of the group: lane r computes dot(q, K[r])
// (one cache row per lane), online-softmax in wave registers, then lanes
// switch to owning 2 d-elements each for the PV accumulation. Waves beyond
// the group size still participate in staging/barriers (no divergent
// __syncthreads). Heads beyond nwaves loop in outer chunks (extra cache
// passes — does not occur for the Llama-3 g=4 config).

#include "kf_common.h"

#define AD_D 128
#define AD_TILE 64

// swizzle for [64 rows][256B] LDS tiles: byte ^= ((row&7)<<4) (guide §6 G4)
__device__ __forceinline__ int kf_swzd(int row, int byte_in_row) {
  return row * (AD_D * 2) + (byte_in_row ^ ((row & 7) << 4));
}

// Grid (N, Hkv, SPLITS): flash-decoding sequence split. N*Hkv blocks
// alone (16x8 = 128 at the serving bucket) fill only half the chip and
// serialize long contexts; each grid.z block handles a contiguous
// [lo, hi) slice of the sequence and (for splits > 1) writes an
// UNNORMALIZED partial (o, m, l) that kf_attn_decode_combine merges
// with the online-softmax rules.
// VDIRECT: read V straight from global in the PV phase (each row-group
// instruction covers 256 contiguous bytes of one cache row, and the 4
// waves' repeat reads hit the XCD's L2) instead of staging it — halves
// the LDS per block, doubling resident blocks per CU. Measured per
// context length; host picks.
template <bool VDIRECT>
__global__ __launch_bounds__(256) void kf_attn_decode_kernel(
    unsigned short* __restrict__ out, float* __restrict__ part_o,
    float* __restrict__ part_ml, const unsigned short* __restrict__ q,
    const unsigned short* __restrict__ kcache,
    const unsigned short* __restrict__ vcache,
    const int* __restrict__ slots, const int* __restrict__ lens,
    int64_t smax, int Hq, int Hkv, float scale) {
  __shared__ unsigned char k_lds[AD_TILE * AD_D * 2];
  __shared__ unsigned char v_lds[VDIRECT ? 1 : AD_TILE * AD_D * 2];
  __shared__ float p_lds[4][AD_TILE];

  const int n = blockIdx.x, hkv = blockIdx.y;
  const int z = blockIdx.z, splits = gridDim.z;
  const int g = Hq / Hkv;
  const int nw = blockDim.x / KF_WAVE;
  const int w = threadIdx.x / KF_WAVE;
  const int lane = threadIdx.x & (KF_WAVE - 1);
  const int len = lens[n];
  const int64_t slot = slots[n];
  const int64_t cbase = (slot * smax * Hkv + hkv) * AD_D;
  const int64_t cstride = (int64_t)Hkv * AD_D;
  // this split's row slice, tile-aligned so staging stays coalesced
  const int chunk =
      ((len + splits - 1) / splits + AD_TILE - 1) / AD_TILE * AD_TILE;
  const int lo = z * chunk, hi = min(len, lo + chunk);

  for (int hbase = 0; hbase < g; hbase += nw) {
    const int hg = hbase + w;
    const bool active = hg < g;
    const int hq = hkv * g + (active ? hg : 0);
    const unsigned short* qrow = q + ((int64_t)n * Hq + hq) * AD_D;

    float qreg[AD_D / 8][8];
#pragma unroll
    for (int i = 0; i < AD_D / 8; ++i) {
      kf_short8 qv = *reinterpret_cast<const kf_short8*>(qrow + i * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        qreg[i][j] = kf_bf16_to_f32((unsigned short)qv[j]);
    }
    float m_run = -INFINITY, l_run = 0.f;
    // PV accumulation is 4-row-group split for b128 V reads: lane l owns
    // d-elems [(l&15)*8, +8) for rows r ≡ (l>>4) (mod 4); the 4 groups'
    // partial o vectors reduce through LDS once per sequence slice.
    const int rg = lane >> 4;          // this lane's row group
    const int dcol = (lane & 15) * 8;  // this lane's 8-elem d-range
    float ov[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};

    for (int t0 = lo; t0 < hi; t0 += AD_TILE) {
      const int rows = min(AD_TILE, hi - t0);
      __syncthreads();
      if (rows == AD_TILE) {
        // full tile: register-buffered unrolled staging (8 loads in
        // flight per thread before any LDS write — the dynamic loop's
        // load->write->load chain left staging HBM-latency-bound)
        kf_short8 kb[4], vb[4];
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          const int vi = (int)threadIdx.x + u * 256;
          const int r = vi >> 4, c8 = vi & 15;
          kb[u] = *reinterpret_cast<const kf_short8*>(
              kcache + cbase + (int64_t)(t0 + r) * cstride + c8 * 8);
          if (!VDIRECT)
            vb[u] = *reinterpret_cast<const kf_short8*>(
                vcache + cbase + (int64_t)(t0 + r) * cstride + c8 * 8);
        }
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          const int vi = (int)threadIdx.x + u * 256;
          const int r = vi >> 4, c8 = vi & 15;
          *reinterpret_cast<kf_short8*>(k_lds + kf_swzd(r, c8 * 16)) = kb[u];
          if (!VDIRECT)
            *reinterpret_cast<kf_short8*>(v_lds + kf_swzd(r, c8 * 16)) =
                vb[u];
        }
      } else {
        for (int vi = threadIdx.x; vi < rows * 16; vi += blockDim.x) {
          const int r = vi >> 4, c8 = vi & 15;
          kf_short8 kv8 = *reinterpret_cast<const kf_short8*>(
              kcache + cbase + (int64_t)(t0 + r) * cstride + c8 * 8);
          *reinterpret_cast<kf_short8*>(k_lds + kf_swzd(r, c8 * 16)) = kv8;
          if (!VDIRECT) {
            kf_short8 vv8 = *reinterpret_cast<const kf_short8*>(
                vcache + cbase + (int64_t)(t0 + r) * cstride + c8 * 8);
            *reinterpret_cast<kf_short8*>(v_lds + kf_swzd(r, c8 * 16)) =
                vv8;
          }
        }
      }
      __syncthreads();

      float s = -INFINITY;
      if (lane < rows) {
        float acc = 0.f;
#pragma unroll
        for (int i = 0; i < AD_D / 8; ++i) {
          kf_short8 kv8 = *reinterpret_cast<const kf_short8*>(
              k_lds + kf_swzd(lane, i * 16));
#pragma unroll
          for (int j = 0; j < 8; ++j)
            acc += qreg[i][j] * kf_bf16_to_f32((unsigned short)kv8[j]);
        }
        s = acc * scale;
      }
      const float tile_max = kf_wave_max(s);
      const float m_new = fmaxf(m_run, tile_max);
      const float alpha = (m_run == -INFINITY) ? 0.f : __expf(m_run - m_new);
      const float p = (s == -INFINITY) ? 0.f : __expf(s - m_new);
      m_run = m_new;
      l_run = l_run * alpha + kf_wave_sum(p);
#pragma unroll
      for (int e = 0; e < 8; ++e) ov[e] *= alpha;
      p_lds[w][lane] = p;
      __builtin_amdgcn_s_waitcnt(0);  // wave-local p_lds write->read fence
      // 16 b128 V reads per tile per lane (was 64 b32): rows 4i + rg
      for (int i = 0; i < AD_TILE / 4; ++i) {
        const int r = 4 * i + rg;
        if (r >= rows) break;
        const float pr = p_lds[w][r];
        kf_short8 vv8;
        if (VDIRECT)
          vv8 = *reinterpret_cast<const kf_short8*>(
              vcache + cbase + (int64_t)(t0 + r) * cstride + dcol);
        else
          vv8 = *reinterpret_cast<const kf_short8*>(
              v_lds + kf_swzd(r, dcol * 2));
#pragma unroll
        for (int e = 0; e < 8; ++e)
          ov[e] += pr * kf_bf16_to_f32((unsigned short)vv8[e]);
      }
    }
    // fold the 4 row groups: lanes {l15, l15+16, l15+32, l15+48} hold
    // partial o for the same d-range. Exchange via this wave's p_lds row
    // (wave-local scratch; 3 rounds of 8 floats each).
    float osum[8];
#pragma unroll
    for (int e = 0; e < 8; ++e) osum[e] = ov[e];
    for (int src = 1; src < 4; ++src) {
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        p_lds[w][lane] = ov[e];
        __builtin_amdgcn_s_waitcnt(0);
        osum[e] += p_lds[w][(lane + 16 * src) & 63];
        __builtin_amdgcn_s_waitcnt(0);
      }
    }
    if (active && rg == 0) {
      // lane l15 writes d-range [dcol, dcol+8)
      if (splits == 1) {
        const float inv_l = l_run > 0.f ? 1.f / l_run : 0.f;
        unsigned short* orow = out + ((int64_t)n * Hq + hq) * AD_D + dcol;
#pragma unroll
        for (int e = 0; e < 8; ++e)
          orow[e] = kf_f32_to_bf16(osum[e] * inv_l);
      } else {
        const int64_t pb = ((int64_t)n * Hq + hq) * splits + z;
        float* po = part_o + pb * AD_D + dcol;
#pragma unroll
        for (int e = 0; e < 8; ++e) po[e] = osum[e];
        if (lane == 0) {
          part_ml[pb * 2] = m_run;
          part_ml[pb * 2 + 1] = l_run;
        }
      }
    }
  }
}

// merge the split partials: one 64-lane wave per (n, hq) row
__global__ __launch_bounds__(64) void kf_attn_decode_combine_kernel(
    unsigned short* __restrict__ out, const float* __restrict__ part_o,
    const float* __restrict__ part_ml, int splits) {
  const int64_t row = blockIdx.x;  // n*Hq + hq
  const int lane = threadIdx.x;
  float M = -INFINITY;
  for (int z = 0; z < splits; ++z)
    M = fmaxf(M, part_ml[(row * splits + z) * 2]);
  float L = 0.f, o0 = 0.f, o1 = 0.f;
  for (int z = 0; z < splits; ++z) {
    const float mz = part_ml[(row * splits + z) * 2];
    if (mz == -INFINITY) continue;
    const float wz = __expf(mz - M);
    L += part_ml[(row * splits + z) * 2 + 1] * wz;
    const float* po = part_o + (row * splits + z) * AD_D;
    o0 += po[lane * 2] * wz;
    o1 += po[lane * 2 + 1] * wz;
  }
  const float inv_l = L > 0.f ? 1.f / L : 0.f;
  out[row * AD_D + lane * 2] = kf_f32_to_bf16(o0 * inv_l);
  out[row * AD_D + lane * 2 + 1] = kf_f32_to_bf16(o1 * inv_l);
}

// splits > 1 requires part_o [N,Hq,splits,128] f32 and part_ml
// [N,Hq,splits,2] f32 workspaces; pass splits = 0 to let the host pick
// enough grid.z to fill the chip (2 blocks/CU).
KF_EXPORT int kf_attn_decode(void* out, void* part_o, void* part_ml,
                             const void* q, const void* kcache,
                             const void* vcache, const int* slots,
                             const int* lens, int64_t N, int64_t smax,
                             int64_t Hq, int64_t Hkv, int64_t D,
                             int64_t splits, float scale, void* stream) {
  if (D != AD_D || Hq % Hkv) return (int)hipErrorInvalidValue;
  if (splits < 1) return (int)hipErrorInvalidValue;
  if (splits > 1 && (!part_o || !part_ml)) return (int)hipErrorInvalidValue;
  dim3 grid((unsigned)N, (unsigned)Hkv, (unsigned)splits);
  const char* vd = getenv("KF_DECODE_VDIRECT");
  if (vd && vd[0] == '1')
    hipLaunchKernelGGL(kf_attn_decode_kernel<true>, grid, dim3(256), 0,
                       (hipStream_t)stream, (unsigned short*)out,
                       (float*)part_o, (float*)part_ml,
                       (const unsigned short*)q,
                       (const unsigned short*)kcache,
                       (const unsigned short*)vcache, slots, lens, smax,
                       (int)Hq, (int)Hkv, scale);
  else
    hipLaunchKernelGGL(kf_attn_decode_kernel<false>, grid, dim3(256), 0,
                       (hipStream_t)stream, (unsigned short*)out,
                       (float*)part_o, (float*)part_ml,
                       (const unsigned short*)q,
                       (const unsigned short*)kcache,
                       (const unsigned short*)vcache, slots, lens, smax,
                       (int)Hq, (int)Hkv, scale);
  if (splits > 1)
    hipLaunchKernelGGL(kf_attn_decode_combine_kernel,
                       dim3((unsigned)(N * Hq)), dim3(64), 0,
                       (hipStream_t)stream, (unsigned short*)out,
                       (const float*)part_o, (const float*)part_ml,
                       (int)splits);
  return (int)hipGetLastError();
}
