// decode_fused.hip — fused RoPE + KV-cache scatter for the decode step.
//
// Eager decode ran: qkv split -> rope wrapper (clone q, clone k, rope
// kernel) -> kv_store, i.e. ~4 kernels and two D2D copies per layer just
// to position-encode one token per sequence and file it in the cache
// (profiles/r02_decode_anatomy.md: non-GEMM time ~53 us/layer across ~11
// launches). This kernel consumes the fused QKV projection output
// directly: applies rotate-half RoPE (same convention as rope.hip) to
// the q and k segments, writes q to a contiguous [N, Hq, D] buffer for
// the decode-attention kernel, and scatters k (rotated) and v (copied)
// into cache row [slot, pos]. One launch, no intermediate copies.
//
// Grid (N, Hq + 2*Hkv): one 64-lane wave per (sequence, head); lane d
// owns the rotate-half pair (x[d], x[d + D/2]). D == 128.

#include "kf_common.h"

#define DF_D 128

__global__ __launch_bounds__(64) void kf_decode_rope_store_kernel(
    unsigned short* __restrict__ qout, unsigned short* __restrict__ ck,
    unsigned short* __restrict__ cv, const unsigned short* __restrict__ qkv,
    const float* __restrict__ cost, const float* __restrict__ sint,
    const int* __restrict__ slots, const long long* __restrict__ positions,
    int Hq, int Hkv, int64_t smax) {
  const int n = blockIdx.x, h = blockIdx.y;
  const int d = threadIdx.x;  // 0..63: rotate-half pair index
  const int halfD = DF_D / 2;
  const int Ht = Hq + 2 * Hkv;
  const long long pos = positions[n];
  const unsigned short* src = qkv + ((int64_t)n * Ht + h) * DF_D;

  if (h < Hq + Hkv) {  // q or k head: rotate
    const float c = cost[pos * halfD + d];
    const float s = sint[pos * halfD + d];
    const float a = kf_bf16_to_f32(src[d]);
    const float b = kf_bf16_to_f32(src[d + halfD]);
    unsigned short* dst;
    if (h < Hq) {
      dst = qout + ((int64_t)n * Hq + h) * DF_D;
    } else {
      dst = ck + (((int64_t)slots[n] * smax + pos) * Hkv + (h - Hq)) * DF_D;
    }
    dst[d] = kf_f32_to_bf16(a * c - b * s);
    dst[d + halfD] = kf_f32_to_bf16(b * c + a * s);
  } else {  // v head: plain copy into the cache
    unsigned short* dst =
        cv + (((int64_t)slots[n] * smax + pos) * Hkv + (h - Hq - Hkv)) * DF_D;
    dst[d] = src[d];
    dst[d + halfD] = src[d + halfD];
  }
}

KF_EXPORT int kf_decode_rope_store(void* qout, void* ck, void* cv,
                                   const void* qkv, const float* cost,
                                   const float* sint, const int* slots,
                                   const void* positions, int64_t N,
                                   int64_t Hq, int64_t Hkv, int64_t D,
                                   int64_t smax, void* stream) {
  if (D != DF_D) return (int)hipErrorInvalidValue;
  dim3 grid((unsigned)N, (unsigned)(Hq + 2 * Hkv));
  hipLaunchKernelGGL(kf_decode_rope_store_kernel, grid, dim3(64), 0,
                     (hipStream_t)stream, (unsigned short*)qout,
                     (unsigned short*)ck, (unsigned short*)cv,
                     (const unsigned short*)qkv, cost, sint, slots,
                     (const long long*)positions, (int)Hq, (int)Hkv, smax);
  return (int)hipGetLastError();
}
