// rope.hip — rotary position embedding, in-place over q and k, CDNA4.
//
// SURVEY.md §2.13 "rope kernel: grid (tokens×heads), vectorized". cos/sin are
// host-precomputed fp32 tables [S, D/2] (guide Appendix B: no device trig on
// the hot path). Layout is bshd: q [B,S,Hq,D], k [B,S,Hkv,D] bf16 contiguous.
// Pairing is rotate-half: (x[d], x[d + D/2]) for d < D/2.
//
// backward=1 applies the inverse rotation (the exact adjoint of forward),
// so the same kernel serves autograd backward on dq/dk.

#include "kf_common.h"

#define ROPE_VEC 4  // dim-pairs per thread

__global__ void kf_rope_kernel(unsigned short* __restrict__ q,
                               unsigned short* __restrict__ k,
                               const float* __restrict__ cost,
                               const float* __restrict__ sint,
                               const int64_t* __restrict__ positions,
                               int64_t B, int S, int Hq, int Hkv, int D,
                               int64_t qts, int64_t kts,  // token strides
                               int64_t pos_offset, int backward) {
  const int Ht = Hq + Hkv;
  const int halfD = D / 2;
  const int quads = halfD / ROPE_VEC;              // vec-chunks per head
  const int64_t total = B * (int64_t)S * Ht * quads;
  for (int64_t it = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; it < total;
       it += gridDim.x * (int64_t)blockDim.x) {
    const int qd = (int)(it % quads);
    int64_t rest = it / quads;
    const int h = (int)(rest % Ht);
    rest /= Ht;
    const int s = (int)(rest % S);
    const int64_t b = rest / S;
    unsigned short* base;
    if (h < Hq) base = q + (b * S + s) * qts + (int64_t)h * D;
    else base = k + (b * S + s) * kts + (int64_t)(h - Hq) * D;
    const int d0 = qd * ROPE_VEC;
    kf_short4 x1 = *reinterpret_cast<const kf_short4*>(base + d0);
    kf_short4 x2 = *reinterpret_cast<const kf_short4*>(base + halfD + d0);
    const int64_t pos = (positions ? positions[b] : pos_offset) + s;
    kf_float4 c = *reinterpret_cast<const kf_float4*>(
        cost + pos * (int64_t)halfD + d0);
    kf_float4 sn = *reinterpret_cast<const kf_float4*>(
        sint + pos * (int64_t)halfD + d0);
    kf_short4 o1, o2;
#pragma unroll
    for (int j = 0; j < ROPE_VEC; ++j) {
      float a = kf_bf16_to_f32((unsigned short)x1[j]);
      float bb = kf_bf16_to_f32((unsigned short)x2[j]);
      float sj = backward ? -sn[j] : sn[j];
      o1[j] = (short)kf_f32_to_bf16(a * c[j] - bb * sj);
      o2[j] = (short)kf_f32_to_bf16(bb * c[j] + a * sj);
    }
    *reinterpret_cast<kf_short4*>(base + d0) = o1;
    *reinterpret_cast<kf_short4*>(base + halfD + d0) = o2;
  }
}

KF_EXPORT int kf_rope(void* q, void* k, const float* cost, const float* sint,
                      const int64_t* positions, int64_t B, int64_t S,
                      int64_t Hq, int64_t Hkv, int64_t D, int64_t qts,
                      int64_t kts, int64_t pos_offset, int backward,
                      void* stream) {
  if ((D / 2) % ROPE_VEC) return (int)hipErrorInvalidValue;
  const int64_t total = B * S * (Hq + Hkv) * (D / 2 / ROPE_VEC);
  hipLaunchKernelGGL(kf_rope_kernel, dim3(kf_grid_for(total, 256)), dim3(256),
                     0, (hipStream_t)stream, (unsigned short*)q,
                     (unsigned short*)k, cost, sint, positions, B, (int)S,
                     (int)Hq, (int)Hkv, (int)D, qts, kts, pos_offset,
                     backward);
  return (int)hipGetLastError();
}
