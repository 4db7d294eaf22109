// adamw.hip — fused decoupled-AdamW over one flat parameter shard, CDNA4.
//
// SURVEY.md §2.13 "fused_adamw kernel: flat 1D grid over param shards,
// master fp32". The trainer keeps ALL model parameters as views into a
// single contiguous bf16 buffer (288 GB HBM3E: big flat allocations, one
// launch per step — MI355X-first), with fp32 master weights and moments in
// matching flat buffers. One kernel updates everything:
//
//   m = b1*m + (1-b1)*g;  v = b2*v + (1-b2)*g^2
//   p32 = p32*(1 - lr*wd) - lr * (m/bc1) / (sqrt(v/bc2) + eps)
//   p_bf16 = round(p32)
//
// Bias corrections are precomputed on host (no pow on device). Weight decay
// masking (no decay for norms/embeddings biases) is handled by the caller via
// a per-element fp32 `wd_mask` {0,1} buffer — optional (null => decay all).
// Memory-bound: 28-32 B/element — everything vectorized float4/short4.

#include "kf_common.h"

__global__ void kf_adamw_kernel(unsigned short* __restrict__ p16,
                                float* __restrict__ p32,
                                const unsigned short* __restrict__ g,
                                float* __restrict__ m, float* __restrict__ v,
                                const float* __restrict__ wd_mask, int64_t n4,
                                float lr, float b1, float b2, float eps,
                                float wd, float inv_bc1, float inv_bc2) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n4;
       i += gridDim.x * (int64_t)blockDim.x) {
    kf_short4 gv = *reinterpret_cast<const kf_short4*>(g + i * 4);
    kf_float4 pv = *reinterpret_cast<const kf_float4*>(p32 + i * 4);
    kf_float4 mv = *reinterpret_cast<const kf_float4*>(m + i * 4);
    kf_float4 vv = *reinterpret_cast<const kf_float4*>(v + i * 4);
    kf_float4 wdv;
    if (wd_mask) wdv = *reinterpret_cast<const kf_float4*>(wd_mask + i * 4);
    kf_short4 out16;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float gf = kf_bf16_to_f32((unsigned short)gv[j]);
      float mf = b1 * mv[j] + (1.f - b1) * gf;
      float vf = b2 * vv[j] + (1.f - b2) * gf * gf;
      float decay = wd_mask ? wdv[j] * wd : wd;
      float pf = pv[j] * (1.f - lr * decay);
      pf -= lr * (mf * inv_bc1) * __frcp_rn(sqrtf(vf * inv_bc2) + eps);
      mv[j] = mf; vv[j] = vf; pv[j] = pf;
      out16[j] = (short)kf_f32_to_bf16(pf);
    }
    *reinterpret_cast<kf_float4*>(m + i * 4) = mv;
    *reinterpret_cast<kf_float4*>(v + i * 4) = vv;
    *reinterpret_cast<kf_float4*>(p32 + i * 4) = pv;
    *reinterpret_cast<kf_short4*>(p16 + i * 4) = out16;
  }
}

KF_EXPORT int kf_adamw(void* p16, float* p32, const void* g, float* m,
                       float* v, const float* wd_mask, int64_t n, float lr,
                       float b1, float b2, float eps, float wd, int64_t step,
                       void* stream) {
  if (n % 4) return (int)hipErrorInvalidValue;
  const float inv_bc1 = 1.f / (1.f - powf(b1, (float)step));
  const float inv_bc2 = 1.f / (1.f - powf(b2, (float)step));
  hipLaunchKernelGGL(kf_adamw_kernel, dim3(kf_grid_for(n / 4, 256)), dim3(256),
                     0, (hipStream_t)stream, (unsigned short*)p16, p32,
                     (const unsigned short*)g, m, v, wd_mask, n / 4, lr, b1,
                     b2, eps, wd, inv_bc1, inv_bc2);
  return (int)hipGetLastError();
}
