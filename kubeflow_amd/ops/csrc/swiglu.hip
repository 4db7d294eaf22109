// swiglu.hip — fused SwiGLU activation for the Llama MLP, CDNA4.
//
//   fwd: h = [g ++ u] (one [T, 2F] tensor from the fused w13 GEMM)
//        y[t,f] = silu(g[t,f]) * u[t,f]
//   bwd: dh = [dg ++ du] written into ONE [T, 2F] buffer so the w13 GEMM
//        backward consumes it directly — no torch.cat of split-grads
//        (CatArrayBatchedCopy showed up at 2.2% of step time in the r01
//        profile, plus the silu/mul elementwise round-trips).
//
// Memory-bound: short8-vectorized, grid-stride (guide G13/G11).

#include "kf_common.h"

__device__ __forceinline__ float kf_sigmoid(float x) {
  return 1.f / (1.f + __expf(-x));
}

__global__ void kf_swiglu_fwd_kernel(unsigned short* __restrict__ y,
                                     const unsigned short* __restrict__ h,
                                     int64_t T, int64_t F) {
  const int64_t nvec = T * F / 8;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < nvec;
       i += gridDim.x * (int64_t)blockDim.x) {
    const int64_t t = (i * 8) / F;
    const int64_t f = (i * 8) % F;
    kf_short8 gv = *reinterpret_cast<const kf_short8*>(h + t * 2 * F + f);
    kf_short8 uv = *reinterpret_cast<const kf_short8*>(h + t * 2 * F + F + f);
    kf_short8 ov;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float g = kf_bf16_to_f32((unsigned short)gv[j]);
      float u = kf_bf16_to_f32((unsigned short)uv[j]);
      ov[j] = (short)kf_f32_to_bf16(g * kf_sigmoid(g) * u);
    }
    *reinterpret_cast<kf_short8*>(y + t * F + f) = ov;
  }
}

__global__ void kf_swiglu_bwd_kernel(unsigned short* __restrict__ dh,
                                     const unsigned short* __restrict__ dy,
                                     const unsigned short* __restrict__ h,
                                     int64_t T, int64_t F) {
  const int64_t nvec = T * F / 8;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < nvec;
       i += gridDim.x * (int64_t)blockDim.x) {
    const int64_t t = (i * 8) / F;
    const int64_t f = (i * 8) % F;
    kf_short8 gv = *reinterpret_cast<const kf_short8*>(h + t * 2 * F + f);
    kf_short8 uv = *reinterpret_cast<const kf_short8*>(h + t * 2 * F + F + f);
    kf_short8 dyv = *reinterpret_cast<const kf_short8*>(dy + t * F + f);
    kf_short8 dgv, duv;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float g = kf_bf16_to_f32((unsigned short)gv[j]);
      float u = kf_bf16_to_f32((unsigned short)uv[j]);
      float d = kf_bf16_to_f32((unsigned short)dyv[j]);
      float sig = kf_sigmoid(g);
      float silu = g * sig;
      dgv[j] = (short)kf_f32_to_bf16(d * u * (sig + silu * (1.f - sig)));
      duv[j] = (short)kf_f32_to_bf16(d * silu);
    }
    *reinterpret_cast<kf_short8*>(dh + t * 2 * F + f) = dgv;
    *reinterpret_cast<kf_short8*>(dh + t * 2 * F + F + f) = duv;
  }
}

KF_EXPORT int kf_swiglu_fwd(void* y, const void* h, int64_t T, int64_t F,
                            void* stream) {
  if (F % 8) return (int)hipErrorInvalidValue;
  hipLaunchKernelGGL(kf_swiglu_fwd_kernel, dim3(kf_grid_for(T * F / 8, 256)),
                     dim3(256), 0, (hipStream_t)stream, (unsigned short*)y,
                     (const unsigned short*)h, T, F);
  return (int)hipGetLastError();
}

KF_EXPORT int kf_swiglu_bwd(void* dh, const void* dy, const void* h,
                            int64_t T, int64_t F, void* stream) {
  if (F % 8) return (int)hipErrorInvalidValue;
  hipLaunchKernelGGL(kf_swiglu_bwd_kernel, dim3(kf_grid_for(T * F / 8, 256)),
                     dim3(256), 0, (hipStream_t)stream, (unsigned short*)dh,
                     (const unsigned short*)dy, (const unsigned short*)h, T, F);
  return (int)hipGetLastError();
}
