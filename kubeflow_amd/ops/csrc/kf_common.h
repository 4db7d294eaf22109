// kf_common.h — shared helpers for kubeflow_amd CDNA4 (gfx950) kernels.
//
// Conventions:
//  * wave width is 64 (CDNA), hard-coded per the CDNA4 programming guide.
//  * bf16 tensors cross the C ABI as raw pointers (ushort storage).
//  * every entry point takes the caller's hipStream_t (torch current stream)
//    and returns hipError_t as int; no allocation/sync inside kernels' host
//    wrappers (graph-capture safe).
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

#define KF_WAVE 64

#define KF_EXPORT extern "C" __attribute__((visibility("default")))

// Vector aliases for wide loads/stores (guide G13: always vectorize bf16).
typedef short kf_short4 __attribute__((ext_vector_type(4)));
typedef short kf_short8 __attribute__((ext_vector_type(8)));
typedef float kf_float4 __attribute__((ext_vector_type(4)));

__device__ __forceinline__ float kf_bf16_to_f32(unsigned short u) {
  union { float f; unsigned int i; } w;
  w.i = ((unsigned int)u) << 16;
  return w.f;
}

// Round-to-nearest-even f32 -> bf16 (matches PyTorch's conversion).
// NaN guard: RNE add would overflow high-payload NaNs (e.g. 0x7FFFFFFF +
// 0x8000 -> -0.0) — propagate a quiet NaN instead, like torch does.
__device__ __forceinline__ unsigned short kf_f32_to_bf16(float f) {
  union { float f; unsigned int i; } w;
  w.f = f;
  unsigned int x = w.i;
  if ((x & 0x7FFFFFFFu) > 0x7F800000u) return 0x7FC0;
  unsigned int rounding = 0x7fff + ((x >> 16) & 1);
  x += rounding;
  return (unsigned short)(x >> 16);
}

// Full-wave (64-lane) butterfly reductions.
__device__ __forceinline__ float kf_wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, KF_WAVE);
  return v;
}

__device__ __forceinline__ float kf_wave_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, KF_WAVE));
  return v;
}

// Block reduction across waves through LDS. `scratch` must hold
// blockDim.x/64 floats. Result valid in all threads.
template <typename Op>
__device__ __forceinline__ float kf_block_reduce(float v, float* scratch, Op op,
                                                 float init) {
  const int lane = threadIdx.x & (KF_WAVE - 1);
  const int wid = threadIdx.x / KF_WAVE;
  const int nw = blockDim.x / KF_WAVE;
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = op(v, __shfl_xor(v, off, KF_WAVE));
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  float r = init;
  for (int i = 0; i < nw; ++i) r = op(r, scratch[i]);
  __syncthreads();
  return r;
}

struct KfSum { __device__ float operator()(float a, float b) const { return a + b; } };
struct KfMax { __device__ float operator()(float a, float b) const { return fmaxf(a, b); } };

// Memory-bound grid sizing (guide G11): cap at ~8 blocks/CU × 256 CUs and
// grid-stride the remainder.
static inline int kf_grid_for(int64_t work_items, int block) {
  int64_t blocks = (work_items + block - 1) / block;
  if (blocks > 2048) blocks = 2048;
  if (blocks < 1) blocks = 1;
  return (int)blocks;
}
