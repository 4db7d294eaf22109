// cross_entropy.hip — fused softmax cross-entropy over a large vocab, CDNA4.
//
// SURVEY.md §2.13 "cross_entropy + softmax kernels: two-pass online softmax,
// split-V reduction" for the 128256-entry Llama-3 vocab. One workgroup per
// token row; each thread streams a vectorized slice of the row keeping an
// online (max, sum) pair, then a block reduction merges them — the row is
// read once in forward.
//
//   fwd: lse[t] = logsumexp(logits[t,:]);  loss_sum += lse[t] - logit[t,y_t]
//   bwd: dlogits[t,v] = scale[t] * (exp(logit - lse) - [v == y_t])
//
// Rows with target == ignore_index contribute 0 loss and 0 grad; the valid
// count is handled by the caller (scale). loss_sum accumulation is one
// atomicAdd per row (guide G12: block-reduce first, one atomic per block).

#include "kf_common.h"

#define CE_BLOCK 256
#define CE_VEC 8

__global__ __launch_bounds__(CE_BLOCK) void kf_ce_fwd_kernel(
    float* __restrict__ loss_sum, float* __restrict__ lse_out,
    const unsigned short* __restrict__ logits,
    const int64_t* __restrict__ targets, int64_t T, int64_t V,
    int64_t ignore_index) {
  __shared__ float scratch[CE_BLOCK / KF_WAVE];
  const int64_t nvec = V / CE_VEC;
  for (int64_t t = blockIdx.x; t < T; t += gridDim.x) {
    const unsigned short* row = logits + t * V;
    float mx = -INFINITY, sm = 0.f;
    for (int64_t i = threadIdx.x; i < nvec; i += CE_BLOCK) {
      kf_short8 lv = *reinterpret_cast<const kf_short8*>(row + i * CE_VEC);
#pragma unroll
      for (int j = 0; j < CE_VEC; ++j) {
        float f = kf_bf16_to_f32((unsigned short)lv[j]);
        if (f > mx) { sm *= __expf(mx - f); mx = f; }
        sm += __expf(f - mx);
      }
    }
    // tail (V % 8) — Llama vocab 128256 is 8-divisible, generic anyway
    for (int64_t i = nvec * CE_VEC + threadIdx.x; i < V; i += CE_BLOCK) {
      float f = kf_bf16_to_f32(row[i]);
      if (f > mx) { sm *= __expf(mx - f); mx = f; }
      sm += __expf(f - mx);
    }
    float gmx = kf_block_reduce(mx, scratch, KfMax{}, -INFINITY);
    float part = sm * __expf(mx - gmx);
    if (mx == -INFINITY) part = 0.f;
    float gsum = kf_block_reduce(part, scratch, KfSum{}, 0.f);
    const float lse = gmx + __logf(gsum);
    if (threadIdx.x == 0) {
      lse_out[t] = lse;
      const int64_t y = targets[t];
      if (y != ignore_index)
        atomicAdd(loss_sum, lse - kf_bf16_to_f32(row[y]));
    }
  }
}

__global__ __launch_bounds__(CE_BLOCK) void kf_ce_bwd_kernel(
    unsigned short* __restrict__ dlogits,
    const unsigned short* __restrict__ logits, const float* __restrict__ lse,
    const int64_t* __restrict__ targets, const float* __restrict__ scale_ptr,
    int64_t T, int64_t V, int64_t ignore_index) {
  const float scale = scale_ptr[0];
  const int64_t nvec = V / CE_VEC;
  for (int64_t t = blockIdx.x; t < T; t += gridDim.x) {
    const unsigned short* row = logits + t * V;
    unsigned short* drow = dlogits + t * V;
    const int64_t y = targets[t];
    const float l = lse[t];
    const float sc = (y == ignore_index) ? 0.f : scale;
    for (int64_t i = threadIdx.x; i < nvec; i += CE_BLOCK) {
      kf_short8 lv = *reinterpret_cast<const kf_short8*>(row + i * CE_VEC);
      kf_short8 ov;
#pragma unroll
      for (int j = 0; j < CE_VEC; ++j) {
        const int64_t v = i * CE_VEC + j;
        float p = __expf(kf_bf16_to_f32((unsigned short)lv[j]) - l);
        ov[j] = (short)kf_f32_to_bf16(sc * (p - (v == y ? 1.f : 0.f)));
      }
      *reinterpret_cast<kf_short8*>(drow + i * CE_VEC) = ov;
    }
    for (int64_t v = nvec * CE_VEC + threadIdx.x; v < V; v += CE_BLOCK) {
      float p = __expf(kf_bf16_to_f32(row[v]) - l);
      drow[v] = kf_f32_to_bf16(sc * (p - (v == y ? 1.f : 0.f)));
    }
  }
}

KF_EXPORT int kf_ce_fwd(float* loss_sum, float* lse, const void* logits,
                        const int64_t* targets, int64_t T, int64_t V,
                        int64_t ignore_index, void* stream) {
  hipLaunchKernelGGL(kf_ce_fwd_kernel, dim3(kf_grid_for(T, 1)), dim3(CE_BLOCK),
                     0, (hipStream_t)stream, loss_sum, lse,
                     (const unsigned short*)logits, targets, T, V,
                     ignore_index);
  return (int)hipGetLastError();
}

KF_EXPORT int kf_ce_bwd(void* dlogits, const void* logits, const float* lse,
                        const int64_t* targets, const float* scale, int64_t T,
                        int64_t V, int64_t ignore_index, void* stream) {
  hipLaunchKernelGGL(kf_ce_bwd_kernel, dim3(kf_grid_for(T, 1)), dim3(CE_BLOCK),
                     0, (hipStream_t)stream, (unsigned short*)dlogits,
                     (const unsigned short*)logits, lse, targets, scale, T, V,
                     ignore_index);
  return (int)hipGetLastError();
}
