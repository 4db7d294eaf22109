// rmsnorm.hip — fused RMSNorm forward/backward for CDNA4 (gfx950).
//
// Replaces the per-token normalization of the PyTorchJob worker hot loop
// ("rmsnorm fused kernel, one workgroup per row, wave
// reduction"). Memory-bound: target is HBM BW, so all bf16 access is
// vectorized as short8 (guide G13) and forward reads x exactly once.
//
//   y = x * rsqrt(mean(x^2) + eps) * w        (fp32 math, bf16 in/out)
//
// Backward (per row, s = sum(dy*w*x)):
//   dx = rstd * dy*w - rstd^3/C * s * x
//   dw_j = sum_rows(dy_j * x_j * rstd)   — block-local LDS accumulation,
//   one fp32 atomicAdd pass per block, then a cast kernel.

#include "kf_common.h"

#define RN_BLOCK 256
#define RN_VEC 8

__global__ __launch_bounds__(RN_BLOCK) void kf_rmsnorm_fwd_kernel(
    unsigned short* __restrict__ y, float* __restrict__ rstd_out,
    const unsigned short* __restrict__ x, const unsigned short* __restrict__ w,
    int64_t rows, int cols, float eps) {
  __shared__ float scratch[RN_BLOCK / KF_WAVE];
  const int nvec = cols / RN_VEC;
  for (int64_t row = blockIdx.x; row < rows; row += gridDim.x) {
    const unsigned short* xr = x + row * cols;
    unsigned short* yr = y + row * cols;
    float ss = 0.f;
    for (int i = threadIdx.x; i < nvec; i += RN_BLOCK) {
      kf_short8 xv = *reinterpret_cast<const kf_short8*>(xr + i * RN_VEC);
#pragma unroll
      for (int j = 0; j < RN_VEC; ++j) {
        float f = kf_bf16_to_f32((unsigned short)xv[j]);
        ss += f * f;
      }
    }
    ss = kf_block_reduce(ss, scratch, KfSum{}, 0.f);
    const float rstd = rsqrtf(ss / (float)cols + eps);
    if (threadIdx.x == 0 && rstd_out) rstd_out[row] = rstd;
    for (int i = threadIdx.x; i < nvec; i += RN_BLOCK) {
      kf_short8 xv = *reinterpret_cast<const kf_short8*>(xr + i * RN_VEC);
      kf_short8 wv = *reinterpret_cast<const kf_short8*>(w + i * RN_VEC);
      kf_short8 ov;
#pragma unroll
      for (int j = 0; j < RN_VEC; ++j) {
        float f = kf_bf16_to_f32((unsigned short)xv[j]) * rstd *
                  kf_bf16_to_f32((unsigned short)wv[j]);
        ov[j] = (short)kf_f32_to_bf16(f);
      }
      *reinterpret_cast<kf_short8*>(yr + i * RN_VEC) = ov;
    }
  }
}

// dw: each block accumulates its rows' contribution in LDS (cols * 4 B;
// hidden 4096 -> 16 KiB) and atomicAdds it once into a zeroed fp32 buffer
// (guide G12: block-level pre-reduction, one atomic pass per block — the
// earlier partials+column-sum design ran the reduction on cols/256 = 16
// blocks, 6% of the chip, ~16 ms/step in the r01 profile).
__global__ __launch_bounds__(RN_BLOCK) void kf_rmsnorm_bwd_kernel(
    unsigned short* __restrict__ dx, float* __restrict__ dw_part,
    const unsigned short* __restrict__ dy, const unsigned short* __restrict__ x,
    const unsigned short* __restrict__ w, const float* __restrict__ rstd,
    int64_t rows, int cols) {
  extern __shared__ float lds[];           // [cols] dw accum + reduce scratch
  float* dw_lds = lds;
  float* scratch = lds + cols;
  const int nvec = cols / RN_VEC;
  for (int i = threadIdx.x; i < cols; i += RN_BLOCK) dw_lds[i] = 0.f;
  __syncthreads();

  for (int64_t row = blockIdx.x; row < rows; row += gridDim.x) {
    const unsigned short* dyr = dy + row * cols;
    const unsigned short* xr = x + row * cols;
    unsigned short* dxr = dx + row * cols;
    const float rs = rstd[row];
    float dot = 0.f;
    for (int i = threadIdx.x; i < nvec; i += RN_BLOCK) {
      kf_short8 dyv = *reinterpret_cast<const kf_short8*>(dyr + i * RN_VEC);
      kf_short8 xv = *reinterpret_cast<const kf_short8*>(xr + i * RN_VEC);
      kf_short8 wv = *reinterpret_cast<const kf_short8*>(w + i * RN_VEC);
#pragma unroll
      for (int j = 0; j < RN_VEC; ++j) {
        float dyf = kf_bf16_to_f32((unsigned short)dyv[j]);
        float xf = kf_bf16_to_f32((unsigned short)xv[j]);
        float wf = kf_bf16_to_f32((unsigned short)wv[j]);
        dot += dyf * wf * xf;
        dw_lds[i * RN_VEC + j] += dyf * xf * rs;  // thread-owned slot
      }
    }
    dot = kf_block_reduce(dot, scratch, KfSum{}, 0.f);
    const float k = rs * rs * rs * dot / (float)cols;
    for (int i = threadIdx.x; i < nvec; i += RN_BLOCK) {
      kf_short8 dyv = *reinterpret_cast<const kf_short8*>(dyr + i * RN_VEC);
      kf_short8 xv = *reinterpret_cast<const kf_short8*>(xr + i * RN_VEC);
      kf_short8 wv = *reinterpret_cast<const kf_short8*>(w + i * RN_VEC);
      kf_short8 ov;
#pragma unroll
      for (int j = 0; j < RN_VEC; ++j) {
        float dyf = kf_bf16_to_f32((unsigned short)dyv[j]);
        float xf = kf_bf16_to_f32((unsigned short)xv[j]);
        float wf = kf_bf16_to_f32((unsigned short)wv[j]);
        ov[j] = (short)kf_f32_to_bf16(rs * dyf * wf - k * xf);
      }
      *reinterpret_cast<kf_short8*>(dxr + i * RN_VEC) = ov;
    }
    __syncthreads();  // dw_lds writes of this row done before next row reuse
  }
  for (int i = threadIdx.x; i < cols; i += RN_BLOCK)
    atomicAdd(&dw_part[i], dw_lds[i]);
}

// fp32 accumulator -> bf16 output.
__global__ void kf_cast_bf16_kernel(unsigned short* __restrict__ dst,
                                    const float* __restrict__ src,
                                    int64_t n) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * (int64_t)blockDim.x)
    dst[i] = kf_f32_to_bf16(src[i]);
}

KF_EXPORT int kf_rmsnorm_fwd(void* y, float* rstd, const void* x, const void* w,
                             int64_t rows, int64_t cols, float eps,
                             void* stream) {
  if (cols % RN_VEC) return (int)hipErrorInvalidValue;
  int grid = kf_grid_for(rows, 1);
  hipLaunchKernelGGL(kf_rmsnorm_fwd_kernel, dim3(grid), dim3(RN_BLOCK), 0,
                     (hipStream_t)stream, (unsigned short*)y, rstd,
                     (const unsigned short*)x, (const unsigned short*)w, rows,
                     (int)cols, eps);
  return (int)hipGetLastError();
}

// dw_acc must be a ZEROED fp32 buffer of `cols` floats.
KF_EXPORT int64_t kf_rmsnorm_bwd_nparts(int64_t rows) {
  (void)rows;
  return 1;  // accumulator is [cols] now (kept for ABI compat)
}

KF_EXPORT int kf_rmsnorm_bwd(void* dx, void* dw, float* dw_acc,
                             const void* dy, const void* x, const void* w,
                             const float* rstd, int64_t rows, int64_t cols,
                             void* stream) {
  if (cols % RN_VEC) return (int)hipErrorInvalidValue;
  int grid = (int)(rows < 1024 ? rows : 1024);
  if (grid < 1) grid = 1;
  size_t lds = (cols + RN_BLOCK / KF_WAVE) * sizeof(float);
  hipLaunchKernelGGL(kf_rmsnorm_bwd_kernel, dim3(grid), dim3(RN_BLOCK), lds,
                     (hipStream_t)stream, (unsigned short*)dx, dw_acc,
                     (const unsigned short*)dy, (const unsigned short*)x,
                     (const unsigned short*)w, rstd, rows, (int)cols);
  int err = (int)hipGetLastError();
  if (err) return err;
  hipLaunchKernelGGL(kf_cast_bf16_kernel, dim3(kf_grid_for(cols, 256)),
                     dim3(256), 0, (hipStream_t)stream, (unsigned short*)dw,
                     dw_acc, cols);
  return (int)hipGetLastError();
}
