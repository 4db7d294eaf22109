// kv_store.hip — fused decode KV-cache scatter.
//
// The decode step writes this token's k,v into cache row [slot, pos] for
// every sequence in the batch. Doing that from Python is two advanced-
// indexing kernels per layer plus an index-dtype conversion (~15 us of
// the ~160 us decode layer, profiles/r02_decode_anatomy.md); this is one
// launch writing both rows. Capture-safe (pure function of its tensors).
//
// k/v are the [N, Hkv*D] bf16 token rows (contiguous); caches are
// [SLOTS, SMAX, Hkv, D] bf16. Row bytes = Hkv*D*2 (2048 for llama3-8b):
// one 128-lane block copies k and v for one sequence in 16 B chunks.

#include "kf_common.h"

typedef unsigned int kf_u32x4 __attribute__((ext_vector_type(4)));

__global__ __launch_bounds__(128, 8) void kf_kv_store_kernel(
    unsigned short* __restrict__ ck, unsigned short* __restrict__ cv,
    const unsigned short* __restrict__ k, const unsigned short* __restrict__ v,
    const int* __restrict__ slots, const long long* __restrict__ positions,
    long long smax, long long row, int n) {
  const int i = blockIdx.x;
  if (i >= n) return;
  const long long dst = ((long long)slots[i] * smax + positions[i]) * row;
  const long long src = (long long)i * row;
  // row is in bf16 elements; copy in 8-element (16 B) chunks
  for (long long e = (long long)threadIdx.x * 8; e < row;
       e += (long long)blockDim.x * 8) {
    *reinterpret_cast<kf_u32x4*>(ck + dst + e) =
        *reinterpret_cast<const kf_u32x4*>(k + src + e);
    *reinterpret_cast<kf_u32x4*>(cv + dst + e) =
        *reinterpret_cast<const kf_u32x4*>(v + src + e);
  }
}

KF_EXPORT int kf_kv_store(void* ck, void* cv, const void* k, const void* v,
                          const void* slots, const void* positions,
                          int64_t smax, int64_t row, int64_t n,
                          void* stream) {
  if (row % 8) return (int)hipErrorInvalidValue;
  hipLaunchKernelGGL(kf_kv_store_kernel, dim3((unsigned)n), dim3(128), 0,
                     (hipStream_t)stream, (unsigned short*)ck,
                     (unsigned short*)cv, (const unsigned short*)k,
                     (const unsigned short*)v, (const int*)slots,
                     (const long long*)positions, smax, row, (int)n);
  return (int)hipGetLastError();
}
