// Common helpers for esr_amd gfx950 (CDNA4) kernels.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define ESR_INLINE __device__ __forceinline__

// Wavefront on CDNA4 is 64 lanes; blocks are multiples of 64.
constexpr int ESR_BLOCK = 256;
// Grid cap for memory-bound grid-stride kernels (256 CUs x 8 blocks).
constexpr int ESR_MAX_BLOCKS = 2048;

__host__ __device__ inline int esr_grid(long long n, int block = ESR_BLOCK,
                                        int cap = ESR_MAX_BLOCKS) {
  long long g = (n + block - 1) / block;
  return (int)(g < cap ? (g > 1 ? g : 1) : cap);
}

#define ESR_KERNEL_LOOP(i, n)                                        \
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; \
       i < (n); i += (long long)blockDim.x * gridDim.x)

// ---- dtype conversion helpers (load as T, compute in fp32) ----
template <typename T> ESR_INLINE float esr_to_f32(T v);
template <> ESR_INLINE float esr_to_f32<float>(float v) { return v; }
template <> ESR_INLINE float esr_to_f32<__hip_bfloat16>(__hip_bfloat16 v) {
  return __bfloat162float(v);
}

template <typename T> ESR_INLINE T esr_from_f32(float v);
template <> ESR_INLINE float esr_from_f32<float>(float v) { return v; }
template <> ESR_INLINE __hip_bfloat16 esr_from_f32<__hip_bfloat16>(float v) {
  return __float2bfloat16(v);
}

ESR_INLINE float esr_sigmoid(float x) { return 1.0f / (1.0f + __expf(-x)); }
