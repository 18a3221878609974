// Common helpers for hypha_amd CDNA4 (gfx950) HIP kernels.
// Wave size is 64 on CDNA4; all block sizes are multiples of 64.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define HYPHA_WAVE 64

#define HIP_CHECK(expr)                                                        \
  do {                                                                         \
    hipError_t _e = (expr);                                                    \
    if (_e != hipSuccess) {                                                    \
      TORCH_CHECK(false, "HIP error: ", hipGetErrorString(_e), " at ",        \
                  __FILE__, ":", __LINE__);                                    \
    }                                                                          \
  } while (0)

using bf16 = __hip_bfloat16;
typedef __attribute__((ext_vector_type(2))) float f32x2;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(16))) float f32x16;
typedef __attribute__((ext_vector_type(4))) short s16x4;
typedef __attribute__((ext_vector_type(8))) short s16x8;

__device__ __forceinline__ float bf2f(short u) {
  union {
    float f;
    unsigned int i;
  } c;
  c.i = ((unsigned int)(unsigned short)u) << 16;
  return c.f;
}

__device__ __forceinline__ short f2bf(float f) {
  union {
    float f;
    unsigned int i;
  } c;
  c.f = f;
  unsigned int x = c.i;
  // round-to-nearest-even, NaN-safe
  if ((x & 0x7fffffffu) > 0x7f800000u) return (short)0x7fc0;  // NaN
  unsigned int round = 0x7fffu + ((x >> 16) & 1u);
  return (short)((x + round) >> 16);
}

// Grid sizing for memory-bound grid-stride kernels: cap at ~8 blocks/CU.
inline int elementwise_grid(long long n_items, int block = 256) {
  long long blocks = (n_items + block - 1) / block;
  if (blocks > 2048) blocks = 2048;
  if (blocks < 1) blocks = 1;
  return (int)blocks;
}

// Wave-level f32 reductions (64 lanes).
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

// torch current-stream helper (native HIP API, no CUDA masquerade)
#ifdef __HIP_PLATFORM_AMD__
#include <c10/hip/HIPStream.h>
inline hipStream_t hypha_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}
#endif
