// Common device helpers for the rbg_amd CDNA4 (gfx950) kernels.
// Wave size is 64 everywhere; block sizes are multiples of 64.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <stdint.h>

#define WAVE_SIZE 64
#define DEV_INLINE __device__ __forceinline__

// 8 bf16 = 16 bytes — the vector width every global bf16 access should use
// (guide: scalar bf16 loads are ~2-2.5x slower).
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;
typedef __attribute__((ext_vector_type(2))) __bf16 bf16x2;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) float f32x8;
typedef __attribute__((ext_vector_type(2))) float f32x2;
// MFMA fragment types (gfx950 16x16x32 bf16: 8 bf16 in, 4 f32 acc)
typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) short short4_t;
typedef __attribute__((ext_vector_type(16))) float f32x16;

DEV_INLINE float bf2f(__hip_bfloat16 v) { return __bfloat162float(v); }
DEV_INLINE __hip_bfloat16 f2bf(float v) { return __float2bfloat16(v); }

union Bf16x8U {
  bf16x8 v;
  uint4 u;                 // one 16-byte load/store
  __hip_bfloat16 e[8];
  short s[8];
};

union F32x4U {
  f32x4 v;
  float e[4];
};

// Full-wave (64-lane) reductions via xor shuffles.
DEV_INLINE float wave_sum(float x) {
#pragma unroll
  for (int off = 32; off >= 1; off >>= 1) x += __shfl_xor(x, off, 64);
  return x;
}

DEV_INLINE float wave_max(float x) {
#pragma unroll
  for (int off = 32; off >= 1; off >>= 1) x = fmaxf(x, __shfl_xor(x, off, 64));
  return x;
}

// Reduction across a 32-lane half (lanes with the same (l>>5)).
DEV_INLINE float half_wave_sum(float x) {
#pragma unroll
  for (int off = 16; off >= 1; off >>= 1) x += __shfl_xor(x, off, 64);
  return x;
}

DEV_INLINE float half_wave_max(float x) {
#pragma unroll
  for (int off = 16; off >= 1; off >>= 1)
    x = fmaxf(x, __shfl_xor(x, off, 64));
  return x;
}

// Reduction across 16-lane groups (same l>>4).
DEV_INLINE float group16_sum(float x) {
#pragma unroll
  for (int off = 8; off >= 1; off >>= 1) x += __shfl_xor(x, off, 64);
  return x;
}

DEV_INLINE float group16_max(float x) {
#pragma unroll
  for (int off = 8; off >= 1; off >>= 1) x = fmaxf(x, __shfl_xor(x, off, 64));
  return x;
}

// Block-level sum over NW waves (NW <= 16); every thread must participate.
template <int NW>
DEV_INLINE float block_sum(float x, float* scratch /* NW floats */) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  x = wave_sum(x);
  if (lane == 0) scratch[wave] = x;
  __syncthreads();
  float total = 0.f;
#pragma unroll
  for (int w = 0; w < NW; ++w) total += scratch[w];
  return total;
}

DEV_INLINE int cdiv(int a, int b) { return (a + b - 1) / b; }

#define HIP_KERNEL_CHECK()                                                    \
  do {                                                                        \
    hipError_t err_ = hipGetLastError();                                      \
    if (err_ != hipSuccess) {                                                 \
      printf("HIP kernel launch error: %s\n", hipGetErrorString(err_));       \
    }                                                                         \
  } while (0)
