// Common helpers for the gfx950 (CDNA4) kernels.
// Wave size is 64 on CDNA — every reduction below is wave64-shaped
// (see /opt/skills/guides/cdna_hip_programming.md §1).
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define WAVE 64

using bf16 = __hip_bfloat16;
using fp16 = __half;

typedef short short4v __attribute__((ext_vector_type(4)));
typedef short short8v __attribute__((ext_vector_type(8)));
typedef float float4v __attribute__((ext_vector_type(4)));
typedef __bf16 bf16x2_t __attribute__((ext_vector_type(2)));
typedef __bf16 bf16x8_t __attribute__((ext_vector_type(8)));

// v_dot2c_f32_bf16: 2 bf16 products + f32 accumulate in ONE VALU instr
// (gfx950) — same numerics as cvt+fma chains with f32 accumulation.
__device__ __forceinline__ float dot2_bf16(bf16x2_t a, bf16x2_t b, float c) {
  return __builtin_amdgcn_fdot2_f32_bf16(a, b, c, false);
}

__device__ __forceinline__ float bf2f(bf16 v) { return __bfloat162float(v); }
__device__ __forceinline__ bf16 f2bf(float v) { return __float2bfloat16(v); }

// bits of a bf16 (stored as short) -> float
__device__ __forceinline__ float bfbits2f(short s) {
  unsigned int u = ((unsigned int)(unsigned short)s) << 16;
  return __uint_as_float(u);
}

// Wave-level reductions (64 lanes).
__device__ __forceinline__ float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

__device__ __forceinline__ float wave_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE));
  return v;
}

// Block-level reduce across NW waves using LDS scratch (NW floats).
template <int BLOCK>
__device__ __forceinline__ float block_sum(float v, float* scratch) {
  constexpr int NW = BLOCK / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  v = wave_sum(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  float r = 0.0f;
#pragma unroll
  for (int i = 0; i < NW; ++i) r += scratch[i];  // LDS broadcast reads
  __syncthreads();
  return r;
}

template <int BLOCK>
__device__ __forceinline__ float block_max(float v, float* scratch) {
  constexpr int NW = BLOCK / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  v = wave_max(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  float r = scratch[0];
#pragma unroll
  for (int i = 1; i < NW; ++i) r = fmaxf(r, scratch[i]);
  __syncthreads();
  return r;
}

#define HIP_CHECK_LAST()                                                     \
  do {                                                                       \
    hipError_t e = hipGetLastError();                                        \
    if (e != hipSuccess) {                                                   \
      printf("HIP error %s at %s:%d\n", hipGetErrorString(e), __FILE__,      \
             __LINE__);                                                      \
    }                                                                        \
  } while (0)
