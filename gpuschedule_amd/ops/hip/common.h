// Shared helpers for gfx950 (MI355X / CDNA4) kernels.
// Wave size is 64 on CDNA4; block sizes are multiples of 64.
#pragma once
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64

// Vector types for wide loads: 8 bf16 = 16 B = one dwordx4.
typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) short short4v;
typedef __attribute__((ext_vector_type(4))) float float4v;
typedef __attribute__((ext_vector_type(8))) float float8v;
typedef __attribute__((ext_vector_type(16))) float float16v;

__device__ __forceinline__ float bf16_to_f32(short u) {
  union { float f; unsigned int i; } c;
  c.i = ((unsigned int)(unsigned short)u) << 16;
  return c.f;
}

__device__ __forceinline__ short f32_to_bf16(float f) {
  // round-to-nearest-even, matching PyTorch's conversion
  union { float f; unsigned int i; } c;
  c.f = f;
  unsigned int x = c.i;
  unsigned int rounding_bias = 0x7FFF + ((x >> 16) & 1);
  if ((x & 0x7FFFFFFF) > 0x7F800000) return (short)0x7FC0;  // NaN
  return (short)((x + rounding_bias) >> 16);
}

// Full-wave (64-lane) float reduction via xor shuffles.
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    v += __shfl_xor(v, off, WAVE);
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_xor(v, off, WAVE));
  return v;
}

// Block-level sum over NWAVES waves (NWAVES <= 16). `scratch` is LDS
// with >= NWAVES floats. Every thread returns the block sum.
template <int NWAVES>
__device__ __forceinline__ float block_reduce_sum(float v, float* scratch) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  v = wave_reduce_sum(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  float s = (lane < NWAVES) ? scratch[lane] : 0.0f;
  s = wave_reduce_sum(s);  // every lane has the total afterwards
  return s;
}

#define HIP_CHECK_LAST()                                             \
  do {                                                               \
    hipError_t e = hipGetLastError();                                \
    if (e != hipSuccess)                                             \
      throw std::runtime_error(std::string("HIP kernel launch: ") +  \
                               hipGetErrorString(e));                \
  } while (0)
