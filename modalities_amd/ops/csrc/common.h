// Common helpers for MI355X (gfx950, CDNA4) kernels.
// Wave size is 64 on CDNA4 — every cross-lane idiom below assumes it.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define WAVE_SIZE 64

// ext_vector types for vectorized access (guide G13: always vectorize bf16).
typedef __attribute__((ext_vector_type(2))) float floatx2;
typedef __attribute__((ext_vector_type(4))) float floatx4;
typedef __attribute__((ext_vector_type(16))) float floatx16;
typedef __attribute__((ext_vector_type(4))) short shortx4;
typedef __attribute__((ext_vector_type(8))) short shortx8;

__device__ __forceinline__ float bf16_to_f32(unsigned short u) {
  union { unsigned int i; float f; } c;
  c.i = ((unsigned int)u) << 16;
  return c.f;
}

__device__ __forceinline__ unsigned short f32_to_bf16(float f) {
  union { float f; unsigned int i; } c;
  c.f = f;
  unsigned int i = c.i;
  // round-to-nearest-even
  unsigned int rounded = i + 0x7FFF + ((i >> 16) & 1);
  if ((i & 0x7F800000) == 0x7F800000) rounded = i;  // inf/nan passthrough
  return (unsigned short)(rounded >> 16);
}

// Wave-wide f32 sum via shuffle tree (64 lanes).
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE_SIZE);
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE_SIZE));
  return v;
}

// Block-level f32 sum: wave reduce + LDS combine. `scratch` needs
// >= blockDim.x / WAVE_SIZE floats. All threads return the total.
__device__ __forceinline__ float block_reduce_sum(float v, float* scratch) {
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wid = threadIdx.x / WAVE_SIZE;
  const int nwaves = blockDim.x / WAVE_SIZE;
  v = wave_reduce_sum(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  float total = 0.f;
#pragma unroll 8
  for (int i = 0; i < nwaves; ++i) total += scratch[i];
  __syncthreads();
  return total;
}

__device__ __forceinline__ float block_reduce_max(float v, float* scratch) {
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wid = threadIdx.x / WAVE_SIZE;
  const int nwaves = blockDim.x / WAVE_SIZE;
  v = wave_reduce_max(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  float m = -INFINITY;
#pragma unroll 8
  for (int i = 0; i < nwaves; ++i) m = fmaxf(m, scratch[i]);
  __syncthreads();
  return m;
}

#define HIP_CHECK_KERNEL()                                              \
  do {                                                                  \
    hipError_t e = hipGetLastError();                                   \
    TORCH_CHECK(e == hipSuccess, "HIP kernel launch failed: ",          \
                hipGetErrorString(e));                                  \
  } while (0)
