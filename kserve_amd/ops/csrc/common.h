// Common device helpers for kserve_amd CDNA4 (gfx950) kernels.
//
// Conventions (per /opt/skills/guides/cdna_hip_programming.md):
//  - wave = 64 lanes, hard-coded
//  - bf16 memory traffic vectorized as short8 (16 B/lane) [Guideline 13]
//  - block sizes are multiples of 64
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <stdint.h>

#define WAVE_SIZE 64

typedef short short8_t __attribute__((ext_vector_type(8)));
typedef short short4_t __attribute__((ext_vector_type(4)));
typedef float float4_t __attribute__((ext_vector_type(4)));
typedef float float2_t __attribute__((ext_vector_type(2)));
typedef __hip_bfloat16 bf16_t;

__device__ __forceinline__ float bf16_bits_to_float(short bits) {
  union {
    unsigned int u;
    float f;
  } cvt;
  cvt.u = ((unsigned int)(unsigned short)bits) << 16;
  return cvt.f;
}

__device__ __forceinline__ short float_to_bf16_bits(float f) {
  union {
    unsigned int u;
    float f;
  } cvt;
  cvt.f = f;
  // round-to-nearest-even
  unsigned int lsb = (cvt.u >> 16) & 1u;
  cvt.u += 0x7fffu + lsb;
  return (short)(cvt.u >> 16);
}

// -- wave reductions (64-wide) ------------------------------------------------

__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE_SIZE);
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_xor(v, off, WAVE_SIZE));
  return v;
}

// reduce within contiguous groups of `width` lanes (width = power of 2 <= 64)
template <int WIDTH>
__device__ __forceinline__ float group_reduce_sum(float v) {
#pragma unroll
  for (int off = WIDTH / 2; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE_SIZE);
  return v;
}

template <int WIDTH>
__device__ __forceinline__ float group_reduce_max(float v) {
#pragma unroll
  for (int off = WIDTH / 2; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_xor(v, off, WAVE_SIZE));
  return v;
}

// -- block reduction via LDS (NWAVES <= 16) ----------------------------------

template <int NWAVES>
__device__ __forceinline__ float block_reduce_sum(float v, float* lds_scratch) {
  int lane = threadIdx.x & (WAVE_SIZE - 1);
  int wave = threadIdx.x / WAVE_SIZE;
  v = wave_reduce_sum(v);
  if (lane == 0) lds_scratch[wave] = v;
  __syncthreads();
  float out = 0.f;
#pragma unroll
  for (int w = 0; w < NWAVES; ++w) out += lds_scratch[w];
  return out;
}

// ceil-div
__host__ __device__ __forceinline__ int cdiv(int a, int b) { return (a + b - 1) / b; }

#define HIP_CHECK_KERNEL()                                  \
  do {                                                      \
    hipError_t e = hipGetLastError();                       \
    if (e != hipSuccess) {                                  \
      return e;                                             \
    }                                                       \
  } while (0)
