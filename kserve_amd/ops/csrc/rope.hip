// Rotary position embedding (neox-style rotate-half), in-place on q and k.
//
// cos/sin table is host-precomputed (CDNA guide Appendix B: no on-device
// trig) as [max_pos, head_dim] f32 = [cos(half) | sin(half)].
#include "common.h"

namespace {

// One wave per (token, head). lane i < half handles dim pair (i, i+half).
// head_dim <= 128 assumed per wave pass (loops for larger).
__global__ void rope_kernel(short* __restrict__ q,  // [T, Hq, D] (row stride sq)
                            short* __restrict__ k,  // [T, Hk, D] (row stride sk)
                            const long* __restrict__ positions,  // [T]
                            const float* __restrict__ cos_sin,   // [P, D]
                            const int T, const int Hq, const int Hk,
                            const int D, const long sq, const long sk) {
  const int half = D >> 1;
  const int waves_per_block = blockDim.x >> 6;
  const int gwave = blockIdx.x * waves_per_block + (threadIdx.x >> 6);
  const int lane = threadIdx.x & 63;
  const int total_heads = Hq + Hk;
  const long total_waves = (long)T * total_heads;
  if (gwave >= total_waves) return;
  const int t = gwave / total_heads;
  const int h = gwave % total_heads;
  const long pos = positions[t];
  const float* cs = cos_sin + pos * D;
  short* base = (h < Hq) ? q + (long)t * sq + (long)h * D
                         : k + (long)t * sk + (long)(h - Hq) * D;
  for (int i = lane; i < half; i += 64) {
    float c = cs[i];
    float s = cs[half + i];
    float x1 = bf16_bits_to_float(base[i]);
    float x2 = bf16_bits_to_float(base[i + half]);
    base[i] = float_to_bf16_bits(x1 * c - x2 * s);
    base[i + half] = float_to_bf16_bits(x2 * c + x1 * s);
  }
}

}  // namespace

extern "C" hipError_t ks_rotary_embedding(void* q, void* k,
                                          const void* positions,
                                          const void* cos_sin, int T, int Hq,
                                          int Hk, int D, long sq, long sk,
                                          hipStream_t stream) {
  if (D % 2 != 0) return hipErrorInvalidValue;
  const long total_waves = (long)T * (Hq + Hk);
  if (total_waves == 0) return hipSuccess;
  const int waves_per_block = 4;  // 256 threads
  int grid = (int)((total_waves + waves_per_block - 1) / waves_per_block);
  hipLaunchKernelGGL(rope_kernel, dim3(grid), dim3(256), 0, stream, (short*)q,
                     (short*)k, (const long*)positions, (const float*)cos_sin,
                     T, Hq, Hk, D, sq, sk);
  HIP_CHECK_KERNEL();
  return hipSuccess;
}
