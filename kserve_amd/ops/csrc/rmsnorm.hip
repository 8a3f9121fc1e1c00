// RMSNorm kernels (bf16, memory-bound).
//
// Replaces the RMSNorm the reference exercises through vLLM/torch
// (SURVEY.md §2.7). Design per CDNA guide Appendix B: vectorized short8
// loads (16 B/lane), one block per token row, grid-stride over rows.
#include "common.h"

namespace {

// Each block handles one row of `hidden` elements; NWAVES=4 (256 threads).
template <bool FUSED_ADD>
__global__ void rms_norm_kernel(
    short* __restrict__ out,          // [rows, hidden] bf16 bits
    short* __restrict__ input,        // [rows, hidden]; FUSED_ADD: in-place normed
    short* __restrict__ residual,     // [rows, hidden] or nullptr
    const short* __restrict__ weight, // [hidden]
    const float eps,
    const int rows,
    const int hidden) {
  constexpr int NWAVES = 4;
  __shared__ float s_partial[NWAVES];
  __shared__ float s_scale;
  const int nvec = hidden >> 3;  // short8 vectors per row

  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    short8_t* row_in = reinterpret_cast<short8_t*>(input) + (size_t)row * nvec;
    short8_t* row_res =
        FUSED_ADD ? reinterpret_cast<short8_t*>(residual) + (size_t)row * nvec
                  : nullptr;
    short8_t* row_out = reinterpret_cast<short8_t*>(out) + (size_t)row * nvec;
    const short8_t* wvec = reinterpret_cast<const short8_t*>(weight);

    float ssq = 0.f;
    for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
      short8_t v = row_in[i];
      float vals[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) vals[j] = bf16_bits_to_float(v[j]);
      if constexpr (FUSED_ADD) {
        short8_t r = row_res[i];
#pragma unroll
        for (int j = 0; j < 8; ++j) vals[j] += bf16_bits_to_float(r[j]);
        short8_t sum;
#pragma unroll
        for (int j = 0; j < 8; ++j) sum[j] = float_to_bf16_bits(vals[j]);
        row_res[i] = sum;  // residual <- x + residual
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) ssq += vals[j] * vals[j];
    }
    // block reduce
    int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
    ssq = wave_reduce_sum(ssq);
    if (lane == 0) s_partial[wave] = ssq;
    __syncthreads();
    if (threadIdx.x == 0) {
      float total = 0.f;
#pragma unroll
      for (int w = 0; w < NWAVES; ++w) total += s_partial[w];
      s_scale = rsqrtf(total / hidden + eps);
    }
    __syncthreads();
    const float scale = s_scale;

    for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
      // FUSED_ADD: residual now holds x+residual; re-read it (L2-hot)
      short8_t v = FUSED_ADD ? row_res[i] : row_in[i];
      short8_t w = wvec[i];
      short8_t o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float val = bf16_bits_to_float(v[j]) * scale * bf16_bits_to_float(w[j]);
        o[j] = float_to_bf16_bits(val);
      }
      row_out[i] = o;
    }
    __syncthreads();
  }
}

}  // namespace

extern "C" {

hipError_t ks_rms_norm(void* out, const void* input, const void* weight,
                       float eps, int rows, int hidden, hipStream_t stream) {
  if (hidden % 8 != 0) return hipErrorInvalidValue;
  int grid = rows < 2048 ? rows : 2048;
  if (grid == 0) return hipSuccess;
  hipLaunchKernelGGL((rms_norm_kernel<false>), dim3(grid), dim3(256), 0, stream,
                     (short*)out, (short*)input, nullptr, (const short*)weight,
                     eps, rows, hidden);
  HIP_CHECK_KERNEL();
  return hipSuccess;
}

// x <- rmsnorm(x + residual), residual <- x + residual (both in-place)
hipError_t ks_fused_add_rms_norm(void* x, void* residual, const void* weight,
                                 float eps, int rows, int hidden,
                                 hipStream_t stream) {
  if (hidden % 8 != 0) return hipErrorInvalidValue;
  int grid = rows < 2048 ? rows : 2048;
  if (grid == 0) return hipSuccess;
  hipLaunchKernelGGL((rms_norm_kernel<true>), dim3(grid), dim3(256), 0, stream,
                     (short*)x, (short*)x, (short*)residual,
                     (const short*)weight, eps, rows, hidden);
  HIP_CHECK_KERNEL();
  return hipSuccess;
}
}
