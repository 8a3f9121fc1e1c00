// LayerNorm (+ optional fused residual add) and GELU — the encoder-path
// (BERT-class) elementwise ops. Memory-bound; vectorized short8 per CDNA
// guide Appendix B.
#include "common.h"

namespace {

template <bool FUSED_ADD>
__global__ void layer_norm_kernel(
    short* __restrict__ out,           // [rows, hidden]
    const short* __restrict__ input,   // [rows, hidden]
    const short* __restrict__ residual,  // [rows, hidden] or nullptr
    const short* __restrict__ weight,  // [hidden]
    const short* __restrict__ bias,    // [hidden]
    const float eps, const int rows, const int hidden) {
  constexpr int NWAVES = 4;
  __shared__ float s_sum[NWAVES];
  __shared__ float s_sq[NWAVES];
  __shared__ float s_mean, s_rstd;
  const int nvec = hidden >> 3;

  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const short8_t* rin = reinterpret_cast<const short8_t*>(input) + (size_t)row * nvec;
    const short8_t* rres =
        FUSED_ADD ? reinterpret_cast<const short8_t*>(residual) + (size_t)row * nvec
                  : nullptr;
    short8_t* rout = reinterpret_cast<short8_t*>(out) + (size_t)row * nvec;
    const short8_t* wv = reinterpret_cast<const short8_t*>(weight);
    const short8_t* bv = reinterpret_cast<const short8_t*>(bias);

    float sum = 0.f, sq = 0.f;
    for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
      short8_t v = rin[i];
      short8_t r;
      if constexpr (FUSED_ADD) r = rres[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = bf16_bits_to_float(v[j]);
        if constexpr (FUSED_ADD) f += bf16_bits_to_float(r[j]);
        sum += f;
        sq += f * f;
      }
    }
    const int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
    sum = wave_reduce_sum(sum);
    sq = wave_reduce_sum(sq);
    if (lane == 0) {
      s_sum[wave] = sum;
      s_sq[wave] = sq;
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      float ts = 0.f, tq = 0.f;
#pragma unroll
      for (int w = 0; w < NWAVES; ++w) {
        ts += s_sum[w];
        tq += s_sq[w];
      }
      const float mean = ts / hidden;
      const float var = tq / hidden - mean * mean;
      s_mean = mean;
      s_rstd = rsqrtf(var + eps);
    }
    __syncthreads();
    const float mean = s_mean, rstd = s_rstd;
    for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
      short8_t v = rin[i];
      short8_t r;
      if constexpr (FUSED_ADD) r = rres[i];
      short8_t w = wv[i];
      short8_t b = bv[i];
      short8_t o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = bf16_bits_to_float(v[j]);
        if constexpr (FUSED_ADD) f += bf16_bits_to_float(r[j]);
        f = (f - mean) * rstd * bf16_bits_to_float(w[j]) +
            bf16_bits_to_float(b[j]);
        o[j] = float_to_bf16_bits(f);
      }
      rout[i] = o;
    }
    __syncthreads();
  }
}

__global__ void gelu_kernel(short* __restrict__ out,
                            const short* __restrict__ input, const long nvec) {
  const short8_t* in = reinterpret_cast<const short8_t*>(input);
  short8_t* ov = reinterpret_cast<short8_t*>(out);
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < nvec;
       i += (long)gridDim.x * blockDim.x) {
    short8_t v = in[i];
    short8_t o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float x = bf16_bits_to_float(v[j]);
      // erf-based gelu (matches torch default / BERT)
      o[j] = float_to_bf16_bits(0.5f * x * (1.f + erff(x * 0.70710678f)));
    }
    ov[i] = o;
  }
}

}  // namespace

extern "C" {

hipError_t ks_layer_norm(void* out, const void* input, const void* weight,
                         const void* bias, float eps, int rows, int hidden,
                         hipStream_t stream) {
  if (hidden % 8 != 0) return hipErrorInvalidValue;
  int grid = rows < 2048 ? rows : 2048;
  if (grid == 0) return hipSuccess;
  hipLaunchKernelGGL((layer_norm_kernel<false>), dim3(grid), dim3(256), 0,
                     stream, (short*)out, (const short*)input, nullptr,
                     (const short*)weight, (const short*)bias, eps, rows,
                     hidden);
  HIP_CHECK_KERNEL();
  return hipSuccess;
}

hipError_t ks_fused_add_layer_norm(void* out, const void* input,
                                   const void* residual, const void* weight,
                                   const void* bias, float eps, int rows,
                                   int hidden, hipStream_t stream) {
  if (hidden % 8 != 0) return hipErrorInvalidValue;
  int grid = rows < 2048 ? rows : 2048;
  if (grid == 0) return hipSuccess;
  hipLaunchKernelGGL((layer_norm_kernel<true>), dim3(grid), dim3(256), 0,
                     stream, (short*)out, (const short*)input,
                     (const short*)residual, (const short*)weight,
                     (const short*)bias, eps, rows, hidden);
  HIP_CHECK_KERNEL();
  return hipSuccess;
}

hipError_t ks_gelu(void* out, const void* input, long numel,
                   hipStream_t stream) {
  if (numel % 8 != 0) return hipErrorInvalidValue;
  long nvec = numel >> 3;
  if (nvec == 0) return hipSuccess;
  int grid = (int)((nvec + 255) / 256);
  if (grid > 2048) grid = 2048;
  hipLaunchKernelGGL(gelu_kernel, dim3(grid), dim3(256), 0, stream,
                     (short*)out, (const short*)input, nvec);
  HIP_CHECK_KERNEL();
  return hipSuccess;
}
}
