// SwiGLU activation: out = silu(gate) * up, input layout [rows, 2*d] = [gate|up].
// Memory-bound elementwise; vectorized short8 (CDNA guide Appendix B).
#include "common.h"

namespace {

__global__ void silu_and_mul_kernel(short* __restrict__ out,
                                    const short* __restrict__ input,
                                    const int rows, const int d) {
  const int nvec = d >> 3;
  const short8_t* in_vec = reinterpret_cast<const short8_t*>(input);
  short8_t* out_vec = reinterpret_cast<short8_t*>(out);
  const long total = (long)rows * nvec;
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const long row = idx / nvec;
    const long col = idx % nvec;
    short8_t g = in_vec[row * (2 * nvec) + col];
    short8_t u = in_vec[row * (2 * nvec) + nvec + col];
    short8_t o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = bf16_bits_to_float(g[j]);
      float uf = bf16_bits_to_float(u[j]);
      float s = gf / (1.f + __expf(-gf));
      o[j] = float_to_bf16_bits(s * uf);
    }
    out_vec[idx] = o;
  }
}

}  // namespace

extern "C" hipError_t ks_silu_and_mul(void* out, const void* input, int rows,
                                      int d, hipStream_t stream) {
  if (d % 8 != 0) return hipErrorInvalidValue;
  long total = (long)rows * (d >> 3);
  if (total == 0) return hipSuccess;
  int grid = (int)((total + 255) / 256);
  if (grid > 2048) grid = 2048;  // grid-stride (G11)
  hipLaunchKernelGGL(silu_and_mul_kernel, dim3(grid), dim3(256), 0, stream,
                     (short*)out, (const short*)input, rows, d);
  HIP_CHECK_KERNEL();
  return hipSuccess;
}
