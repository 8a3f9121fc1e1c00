// On-device verification of the MFMA operand layouts in mfma_layouts.h:
// computes C = A @ B (16x16x32 bf16) loading A/B from row-major global
// matrices via the assumed fragment layouts. The GPU test compares against
// a torch matmul — a mismatch means the layout macros are wrong.
#include "common.h"
#include "mfma_layouts.h"

namespace {

__global__ void mfma_probe_kernel(float* __restrict__ c_out,   // [16][16]
                                  const short* __restrict__ a,  // [16][32]
                                  const short* __restrict__ b   // [32][16]
) {
  const int lane = threadIdx.x & 63;
  bf16x8_t a_frag, b_frag;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const int k = MFMA_K_OF(lane, j);
    union { short s; __bf16 h; } ca, cb;
    ca.s = a[MFMA_RC_OF(lane) * 32 + k];      // A[row][k]
    cb.s = b[k * 16 + MFMA_RC_OF(lane)];      // B[k][col]
    a_frag[j] = ca.h;
    b_frag[j] = cb.h;
  }
  f32x4_t acc = {0.f, 0.f, 0.f, 0.f};
  acc = mfma16x16x32(a_frag, b_frag, acc);
#pragma unroll
  for (int reg = 0; reg < 4; ++reg) {
    c_out[MFMA_C_ROW(lane, reg) * 16 + MFMA_C_COL(lane)] = acc[reg];
  }
}

}  // namespace

extern "C" hipError_t ks_mfma_probe(void* c_out, const void* a, const void* b,
                                    hipStream_t stream) {
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, stream,
                     (float*)c_out, (const short*)a, (const short*)b);
  HIP_CHECK_KERNEL();
  return hipSuccess;
}
