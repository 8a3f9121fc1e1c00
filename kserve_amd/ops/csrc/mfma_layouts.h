// MFMA fragment layouts for v_mfma_f32_16x16x32_bf16 on gfx950.
//
// C/D layout is hardware-verified per the CDNA guide (§3, learn_hip m89/m91):
//   C[row][col] with col = lane & 15, row = (lane >> 4) * 4 + reg.
//
// A/B layouts below are the standard CDNA mapping (A: row = lane & 15,
// k = (lane >> 4) * 8 + j; B: col = lane & 15, same k) — verified on-device
// by the ks_mfma_probe kernel (tests/test_gpu_ops.py::test_mfma_probe).
// If the probe ever fails, fix K_OF_J here and everything downstream follows.
#pragma once

#include "common.h"

typedef __bf16 bf16x8_t __attribute__((ext_vector_type(8)));
typedef float f32x4_t __attribute__((ext_vector_type(4)));

// k index of element j (0..7) for a lane: k = (lane>>4)*8 + j
#define MFMA_K_OF(lane, j) ((((lane) >> 4) << 3) + (j))
// A row / B col for a lane
#define MFMA_RC_OF(lane) ((lane) & 15)
// C/D: col = lane&15, row = (lane>>4)*4 + reg
#define MFMA_C_COL(lane) ((lane) & 15)
#define MFMA_C_ROW(lane, reg) ((((lane) >> 4) << 2) + (reg))

__device__ __forceinline__ f32x4_t mfma16x16x32(bf16x8_t a, bf16x8_t b,
                                                f32x4_t c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

// AGPR-accumulator variant: the "a" constraint pins the accumulator to
// the accumulation register file, freeing ~4 VGPRs per fragment for
// deep-pipelined kernels the allocator otherwise cannot fit (the
// builtin's heuristic keeps accumulators in VGPRs even at full
// pressure). acc is read back to VGPRs only at the epilogue.
__device__ __forceinline__ void mfma16x16x32_agpr(bf16x8_t a, bf16x8_t b,
                                                  f32x4_t& acc) {
  __asm__ volatile("v_mfma_f32_16x16x32_bf16 %0, %1, %2, %0"
                   : "+a"(acc)
                   : "v"(a), "v"(b));
}
