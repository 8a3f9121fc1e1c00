// 8-phase 256x256 MFMA GEMM (EXPERIMENTAL — round-2 candidate).
//
// Plain-HIP port of the verified 8-phase schedule from the CDNA4 guide
// ("The 256-sq 8-phase template", measured 1563 TF @4k / 1728 @8k on
// MI355X with the st_16x32 swizzle): 256x256 output tile, BK=64, 8 waves
// (2M x 4N), double-buffered LDS staged with global_load_lds (16 B), one
// C-quadrant (4Mx2N fragments) x K=64 per phase, counted vmcnt at phases
// 4/8 only, s_setprio(1) around the MFMA burst.
//
// Status: UNVALIDATED ON HARDWARE in round 1 (written after the GPU
// budget was spent). Gated off everywhere: the dispatcher never selects
// it unless KS_GEMM8=1, and its GPU test requires KS_GEMM8=1. Round 2:
// run tests/test_gpu_ops.py -k gemm8 + tools/gemm8_bench.py first thing.
//
// Computes D[M,N] = A[M,K] @ W[N,K]^T (torch F.linear convention), bf16
// in / bf16 out, fp32 accumulate. M,N,K must be multiples of 256/256/64.
#include "common.h"
#include "mfma_layouts.h"

namespace {

constexpr int BM = 256, BN = 256, BK = 64;
constexpr int WM = 2, WN = 4;           // wave grid (8 waves, 512 threads)
constexpr int FRAG_M = 8, FRAG_N = 4;   // 16x16 fragments per wave
// per-quadrant fragment split: 4M x 2N, four quadrants per K-tile
constexpr int QM = 4, QN = 2;

// st_16x32 swizzle on a byte offset: XOR three row bits into the 16-B
// chunk index (guide T2 recipe: byte ^= ((row&7)<<4), row = byte>>7 for a
// 128-B row). Spreads a 16-lane column-slice read across all 32 banks —
// the 1-bit (bit9->bit5) variant of round 1 left ~4-8-way conflicts
// (measured 126M SQ_LDS_BANK_CONFLICT at 4096^3; PMC 2026-09-12).
// Involution: bits >=7 are untouched, so applying twice is identity.
template <bool SW>
__device__ __forceinline__ int swz(int byte_off) {
  if constexpr (SW) return byte_off ^ (((byte_off >> 7) & 7) << 4);
  return byte_off;
}

// Stage one half-tile (128 rows x 64 K = 16 KB) of a row-major [rows,K]
// bf16 source into LDS with 2 x global_load_lds(16B) per wave.
//
// global_load_lds semantics (guide m104/m108): the LDS destination is the
// WAVE-UNIFORM base + lane*16; the GLOBAL source is per-lane. One
// instruction therefore stages a contiguous 1 KiB LDS chunk. The st_16x32
// swizzle is realised by permuting the per-lane global source byte while
// the LDS write stays linear (HipKittens' pre-swizzled-source pattern):
// LDS[lin] must hold data[swz(lin)] (swz is an XOR involution).
template <bool SW>
__device__ __forceinline__ void stage_half(
    const short* __restrict__ src,  // tile base (row 0, k 0 of this tile)
    long ld,                        // source leading dim (elements)
    int row0,                       // first row of the half-tile (0 or 128)
    short* lds_base,                // LDS base of the FULL 256x64 tile
    int wave, int lane) {
  // 16 KiB = 16 chunks of 1 KiB; 8 waves stage 2 chunks each
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    const int chunk = wave + i * 8;
    const int lin = row0 * 128 + chunk * 1024;        // wave-uniform
    const int sb = swz<SW>(lin + lane * 16);          // per-lane source
    const int srow = sb >> 7;                         // /128 bytes per row
    const int scol = (sb & 127) >> 1;                 // byte -> element
    __builtin_amdgcn_global_load_lds(
        reinterpret_cast<const unsigned int*>(src + (long)srow * ld + scol),
        reinterpret_cast<unsigned int*>(
            reinterpret_cast<char*>(lds_base) + lin),
        16, 0, 0);
  }
}

// read a 16-B bf16x8 from the (possibly swizzled) LDS tile
template <bool SW>
__device__ __forceinline__ bf16x8_t lds_frag(const short* lds_base,
                                             int row, int col) {
  const int byte = swz<SW>(row * 128 + col * 2);
  return *reinterpret_cast<const bf16x8_t*>(
      reinterpret_cast<const char*>(lds_base) + byte);
}

// SCHED 0: stage one half-tile per phase, drain (vmcnt 0) at phase 3 —
//   the last load has <1 phase of MFMA to hide under.
// SCHED 1: front-load the stages (A0+A1 at phase 0, B0+B1 at phase 1) with
//   a counted vmcnt(4) at phase 2 (A halves landed; FIFO retirement) and
//   vmcnt(0) at phase 3 covering only the B halves, which then have 2-3
//   phases in flight. (Step toward the guide's counted-vmcnt discipline —
//   its full 3-half-tiles-in-flight schedule needs half-granular read
//   ordering; this keeps the simple whole-tile flip.)
template <bool SW, int SCHED = 0>
__global__ __launch_bounds__(512, 1) void gemm8_kernel(
    short* __restrict__ D,        // [M, N] bf16
    const short* __restrict__ A,  // [M, K] bf16
    const short* __restrict__ W,  // [N, K] bf16
    const int M, const int N, const int K) {
  // bijective XCD-aware workgroup swizzle (guide ERRATA #11)
  const int nwg = gridDim.x * gridDim.y;
  const int orig = blockIdx.y * gridDim.x + blockIdx.x;
  const int q = nwg / 8, r = nwg % 8;
  const int xcd = orig % 8, idx = orig / 8;
  const int wgid =
      (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  const int tiles_n = N / BN;
  const int tile_m = (wgid / tiles_n) * BM;
  const int tile_n = (wgid % tiles_n) * BN;

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int wm = wave >> 2;          // 0..1
  const int wn = wave & 3;           // 0..3
  const int wrow0 = wm * 128;        // wave's first row within the tile
  const int wcol0 = wn * 64;

  // LDS: A double-buffered [2][256][64], B the same — 128 KiB total
  // (no pointer arrays: addrspace(3) pointers in aggregates miscompile)
  extern __shared__ short lds[];
#define A_BUF(i) (lds + (i) * BM * BK)
#define B_BUF(i) (lds + (2 + (i)) * BM * BK)

  f32x4_t acc[FRAG_M][FRAG_N];
#pragma unroll
  for (int m = 0; m < FRAG_M; ++m)
#pragma unroll
    for (int n = 0; n < FRAG_N; ++n) acc[m][n] = f32x4_t{0.f, 0.f, 0.f, 0.f};

  const short* a_tile = A + (long)tile_m * K;
  const short* w_tile = W + (long)tile_n * K;

  // ---- prologue: stage K-tile 0 fully ----
  stage_half<SW>(a_tile, K, 0, A_BUF(0), wave, lane);
  stage_half<SW>(a_tile, K, 128, A_BUF(0), wave, lane);
  stage_half<SW>(w_tile, K, 0, B_BUF(0), wave, lane);
  stage_half<SW>(w_tile, K, 128, B_BUF(0), wave, lane);
  asm volatile("s_waitcnt vmcnt(0)");
  __builtin_amdgcn_s_barrier();

  const int ktiles = K / BK;
  for (int kt = 0; kt < ktiles; ++kt) {
    const int buf = kt & 1;
    const short* ab = A_BUF(buf);
    const short* bb = B_BUF(buf);
    const int nxt = kt + 1;
    const short* a_next = a_tile + nxt * BK;
    const short* w_next = w_tile + nxt * BK;
    const bool have_next = nxt < ktiles;

    // 4 phases per K-tile: one C-quadrant each, K=64 (2 mfma k-steps)
#pragma unroll
    for (int ph = 0; ph < 4; ++ph) {
      const int qm = (ph >> 1) * QM;  // fragment-row offset of quadrant
      const int qn = (ph & 1) * QN;
      // ---- ds-read the quadrant's A/B fragments (12 x ds_read_b128) ----
      bf16x8_t a_frag[QM][2];
#pragma unroll
      for (int m = 0; m < QM; ++m) {
        const int row = wrow0 + (qm + m) * 16 + MFMA_RC_OF(lane);
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
          a_frag[m][ks] =
              lds_frag<SW>(ab, row, ks * 32 + ((lane >> 4) << 3));
      }
      bf16x8_t b_frag[QN][2];
#pragma unroll
      for (int n = 0; n < QN; ++n) {
        const int row = wcol0 + (qn + n) * 16 + MFMA_RC_OF(lane);
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
          b_frag[n][ks] =
              lds_frag<SW>(bb, row, ks * 32 + ((lane >> 4) << 3));
      }
      // ---- stage NEXT K-tile half-tiles (schedule-dependent) ----
      if (have_next) {
        const int other = buf ^ 1;
        if constexpr (SCHED == 0) {
          if (ph == 0) stage_half<SW>(a_next, K, 0, A_BUF(other), wave, lane);
          else if (ph == 1)
            stage_half<SW>(a_next, K, 128, A_BUF(other), wave, lane);
          else if (ph == 2)
            stage_half<SW>(w_next, K, 0, B_BUF(other), wave, lane);
          else stage_half<SW>(w_next, K, 128, B_BUF(other), wave, lane);
        } else {
          if (ph == 0) {
            stage_half<SW>(a_next, K, 0, A_BUF(other), wave, lane);
            stage_half<SW>(a_next, K, 128, A_BUF(other), wave, lane);
          } else if (ph == 1) {
            stage_half<SW>(w_next, K, 0, B_BUF(other), wave, lane);
            stage_half<SW>(w_next, K, 128, B_BUF(other), wave, lane);
          }
        }
      }
      __builtin_amdgcn_s_barrier();
      asm volatile("s_waitcnt lgkmcnt(0)");
      __builtin_amdgcn_sched_barrier(0);  // MFMA must not hoist past (G#18)
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int m = 0; m < QM; ++m)
#pragma unroll
        for (int n = 0; n < QN; ++n)
#pragma unroll
          for (int ks = 0; ks < 2; ++ks)
            acc[qm + m][qn + n] =
                mfma16x16x32(a_frag[m][ks], b_frag[n][ks], acc[qm + m][qn + n]);
      __builtin_amdgcn_s_setprio(0);
      // drain staged loads before the buffer flips. SCHED 1: counted
      // vmcnt(4) at phase 2 retires the A halves early (FIFO), leaving
      // only the B halves for the phase-3 wait with 2 phases in flight.
      if constexpr (SCHED == 1) {
        if (ph == 2 && have_next) {
          asm volatile("s_waitcnt vmcnt(4)");
        }
      }
      if (ph == 3 && have_next) {
        asm volatile("s_waitcnt vmcnt(0)");
      }
      __builtin_amdgcn_s_barrier();
    }
  }

  // ---- epilogue: D write (C layout: col=lane&15, row=(lane>>4)*4+reg) ----
#pragma unroll
  for (int m = 0; m < FRAG_M; ++m) {
#pragma unroll
    for (int n = 0; n < FRAG_N; ++n) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int row = tile_m + wrow0 + m * 16 + MFMA_C_ROW(lane, reg);
        const int col = tile_n + wcol0 + n * 16 + MFMA_C_COL(lane);
        D[(long)row * N + col] = float_to_bf16_bits(acc[m][n][reg]);
      }
    }
  }
}

}  // namespace

extern "C" hipError_t ks_gemm8(void* d, const void* a, const void* w, int M,
                               int N, int K, int use_swizzle,
                               hipStream_t stream) {
  if (M % BM || N % BN || K % BK) return hipErrorInvalidValue;
  dim3 grid(N / BN, M / BM);
  const size_t lds_bytes = 4 * BM * BK * sizeof(short);  // 128 KiB
  // dynamic LDS above the 64 KiB default needs an explicit opt-in
  static bool attr_set = [] {
    (void)hipFuncSetAttribute((const void*)&gemm8_kernel<true, 0>,
                        hipFuncAttributeMaxDynamicSharedMemorySize,
                        4 * BM * BK * sizeof(short));
    (void)hipFuncSetAttribute((const void*)&gemm8_kernel<false, 0>,
                        hipFuncAttributeMaxDynamicSharedMemorySize,
                        4 * BM * BK * sizeof(short));
    (void)hipFuncSetAttribute((const void*)&gemm8_kernel<true, 1>,
                        hipFuncAttributeMaxDynamicSharedMemorySize,
                        4 * BM * BK * sizeof(short));
    (void)hipFuncSetAttribute((const void*)&gemm8_kernel<false, 1>,
                        hipFuncAttributeMaxDynamicSharedMemorySize,
                        4 * BM * BK * sizeof(short));
    return true;
  }();
  (void)attr_set;
  static const int sched = [] {
    const char* e = getenv("KS_GEMM8_SCHED");
    return (e && e[0] == '1') ? 1 : 0;
  }();
  if (use_swizzle) {
    if (sched == 1)
      hipLaunchKernelGGL((gemm8_kernel<true, 1>), grid, dim3(512), lds_bytes,
                         stream, (short*)d, (const short*)a, (const short*)w,
                         M, N, K);
    else
      hipLaunchKernelGGL((gemm8_kernel<true, 0>), grid, dim3(512), lds_bytes,
                         stream, (short*)d, (const short*)a, (const short*)w,
                         M, N, K);
  } else {
    if (sched == 1)
      hipLaunchKernelGGL((gemm8_kernel<false, 1>), grid, dim3(512),
                         lds_bytes, stream, (short*)d, (const short*)a,
                         (const short*)w, M, N, K);
    else
      hipLaunchKernelGGL((gemm8_kernel<false, 0>), grid, dim3(512),
                         lds_bytes, stream, (short*)d, (const short*)a,
                         (const short*)w, M, N, K);
  }
  HIP_CHECK_KERNEL();
  return hipSuccess;
}
