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

// LDS swizzle on a byte offset (applied identically at stage and read,
// so any involution is correct; the choice only moves bank conflicts).
//
// SW=1 — guide T2 st_16x32: byte ^= ((row&7)<<4) (row = byte>>7).
//   Fixed the 126M-conflict 1-bit variant (PMC 2026-09-12), but the
//   fragment read pattern still 2-way conflicts: a quarter-wave reads
//   rows i and i+8 at the same column, and (i&7) == ((i+8)&7) maps both
//   to the SAME 16-B granule -> same bank pair.
// SW=2 — conflict-free for this kernel's reads: XOR granule bits 4..7
//   with (row>>1)&15 (byte bits 8..11, which the XOR does not touch, so
//   it stays an involution). For a quarter-wave (16 consecutive rows,
//   fixed column) the granule index becomes
//     g(i) = [c ^ (i>>1)] | ((i&1)<<3)
//   — a permutation of 0..15, i.e. the 16 lanes' 16-B reads tile all
//   64 banks exactly once.
template <int SW>
__device__ __forceinline__ int swz(int byte_off) {
  if constexpr (SW == 1) return byte_off ^ (((byte_off >> 7) & 7) << 4);
  if constexpr (SW == 2) return byte_off ^ (((byte_off >> 8) & 15) << 4);
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
template <int SW>
__device__ __forceinline__ void stage_half(
    const short* __restrict__ src,  // tile base (row 0, k 0 of this tile)
    long ld,                        // source leading dim (elements)
    int row0,                       // first row of the half-tile (0 or 128)
    short* lds_base,                // LDS base of the FULL 256x64 tile
    int wave, int lane) {
  // SW==1 source-address hoist: every chunk base is an 8-row multiple,
  // so the XOR key is ((lane>>3)&7)<<4 and the per-lane source offset
  // (row delta * ld + column) is loop- and chunk-invariant.
  long lane_src = 0;
  if constexpr (SW == 1) {
    const int lo = (lane * 16) ^ (((lane >> 3) & 7) << 4);
    lane_src = (long)(lo >> 7) * ld + ((lo & 127) >> 1);
  }
  // 16 KiB = 16 chunks of 1 KiB; 8 waves stage 2 chunks each
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    const int chunk = wave + i * 8;
    const int lin = row0 * 128 + chunk * 1024;        // wave-uniform
    const short* sp;
    if constexpr (SW == 1) {
      sp = src + (long)(lin >> 7) * ld + lane_src;
    } else {
      const int sb = swz<SW>(lin + lane * 16);        // per-lane source
      sp = src + (long)(sb >> 7) * ld + ((sb & 127) >> 1);
    }
    __builtin_amdgcn_global_load_lds(
        reinterpret_cast<const unsigned int*>(sp),
        reinterpret_cast<unsigned int*>(
            reinterpret_cast<char*>(lds_base) + lin),
        16, 0, 0);
  }
}

// read a 16-B bf16x8 from the (possibly swizzled) LDS tile
template <int SW>
__device__ __forceinline__ bf16x8_t lds_frag(const short* lds_base,
                                             int row, int col) {
  const int byte = swz<SW>(row * 128 + col * 2);
  return *reinterpret_cast<const bf16x8_t*>(
      reinterpret_cast<const char*>(lds_base) + byte);
}

// Stage one 16 KiB unit cut to the PHASE-GRANULAR read sets (the
// half-granular re-cut NEXT.md item 1 derives): KIND 0 is an A
// quarter-pair {rows r..r+63, 128+r..128+r+63} (phases 0-1 read r=0,
// phases 2-3 read r=64 of each wave's 128-row strip), KIND 1 is a B
// qn-strip {rows r+64w..r+64w+31, w<4} (even phases read r=0, odd r=32).
// Cutting stage units to these sets lets a unit be overwritten one or
// two phases after its last read — the prerequisite for the counted
// per-K-tile vmcnt (guide T3/T4) without a third LDS buffer.
template <int SW, int KIND>
__device__ __forceinline__ void stage_unit(
    const short* __restrict__ src, long ld, int row_add, short* lds_base,
    int wave, int lane) {
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    const int chunk = wave + i * 8;  // 16 chunks of 1 KiB (8 rows each)
    int row0;
    if constexpr (KIND == 0)
      row0 = ((chunk >> 3) << 7) + ((chunk & 7) << 3);
    else
      row0 = ((chunk >> 2) << 6) + ((chunk & 3) << 3);
    row0 += row_add;
    const int lin = row0 * 128;                        // wave-uniform
    const short* sp;
    if constexpr (SW == 1) {
      // same hoist as stage_half: chunk bases are 8-row multiples
      const int lo = (lane * 16) ^ (((lane >> 3) & 7) << 4);
      sp = src + (long)(row0 + (lo >> 7)) * ld + ((lo & 127) >> 1);
    } else {
      const int sb = swz<SW>(lin + lane * 16);         // per-lane source
      sp = src + (long)(sb >> 7) * ld + ((sb & 127) >> 1);
    }
    __builtin_amdgcn_global_load_lds(
        reinterpret_cast<const unsigned int*>(sp),
        reinterpret_cast<unsigned int*>(
            reinterpret_cast<char*>(lds_base) + lin),
        16, 0, 0);
  }
}

// SW==1 fast fragment reads: the st_16x32 XOR key is (row&7)<<4 and the
// fragment rows differ only by 16-multiples, so the key reduces to
// (lane&7)<<4 — invariant across phases and K-tiles. The whole swizzled
// per-lane byte offset hoists out of the loop (one value per ks), and
// each fragment read becomes a single ds_read at base + a compile-time
// immediate ((qm+m)*2048 / (qn+n)*2048). Verified exhaustively against
// swz<1>() for every (lane, wave, quadrant, ks) by test_gemm8_sim.py.
__device__ __forceinline__ int swz1_lane_low(int lane, int ks) {
  const int low = ks * 64 + ((lane >> 4) << 4) + (lane & 15) * 128;
  return low ^ ((lane & 7) << 4);
}

// SCHED 0: stage one half-tile per phase, drain (vmcnt 0) at phase 3 —
//   the last load has <1 phase of MFMA to hide under.
// SCHED 1: front-load the stages (A0+A1 at phase 0, B0+B1 at phase 1) with
//   a counted vmcnt(4) at phase 2 (A halves landed; FIFO retirement) and
//   vmcnt(0) at phase 3 covering only the B halves, which then have 2-3
//   phases in flight. (Step toward the guide's counted-vmcnt discipline —
//   its full 3-half-tiles-in-flight schedule needs half-granular read
//   ordering; this keeps the simple whole-tile flip.)
// ABLATE (perf decomposition only — results are WRONG for !=0, guide
// m233 methodology): 1 = no staging in the K-loop (ds_read+MFMA+barrier
// structure on stale LDS), 2 = no per-phase ds_reads (fragments read
// once; stage+MFMA+barriers), 3 = MFMA-only loop (matrix-pipe ceiling).
template <int SW, int SCHED = 0, int ABLATE = 0>
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
  // (2-K-tile unroll measured neutral-to-slightly-negative — ladder
  // addendum 4; the simple loop stays.)
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
      bf16x8_t b_frag[QN][2];
      if constexpr (ABLATE < 2) {
      if constexpr (SW == 1) {
#pragma unroll
        for (int ks = 0; ks < 2; ++ks) {
          const char* abase = reinterpret_cast<const char*>(ab) +
                              wm * 16384 + swz1_lane_low(lane, ks);
          const char* bbase = reinterpret_cast<const char*>(bb) +
                              wn * 8192 + swz1_lane_low(lane, ks);
#pragma unroll
          for (int m = 0; m < QM; ++m)
            a_frag[m][ks] = *reinterpret_cast<const bf16x8_t*>(
                abase + (qm + m) * 2048);
#pragma unroll
          for (int n = 0; n < QN; ++n)
            b_frag[n][ks] = *reinterpret_cast<const bf16x8_t*>(
                bbase + (qn + n) * 2048);
        }
      } else {
#pragma unroll
      for (int m = 0; m < QM; ++m) {
        const int row = wrow0 + (qm + m) * 16 + MFMA_RC_OF(lane);
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
          a_frag[m][ks] =
              lds_frag<SW>(ab, row, ks * 32 + ((lane >> 4) << 3));
      }
#pragma unroll
      for (int n = 0; n < QN; ++n) {
        const int row = wcol0 + (qn + n) * 16 + MFMA_RC_OF(lane);
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
          b_frag[n][ks] =
              lds_frag<SW>(bb, row, ks * 32 + ((lane >> 4) << 3));
      }
      }
      } else {
        // stale single fragment pair (timing structure only)
#pragma unroll
        for (int m = 0; m < QM; ++m)
#pragma unroll
          for (int ks = 0; ks < 2; ++ks)
            a_frag[m][ks] = lds_frag<SW>(ab, wrow0 + MFMA_RC_OF(lane), 0);
#pragma unroll
        for (int n = 0; n < QN; ++n)
#pragma unroll
          for (int ks = 0; ks < 2; ++ks)
            b_frag[n][ks] = lds_frag<SW>(bb, wcol0 + MFMA_RC_OF(lane), 0);
      }
      // ---- stage NEXT K-tile half-tiles (schedule-dependent) ----
      if (have_next && ABLATE != 1 && ABLATE != 3) {
        const int other = buf ^ 1;
        if constexpr (SCHED != 1) {
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
      // SCHED 3/4: no mid-phase barrier — in the SCHED-0 staging pattern
      // the stage always targets the inactive buffer, so phase cohesion
      // is a scheduling choice, not a correctness requirement.
      if constexpr (SCHED < 3) __builtin_amdgcn_s_barrier();
      // SCHED 6: no explicit wait/fence — the compiler tracks the
      // ds_read->MFMA dependencies itself and inserts minimal counted
      // lgkmcnt waits, software-pipelining reads into the MFMA burst.
      if constexpr (SCHED < 6) {
        asm volatile("s_waitcnt lgkmcnt(0)");
        __builtin_amdgcn_sched_barrier(0);  // MFMA must not hoist past (G#18)
      }
      if constexpr (SCHED < 4) __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int m = 0; m < QM; ++m)
#pragma unroll
        for (int n = 0; n < QN; ++n)
#pragma unroll
          for (int ks = 0; ks < 2; ++ks)
            acc[qm + m][qn + n] =
                mfma16x16x32(a_frag[m][ks], b_frag[n][ks], acc[qm + m][qn + n]);
      if constexpr (SCHED < 4) __builtin_amdgcn_s_setprio(0);
      // drain staged loads before the buffer flips. SCHED 1: counted
      // vmcnt(4) at phase 2 retires the A halves early (FIFO), leaving
      // only the B halves for the phase-3 wait with 2 phases in flight.
      if constexpr (SCHED == 1) {
        if (ph == 2 && have_next && ABLATE == 0) {
          asm volatile("s_waitcnt vmcnt(4)");
        }
      }
      if (ph == 3 && have_next && (ABLATE == 0 || ABLATE == 2)) {
        asm volatile("s_waitcnt vmcnt(0)");
      }
      // SCHED 5: barrier only at the K-tile boundary — within a tile
      // every read hits the pre-staged buffer and every stage hits the
      // other one, so the intermediate end-of-phase barriers
      // synchronize nothing; only the post-vmcnt(0) flip point needs
      // all waves' loads retired.
      if constexpr (SCHED >= 5) {
        if (ph == 3) __builtin_amdgcn_s_barrier();
      } else {
        __builtin_amdgcn_s_barrier();
      }
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

// Half-granular 8-phase schedule (guide T3+T4, the full counted-vmcnt
// discipline; NEXT.md round-3 item 1). One 16 KiB stage unit per phase,
// issued in future-consumption order, landing checked by ONE counted
// s_waitcnt vmcnt(4) per K-tile (never 0 in the main loop):
//
//   ph0: stage B-qn1(kt+1) -> other buf   (region last read kt-1 ph3)
//   ph1: stage A-qp1(kt+1) -> other buf   (last read kt-1 ph3)
//   ph2: stage A-qp0(kt+2) -> THIS buf    (last read kt ph1 — in-place)
//   ph3: stage B-qn0(kt+2) -> THIS buf    (last read kt ph2 — in-place)
//
// FIFO vmem retirement means vmcnt(4) at end of ph3 proves everything
// up to A-qp1(kt+1) landed (only the two kt+2 units may remain in
// flight), so every unit gets >=4 phases of memory latency hiding vs
// <1 phase for the SCHED-0 drain. In-place staging into the live
// buffer is safe because each phase ends with lgkmcnt(0)+s_barrier:
// all waves' ds_reads of the overwritten region completed one phase
// before the overwriting global_load_lds issues.
template <int SW>
__global__ __launch_bounds__(512, 1) void gemm8_hg_kernel(
    short* __restrict__ D, const short* __restrict__ A,
    const short* __restrict__ W, const int M, const int N, const int K) {
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
  const int wm = wave >> 2;
  const int wn = wave & 3;
  const int wrow0 = wm * 128;
  const int wcol0 = wn * 64;

  extern __shared__ short lds[];

  f32x4_t acc[FRAG_M][FRAG_N];
#pragma unroll
  for (int m = 0; m < FRAG_M; ++m)
#pragma unroll
    for (int n = 0; n < FRAG_N; ++n) acc[m][n] = f32x4_t{0.f, 0.f, 0.f, 0.f};

  const short* a_tile = A + (long)tile_m * K;
  const short* w_tile = W + (long)tile_n * K;
  const int ktiles = K / BK;

  // ---- prologue: tile 0 fully (consumption order), then the two tile-1
  // units the steady-state loop can't cover (they'd be kt-1 ph2/ph3) ----
  stage_unit<SW, 0>(a_tile, K, 0, A_BUF(0), wave, lane);    // A-qp0(0)
  stage_unit<SW, 1>(w_tile, K, 0, B_BUF(0), wave, lane);    // B-qn0(0)
  stage_unit<SW, 1>(w_tile, K, 32, B_BUF(0), wave, lane);   // B-qn1(0)
  stage_unit<SW, 0>(a_tile, K, 64, A_BUF(0), wave, lane);   // A-qp1(0)
  if (ktiles > 1) {
    stage_unit<SW, 0>(a_tile + BK, K, 0, A_BUF(1), wave, lane);
    stage_unit<SW, 1>(w_tile + BK, K, 0, B_BUF(1), wave, lane);
    asm volatile("s_waitcnt vmcnt(4)");  // tile-0 units landed (FIFO)
  } else {
    asm volatile("s_waitcnt vmcnt(0)");
  }
  __builtin_amdgcn_s_barrier();

  for (int kt = 0; kt < ktiles; ++kt) {
    const int buf = kt & 1;
    const short* ab = A_BUF(buf);
    const short* bb = B_BUF(buf);
    const bool have1 = kt + 1 < ktiles;
    const bool have2 = kt + 2 < ktiles;
    const short* a1 = a_tile + (kt + 1) * BK;
    const short* w1 = w_tile + (kt + 1) * BK;
    const short* a2 = a_tile + (kt + 2) * BK;
    const short* w2 = w_tile + (kt + 2) * BK;

#pragma unroll
    for (int ph = 0; ph < 4; ++ph) {
      const int qm = (ph >> 1) * QM;
      const int qn = (ph & 1) * QN;
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
      // one stage unit per phase, future-consumption order
      if (ph == 0 && have1)
        stage_unit<SW, 1>(w1, K, 32, B_BUF(buf ^ 1), wave, lane);
      else if (ph == 1 && have1)
        stage_unit<SW, 0>(a1, K, 64, A_BUF(buf ^ 1), wave, lane);
      else if (ph == 2 && have2)
        stage_unit<SW, 0>(a2, K, 0, A_BUF(buf), wave, lane);
      else if (ph == 3 && have2)
        stage_unit<SW, 1>(w2, K, 0, B_BUF(buf), wave, lane);
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
      if (ph == 3 && have1) {
        if (have2) {
          // only the two kt+2 units (4 loads) may stay in flight
          asm volatile("s_waitcnt vmcnt(4)");
        } else {
          // no ph2/ph3 stages were issued: vmcnt(4) would pass with the
          // kt+1 units still in flight — drain instead (epilogue only)
          asm volatile("s_waitcnt vmcnt(0)");
        }
      }
      __builtin_amdgcn_s_barrier();
    }
  }

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

// Read-ahead schedule (SCHED 7): SCHED-5 sync structure (one barrier
// per K-tile, no setprio) + one-phase ds_read readahead. Phase p's MFMA
// runs while phase p+1's 12 ds_reads are in flight; the counted
// s_waitcnt lgkmcnt(12) before the MFMA retires exactly phase p's reads
// (LDS ops retire in order) and leaves p+1's outstanding. Doubles the
// live fragment set (2 x 48 VGPRs) so the allocator moves the 128-VGPR
// accumulator file to AGPRs — exactly what they exist for.
template <int SW>
__global__ __launch_bounds__(512, 1) void gemm8_ra_kernel(
    short* __restrict__ D, const short* __restrict__ A,
    const short* __restrict__ W, const int M, const int N, const int K) {
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
  const int wm = wave >> 2;
  const int wn = wave & 3;
  const int wrow0 = wm * 128;
  const int wcol0 = wn * 64;

  extern __shared__ short lds[];

  f32x4_t acc[FRAG_M][FRAG_N];
#pragma unroll
  for (int m = 0; m < FRAG_M; ++m)
#pragma unroll
    for (int n = 0; n < FRAG_N; ++n) acc[m][n] = f32x4_t{0.f, 0.f, 0.f, 0.f};

  const short* a_tile = A + (long)tile_m * K;
  const short* w_tile = W + (long)tile_n * K;
  const int ktiles = K / BK;

  stage_half<SW>(a_tile, K, 0, A_BUF(0), wave, lane);
  stage_half<SW>(a_tile, K, 128, A_BUF(0), wave, lane);
  stage_half<SW>(w_tile, K, 0, B_BUF(0), wave, lane);
  stage_half<SW>(w_tile, K, 128, B_BUF(0), wave, lane);
  asm volatile("s_waitcnt vmcnt(0)");
  __builtin_amdgcn_s_barrier();

  for (int kt = 0; kt < ktiles; ++kt) {
    const int buf = kt & 1;
    const short* ab = A_BUF(buf);
    const short* bb = B_BUF(buf);
    const int nxt = kt + 1;
    const short* a_next = a_tile + nxt * BK;
    const short* w_next = w_tile + nxt * BK;
    const bool have_next = nxt < ktiles;

    // double fragment set: [ph & 1]
    bf16x8_t a_frag[2][QM][2];
    bf16x8_t b_frag[2][QN][2];

    // reads for ph0 (buffer just validated by the tile-boundary barrier)
#pragma unroll
    for (int m = 0; m < QM; ++m)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
        a_frag[0][m][ks] = lds_frag<SW>(
            ab, wrow0 + m * 16 + MFMA_RC_OF(lane),
            ks * 32 + ((lane >> 4) << 3));
#pragma unroll
    for (int n = 0; n < QN; ++n)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
        b_frag[0][n][ks] = lds_frag<SW>(
            bb, wcol0 + n * 16 + MFMA_RC_OF(lane),
            ks * 32 + ((lane >> 4) << 3));

#pragma unroll
    for (int ph = 0; ph < 4; ++ph) {
      const int qm = (ph >> 1) * QM;
      const int qn = (ph & 1) * QN;
      const int cur = ph & 1;
      if (ph < 3) {
        const int nqm = ((ph + 1) >> 1) * QM;
        const int nqn = ((ph + 1) & 1) * QN;
#pragma unroll
        for (int m = 0; m < QM; ++m)
#pragma unroll
          for (int ks = 0; ks < 2; ++ks)
            a_frag[cur ^ 1][m][ks] = lds_frag<SW>(
                ab, wrow0 + (nqm + m) * 16 + MFMA_RC_OF(lane),
                ks * 32 + ((lane >> 4) << 3));
#pragma unroll
        for (int n = 0; n < QN; ++n)
#pragma unroll
          for (int ks = 0; ks < 2; ++ks)
            b_frag[cur ^ 1][n][ks] = lds_frag<SW>(
                bb, wcol0 + (nqn + n) * 16 + MFMA_RC_OF(lane),
                ks * 32 + ((lane >> 4) << 3));
      }
      if (have_next) {
        const int other = buf ^ 1;
        if (ph == 0) stage_half<SW>(a_next, K, 0, A_BUF(other), wave, lane);
        else if (ph == 1)
          stage_half<SW>(a_next, K, 128, A_BUF(other), wave, lane);
        else if (ph == 2)
          stage_half<SW>(w_next, K, 0, B_BUF(other), wave, lane);
        else stage_half<SW>(w_next, K, 128, B_BUF(other), wave, lane);
      }
      // retire exactly this phase's reads; leave the 12 readahead
      // ds_reads in flight under the MFMA burst
      if (ph < 3)
        asm volatile("s_waitcnt lgkmcnt(12)");
      else
        asm volatile("s_waitcnt lgkmcnt(0)");
      __builtin_amdgcn_sched_barrier(0);
#pragma unroll
      for (int m = 0; m < QM; ++m)
#pragma unroll
        for (int n = 0; n < QN; ++n)
#pragma unroll
          for (int ks = 0; ks < 2; ++ks)
            mfma16x16x32_agpr(a_frag[cur][m][ks], b_frag[cur][n][ks],
                              acc[qm + m][qn + n]);
      if (ph == 3) {
        if (have_next) asm volatile("s_waitcnt vmcnt(0)");
        __builtin_amdgcn_s_barrier();
      }
    }
  }

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

// use_swizzle: 0 = none, 1 = T2 st_16x32 (3-bit), 2 = conflict-free 4-bit
// (row>>1). Schedule via KS_GEMM8_SCHED: 0/1 = whole/front-loaded
// half-tile double buffer, 2 = half-granular counted-vmcnt pipeline.
extern "C" hipError_t ks_gemm8(void* d, const void* a, const void* w, int M,
                               int N, int K, int use_swizzle,
                               hipStream_t stream) {
  if (M % BM || N % BN || K % BK) return hipErrorInvalidValue;
  if (use_swizzle < 0 || use_swizzle > 2) return hipErrorInvalidValue;
  dim3 grid(N / BN, M / BM);
  const size_t lds_bytes = 4 * BM * BK * sizeof(short);  // 128 KiB
  using Kfn = void (*)(short*, const short*, const short*, int, int, int);
  // [swizzle][sched] dispatch; sched 2 = half-granular kernel,
  // 3 = no mid-phase barrier, 4 = 3 + no setprio, 5 = 4 + barrier only
  // at the K-tile boundary, 6 = 5 + compiler-scheduled waits,
  // 7 = 5 + one-phase ds_read readahead (counted lgkmcnt, AGPR acc)
  static const Kfn table[3][8] = {
      {gemm8_kernel<0, 0>, gemm8_kernel<0, 1>, gemm8_hg_kernel<0>,
       gemm8_kernel<0, 3>, gemm8_kernel<0, 4>, gemm8_kernel<0, 5>,
       gemm8_kernel<0, 6>, gemm8_ra_kernel<0>},
      {gemm8_kernel<1, 0>, gemm8_kernel<1, 1>, gemm8_hg_kernel<1>,
       gemm8_kernel<1, 3>, gemm8_kernel<1, 4>, gemm8_kernel<1, 5>,
       gemm8_kernel<1, 6>, gemm8_ra_kernel<1>},
      {gemm8_kernel<2, 0>, gemm8_kernel<2, 1>, gemm8_hg_kernel<2>,
       gemm8_kernel<2, 3>, gemm8_kernel<2, 4>, gemm8_kernel<2, 5>,
       gemm8_kernel<2, 6>, gemm8_ra_kernel<2>},
  };
  // dynamic LDS above the 64 KiB default needs an explicit opt-in
  static bool attr_set = [] {
    for (int s = 0; s < 3; ++s)
      for (int j = 0; j < 8; ++j)
        (void)hipFuncSetAttribute((const void*)table[s][j],
                                  hipFuncAttributeMaxDynamicSharedMemorySize,
                                  4 * BM * BK * sizeof(short));
    return true;
  }();
  (void)attr_set;
  static const int sched = [] {
    const char* e = getenv("KS_GEMM8_SCHED");
    const int s = e ? atoi(e) : 5;  // 5 measured best (profiles/gemm8_ladder.md)
    return (s >= 0 && s <= 7) ? s : 5;
  }();
  // KS_GEMM8_ABLATE: m233-style decomposition (WRONG results; perf only)
  static const int ablate = [] {
    const char* e = getenv("KS_GEMM8_ABLATE");
    const int s = e ? atoi(e) : 0;
    return (s >= 1 && s <= 3) ? s : 0;
  }();
  if (ablate) {
    // decomposition runs against the best schedule (5)
    static const Kfn abl[4] = {nullptr, gemm8_kernel<1, 5, 1>,
                               gemm8_kernel<1, 5, 2>, gemm8_kernel<1, 5, 3>};
    static bool abl_attr = [] {
      for (int i = 1; i < 4; ++i)
        (void)hipFuncSetAttribute((const void*)abl[i],
                                  hipFuncAttributeMaxDynamicSharedMemorySize,
                                  4 * BM * BK * sizeof(short));
      return true;
    }();
    (void)abl_attr;
    hipLaunchKernelGGL(abl[ablate], grid, dim3(512), lds_bytes, stream,
                       (short*)d, (const short*)a, (const short*)w, M, N, K);
    HIP_CHECK_KERNEL();
    return hipSuccess;
  }
  hipLaunchKernelGGL(table[use_swizzle][sched], grid, dim3(512), lds_bytes,
                     stream, (short*)d, (const short*)a, (const short*)w, M,
                     N, K);
  HIP_CHECK_KERNEL();
  return hipSuccess;
}
