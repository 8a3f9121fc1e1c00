// Skinny GEMM for decode-path projections: C[N][M] = X[N][K] @ W[M][K]^T,
// N <= 256 tokens (weight-streaming regime where hipBLASLt reaches only
// ~1-4 TB/s on gfx950; measured in tools/gemm_bench.py).
//
// m97-shaped CDNA4 structure (guide §5): both operands staged through LDS
// via global_load_lds (16 B/lane) with PRE-SWIZZLED global sources so the
// linear LDS destination equals the XOR-swizzled layout that makes the
// ds_read_b128 fragment reads bank-conflict-free (rule #21: linear dest +
// inverse-swizzled source + swizzled read share one involution). Two
// barriers per K-step; cross-workgroup overlap hides HBM latency (m97
// finding: implicit wave-level overlap ~ explicit pipelining).
//
// The X tile is shared by all 4 waves (each wave owns 16 output features,
// all token tiles). K-split fills the chip for small-M projections; the
// splits write disjoint f32 partials reduced by a second kernel (no atomic
// RMW contention).
#include "common.h"
#include "mfma_layouts.h"

namespace {

constexpr int BK = 64;           // K per stage step
constexpr int FEAT_PER_WAVE = 16;
constexpr int NWAVES = 4;        // 64 features per workgroup

// chunk = 8 bf16 = 16 B. Row-chunk swizzle: kc_lds = kc_src ^ (row & 7).
__device__ __forceinline__ int lds_row_idx(int row, int kc) {
  return (row * (BK / 8) + (kc ^ (row & 7))) * 8;  // shorts
}

template <int NT, bool SPLIT>
__global__ __launch_bounds__(256) void skinny_gemm_kernel(
    short* __restrict__ out,      // [N][M] bf16 (!SPLIT)
    short* __restrict__ out_ws,   // [nsplit][N][M] bf16 partials (SPLIT)
    const short* __restrict__ x,  // [N][K] row stride xs
    const short* __restrict__ w,  // [M][K]
    const int M, const int K, const int N, const long xs,
    const int k_per_split) {
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int f_base = blockIdx.x * (NWAVES * FEAT_PER_WAVE);
  const int k_lo = blockIdx.y * k_per_split;
  const int k_hi = min(K, k_lo + k_per_split);

  // LDS: double-buffered {X tile [NT*16][BK] | W tile [64][BK]}, swizzled,
  // + one 1 KB scratch slot for stage-count padding
  constexpr int X_CHUNKS = NT * 16 * (BK / 8);  // 16B chunks
  constexpr int W_CHUNKS = 64 * (BK / 8);
  constexpr int TOTAL_CHUNKS = X_CHUNKS + W_CHUNKS;
  constexpr int NINSTR = TOTAL_CHUNKS / 64;          // wg-wide 1 KB instrs
  constexpr int NI = (NINSTR + NWAVES - 1) / NWAVES;  // per-wave, padded
  constexpr int TILE_SHORTS = (NT * 16 + 64) * BK;
  __shared__ short tile[2][TILE_SHORTS];
  __shared__ short scratch[512];  // dump target for padding instrs

  f32x4_t acc[NT];
#pragma unroll
  for (int t = 0; t < NT; ++t) acc[t] = f32x4_t{0.f, 0.f, 0.f, 0.f};

  const int m_local = lane & 15;
  const int kgrp = lane >> 4;

  // stage buf with the BK-slice at k-offset k0; every wave issues exactly
  // NI instructions so the counted s_waitcnt below is uniform (T4)
  auto stage = [&](int buf, int k0) {
#pragma unroll
    for (int ii = 0; ii < NI; ++ii) {
      const int i = wave + ii * NWAVES;
      const int g = i * 64 + lane;
      const short* src;
      short* dst;
      if (i < NINSTR) {
        dst = &tile[buf][i * 64 * 8];
        if (g < X_CHUNKS) {
          int row = g >> 3;  // token row
          const int kc_src = (g & 7) ^ (row & 7);
          if (row >= N) row = N - 1;  // clamp padded token rows
          src = x + (long)row * xs + (kc_src << 3) + k0;
        } else {
          const int gw = g - X_CHUNKS;
          const int m = gw >> 3;
          const int kc_src = (gw & 7) ^ (m & 7);
          src = w + (long)(f_base + m) * K + (kc_src << 3) + k0;
        }
      } else {  // padding instr keeps per-wave vmcnt uniform
        dst = scratch;
        src = x;
      }
      __builtin_amdgcn_global_load_lds(
          reinterpret_cast<const unsigned int*>(src),
          reinterpret_cast<unsigned int*>(dst), 16, 0, 0);
    }
  };

  // s_waitcnt imm waiting vmcnt==NI, ignore lgkm/exp (gfx9 encoding:
  // vmcnt[3:0]|vmcnt[5:4]<<14, expcnt[6:4], lgkmcnt[11:8])
  constexpr int WAIT_NI =
      (0xF << 8) | (0x7 << 4) | (NI & 0xF) | (((NI >> 4) & 0x3) << 14);
  constexpr int WAIT_0 = (0xF << 8) | (0x7 << 4);

  // prologue: stage first slice into buf 0
  int cur = 0;
  stage(0, k_lo);
  for (int k0 = k_lo; k0 < k_hi; k0 += BK) {
    const int k_next = k0 + BK;
    if (k_next < k_hi) {
      stage(cur ^ 1, k_next);
      __builtin_amdgcn_s_waitcnt(WAIT_NI);  // buf[cur] landed; next in flight
    } else {
      __builtin_amdgcn_s_waitcnt(WAIT_0);
    }
    __builtin_amdgcn_sched_barrier(0);
    __builtin_amdgcn_s_barrier();  // raw: no vmcnt drain (counted pipeline)
    // ---- fragments + MFMA from buf[cur] ----
    const int wf = NT * 16 + wave * FEAT_PER_WAVE;  // W row base in tile
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      const int kc = kk * 4 + kgrp;
      bf16x8_t b = *reinterpret_cast<bf16x8_t*>(
          &tile[cur][lds_row_idx(wf + m_local, kc)]);
#pragma unroll
      for (int t = 0; t < NT; ++t) {
        bf16x8_t a = *reinterpret_cast<bf16x8_t*>(
            &tile[cur][lds_row_idx(t * 16 + m_local, kc)]);
        acc[t] = mfma16x16x32(a, b, acc[t]);
      }
    }
    __builtin_amdgcn_s_barrier();  // readers done before buf[cur] is reused
    cur ^= 1;
  }

  // ---- epilogue: C row = token, col = feature ----
  const int feat = f_base + wave * FEAT_PER_WAVE + MFMA_C_COL(lane);
#pragma unroll
  for (int t = 0; t < NT; ++t) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int tokn = t * 16 + MFMA_C_ROW(lane, reg);
      if (tokn < N) {
        if constexpr (SPLIT) {
          // bf16 partials: each is a full K/nsplit-length dot in f32 first,
          // so the rounding is one bf16 quantization per partial
          out_ws[((long)blockIdx.y * N + tokn) * M + feat] =
              float_to_bf16_bits(acc[t][reg]);
        } else {
          out[(long)tokn * M + feat] = float_to_bf16_bits(acc[t][reg]);
        }
      }
    }
  }
}

// sum bf16 partials (f32 accumulate) and store bf16
__global__ void reduce_ws_kernel(short* __restrict__ out,
                                 const short* __restrict__ ws, const long nm,
                                 const int nsplit) {
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < nm;
       i += (long)gridDim.x * blockDim.x) {
    float acc = 0.f;
    for (int s = 0; s < nsplit; ++s)
      acc += bf16_bits_to_float(ws[s * nm + i]);
    out[i] = float_to_bf16_bits(acc);
  }
}

template <int NT>
void launch_nt(short* out, short* ws, const short* x, const short* w, int M,
               int K, int N, long xs, int nsplit, int k_per_split,
               hipStream_t stream) {
  dim3 grid(M / (NWAVES * FEAT_PER_WAVE), nsplit);
  if (nsplit > 1) {
    hipLaunchKernelGGL((skinny_gemm_kernel<NT, true>), grid, dim3(256), 0,
                       stream, out, ws, x, w, M, K, N, xs, k_per_split);
  } else {
    hipLaunchKernelGGL((skinny_gemm_kernel<NT, false>), grid, dim3(256), 0,
                       stream, out, ws, x, w, M, K, N, xs, k_per_split);
  }
}

}  // namespace

extern "C" hipError_t ks_skinny_gemm(void* out, void* workspace,
                                     const void* x, const void* w, int M,
                                     int K, int N, long x_row_stride,
                                     hipStream_t stream) {
  if (M % 64 != 0 || K % BK != 0 || N > 256) return hipErrorInvalidValue;
  const int feat_wgs = M / 64;
  // fill the chip (G11): aim for >= 768 workgroups via K-splits
  int nsplit = 1;
  while (feat_wgs * nsplit < 768 && nsplit < 8 && (K / (nsplit * 2)) >= BK)
    nsplit *= 2;
  if (nsplit > 1 && workspace == nullptr) return hipErrorInvalidValue;
  int k_per_split = ((K / nsplit + BK - 1) / BK) * BK;
  const int NT = (N + 15) / 16;
  short* ws = (short*)workspace;
  const short* xs = (const short*)x;
  const short* wp = (const short*)w;
#define CASE(nt)                                                            \
  launch_nt<nt>((short*)out, ws, xs, wp, M, K, N, x_row_stride, nsplit,     \
                k_per_split, stream)
  switch (NT) {
    case 1: CASE(1); break;
    case 2: CASE(2); break;
    case 3: CASE(3); break;
    case 4: CASE(4); break;
    case 5: CASE(5); break;
    case 6: CASE(6); break;
    case 7: CASE(8); break;
    case 8: CASE(8); break;
    default:
      if (NT <= 12) CASE(12);
      else CASE(16);
      break;
  }
#undef CASE
  HIP_CHECK_KERNEL();
  if (nsplit > 1) {
    const long nm = (long)N * M;
    hipLaunchKernelGGL(reduce_ws_kernel, dim3(2048), dim3(256), 0, stream,
                       (short*)out, ws, nm, nsplit);
    HIP_CHECK_KERNEL();
  }
  return hipSuccess;
}
