// Skinny GEMM for decode-path projections: C[N][M] = X[N][K] @ W[M][K]^T,
// N <= 256 tokens (weight-streaming-bound regime where hipBLASLt reaches
// only ~1-4 TB/s on gfx950; measured tools/gemm_bench.py).
//
// CDNA4 design (guide §5 + §5.5 T2/T4, rule #21):
//  - roles: MFMA A = X fragments (L2-resident, read direct from global),
//    MFMA B = W fragments k-contiguous -> W streamed through LDS via
//    global_load_lds (16 B/lane) with PRE-SWIZZLED global source so the
//    linear LDS destination equals the XOR-swizzled layout the conflict-free
//    ds_read_b128 wants (linear dest + inverse-swizzled source + swizzled
//    read must share one involution).
//  - per-wave-private LDS slice (16 features x BK) -> no __syncthreads at
//    all; cross-wave overlap hides HBM latency (m97 finding).
//  - K-split via f32 global atomics when the feature grid alone cannot fill
//    the 256 CUs (guide G11/G12).
#include "common.h"
#include "mfma_layouts.h"

namespace {

constexpr int BK = 64;          // K per stage step
constexpr int FEAT_PER_WAVE = 16;
constexpr int NWAVES = 4;       // 64 features per workgroup

// LDS 16-bit-unit index for W element (m_local in [0,16), k in [0,BK)):
// chunk kc = k/8 is XOR-swizzled with m&7.
__device__ __forceinline__ int w_lds_idx(int m_local, int kc) {
  return (m_local * (BK / 8) + (kc ^ (m_local & 7))) * 8;
}

template <int NT, bool SPLIT>  // NT 16-token tiles; SPLIT -> atomic f32 out
__global__ __launch_bounds__(256) void skinny_gemm_kernel(
    short* __restrict__ out,      // [N][M] bf16 (!SPLIT)
    float* __restrict__ out_ws,   // [N][M] f32 (SPLIT)
    const short* __restrict__ x,  // [N][K] (row stride xs)
    const short* __restrict__ w,  // [M][K]
    const int M, const int K, const int N, const long xs,
    const int k_per_split) {
  const int f0 = blockIdx.x * (NWAVES * FEAT_PER_WAVE) +
                 (threadIdx.x >> 6) * FEAT_PER_WAVE;
  const int lane = threadIdx.x & 63;
  const int k_lo = blockIdx.y * k_per_split;
  const int k_hi = min(K, k_lo + k_per_split);

  // per-wave-private LDS slice: 16 x BK bf16 = 2 KB
  __shared__ short w_lds[NWAVES][FEAT_PER_WAVE * BK];
  short* my_lds = w_lds[threadIdx.x >> 6];

  const int m_local = lane & 15;    // A row (token) AND B col (feature) idx
  const int kgrp = lane >> 4;       // k-group of this lane (0..3)

  f32x4_t acc[NT];
#pragma unroll
  for (int t = 0; t < NT; ++t) acc[t] = f32x4_t{0.f, 0.f, 0.f, 0.f};

  // pre-swizzled global source for the two 1 KB stage instructions:
  // 16B chunk p = lane (+64): m = p/8, kc_lds = p%8, src chunk = kc_lds^(m&7)
  const int p0 = lane;
  const int sm0 = p0 >> 3, sk0 = p0 & 7;
  const int p1 = lane + 64;
  const int sm1 = p1 >> 3, sk1 = p1 & 7;
  const long wrow0 = (long)(f0 + sm0) * K + ((sk0 ^ (sm0 & 7)) << 3);
  const long wrow1 = (long)(f0 + sm1) * K + ((sk1 ^ (sm1 & 7)) << 3);

  // A-frag row pointer: token = t*16 + m_local (clamped), k-chunk = kgrp
  long a_rows[NT];
#pragma unroll
  for (int t = 0; t < NT; ++t) {
    int tokn = t * 16 + m_local;
    if (tokn >= N) tokn = N - 1;  // padded rows discarded at store
    a_rows[t] = (long)tokn * xs + (kgrp << 3);
  }

  for (int k0 = k_lo; k0 < k_hi; k0 += BK) {
    // ---- stage W slice [16 feat][64 k] via 2x global_load_lds ----
    __builtin_amdgcn_global_load_lds(
        reinterpret_cast<const unsigned int*>(w + wrow0 + k0),
        reinterpret_cast<unsigned int*>(my_lds), 16, 0, 0);
    __builtin_amdgcn_global_load_lds(
        reinterpret_cast<const unsigned int*>(w + wrow1 + k0),
        reinterpret_cast<unsigned int*>(my_lds + 64 * 8), 16, 0, 0);
    asm volatile("s_waitcnt vmcnt(0)");
    __builtin_amdgcn_sched_barrier(0);
    // ---- B fragments (features x k32) for kk = 0,1 ----
    bf16x8_t b0 = *reinterpret_cast<bf16x8_t*>(
        &my_lds[w_lds_idx(m_local, 0 * 4 + kgrp)]);
    bf16x8_t b1 = *reinterpret_cast<bf16x8_t*>(
        &my_lds[w_lds_idx(m_local, 1 * 4 + kgrp)]);
    // ---- per token-tile: A direct from global (L2-hot), 2 MFMA ----
#pragma unroll
    for (int t = 0; t < NT; ++t) {
      const short* xr = x + a_rows[t] + k0;
      bf16x8_t a0 = *reinterpret_cast<const bf16x8_t*>(xr);
      bf16x8_t a1 = *reinterpret_cast<const bf16x8_t*>(xr + 32);
      acc[t] = mfma16x16x32(a0, b0, acc[t]);
      acc[t] = mfma16x16x32(a1, b1, acc[t]);
    }
  }

  // ---- epilogue: C row = token, col = feature (C layout §3) ----
  const int feat = f0 + MFMA_C_COL(lane);
  if (feat >= M) return;
#pragma unroll
  for (int t = 0; t < NT; ++t) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int tokn = t * 16 + MFMA_C_ROW(lane, reg);
      if (tokn < N) {
        if constexpr (SPLIT) {
          atomicAdd(&out_ws[(long)tokn * M + feat], acc[t][reg]);
        } else {
          out[(long)tokn * M + feat] = float_to_bf16_bits(acc[t][reg]);
        }
      }
    }
  }
}

__global__ void cast_ws_kernel(short* __restrict__ out,
                               const float* __restrict__ ws, const long n) {
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x)
    out[i] = float_to_bf16_bits(ws[i]);
}

__global__ void zero_ws_kernel(float* __restrict__ ws, const long n) {
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x)
    ws[i] = 0.f;
}

template <int NT>
void launch_nt(short* out, float* ws, const short* x, const short* w, int M,
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
  // fill the chip: want >= ~1024 workgroups (G11)
  int nsplit = 1;
  while (feat_wgs * nsplit < 1024 && nsplit < 16 &&
         (K / (nsplit * 2)) >= BK)
    nsplit *= 2;
  int k_per_split = ((K / nsplit + BK - 1) / BK) * BK;
  const int NT = (N + 15) / 16;
  float* ws = (float*)workspace;
  if (nsplit > 1) {
    const long n = (long)N * M;
    hipLaunchKernelGGL(zero_ws_kernel, dim3(1024), dim3(256), 0, stream, ws,
                       n);
  }
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
    case 8: CASE(8); break;
    case 12: CASE(12); break;
    case 16: CASE(16); break;
    default: {
      // round NT up to a supported bucket
      int nt = NT <= 8 ? 8 : (NT <= 12 ? 12 : 16);
      if (nt == 8) CASE(8);
      else if (nt == 12) CASE(12);
      else CASE(16);
      break;
    }
  }
#undef CASE
  HIP_CHECK_KERNEL();
  if (nsplit > 1) {
    const long n = (long)N * M;
    hipLaunchKernelGGL(cast_ws_kernel, dim3(1024), dim3(256), 0, stream,
                       (short*)out, ws, n);
    HIP_CHECK_KERNEL();
  }
  return hipSuccess;
}
