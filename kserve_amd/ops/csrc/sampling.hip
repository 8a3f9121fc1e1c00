// Sampling kernels: greedy argmax and Gumbel-max temperature sampling.
//
// Gumbel-max: argmax(logit/T + G_i) with G_i = -log(-log(U_i)) samples the
// softmax(logit/T) distribution exactly, without materializing probabilities
// or sorting. Per-element U_i comes from a counter-based hash (seed, idx) so
// the kernel is stateless and hipGraph-replayable.
// (SURVEY.md §2.7 sampling row; north star: hand-written sampling kernel.)
#include "common.h"

namespace {

// xxhash-like 2x32 mix: deterministic per (seed, idx)
__device__ __forceinline__ unsigned int hash2(unsigned int a, unsigned int b) {
  unsigned int h = a * 0x9E3779B1u + b * 0x85EBCA77u + 0x165667B1u;
  h ^= h >> 15;
  h *= 0x2C1B3C6Du;
  h ^= h >> 12;
  h *= 0x297A2D39u;
  h ^= h >> 15;
  return h;
}

// logits bf16 [S, V]; one block (256 threads) per row.
template <bool GUMBEL>
__global__ __launch_bounds__(256) void argmax_kernel(
    long* __restrict__ out,            // [S]
    const short* __restrict__ logits,  // [S, V] bf16
    const float* __restrict__ temperatures,  // [S] (GUMBEL)
    const int* __restrict__ top_k,           // [S] (GUMBEL; -1 = off)
    const long* __restrict__ seeds,          // [S] (GUMBEL)
    const int V) {
  const int s = blockIdx.x;
  const short* row = logits + (long)s * V;
  const float inv_t = GUMBEL ? 1.f / fmaxf(temperatures[s], 1e-5f) : 1.f;
  const unsigned int seed = GUMBEL ? (unsigned int)(seeds[s] & 0xFFFFFFFF) : 0;

  float best = -1e38f;
  int best_idx = 0;
  // vectorized short8 loads over the vocab row
  const int nvec = V >> 3;
  const short8_t* rv = reinterpret_cast<const short8_t*>(row);
  for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
    short8_t v = rv[i];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int idx = i * 8 + j;
      float val = bf16_bits_to_float(v[j]);
      if constexpr (GUMBEL) {
        val *= inv_t;
        const unsigned int h = hash2(seed, (unsigned int)idx);
        // u in (0,1): (h + 1) / 2^32
        const float u = ((float)h + 1.0f) * 2.3283064e-10f;
        val += -__logf(-__logf(u));
      }
      if (val > best) {
        best = val;
        best_idx = idx;
      }
    }
  }
  // tail (V not divisible by 8)
  for (int idx = nvec * 8 + threadIdx.x; idx < V; idx += blockDim.x) {
    float val = bf16_bits_to_float(row[idx]);
    if constexpr (GUMBEL) {
      val *= inv_t;
      const unsigned int h = hash2(seed, (unsigned int)idx);
      const float u = ((float)h + 1.0f) * 2.3283064e-10f;
      val += -__logf(-__logf(u));
    }
    if (val > best) {
      best = val;
      best_idx = idx;
    }
  }
  // block argmax reduce: pack (value, idx); ties -> lower idx wins
  __shared__ float s_val[4];
  __shared__ int s_idx[4];
  const int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    float ov = __shfl_xor(best, off, 64);
    int oi = __shfl_xor(best_idx, off, 64);
    if (ov > best || (ov == best && oi < best_idx)) {
      best = ov;
      best_idx = oi;
    }
  }
  if (lane == 0) {
    s_val[wave] = best;
    s_idx[wave] = best_idx;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    for (int w = 1; w < 4; ++w) {
      if (s_val[w] > best || (s_val[w] == best && s_idx[w] < best_idx)) {
        best = s_val[w];
        best_idx = s_idx[w];
      }
    }
    out[s] = best_idx;
  }
}

}  // namespace

extern "C" {

hipError_t ks_greedy_sample(void* out, const void* logits, int num_seqs,
                            int vocab, hipStream_t stream) {
  if (num_seqs == 0) return hipSuccess;
  hipLaunchKernelGGL((argmax_kernel<false>), dim3(num_seqs), dim3(256), 0,
                     stream, (long*)out, (const short*)logits, nullptr,
                     nullptr, nullptr, vocab);
  HIP_CHECK_KERNEL();
  return hipSuccess;
}

hipError_t ks_gumbel_sample(void* out, const void* logits,
                            const void* temperatures, const void* top_k,
                            const void* seeds, int num_seqs, int vocab,
                            hipStream_t stream) {
  if (num_seqs == 0) return hipSuccess;
  hipLaunchKernelGGL((argmax_kernel<true>), dim3(num_seqs), dim3(256), 0,
                     stream, (long*)out, (const short*)logits,
                     (const float*)temperatures, (const int*)top_k,
                     (const long*)seeds, vocab);
  HIP_CHECK_KERNEL();
  return hipSuccess;
}
}
