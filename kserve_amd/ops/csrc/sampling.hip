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

// ---------------------------------------------------------------------------
// Fused top-k/top-p + Gumbel-max sampler.
//
// bf16 logits have 16-bit patterns that order monotonically once sign-flipped
// (key = bits<0 ? ~bits : bits|0x8000), so a 256-bin histogram over the high
// byte plus a second one over the boundary bin's low byte finds the EXACT
// k-th-largest value and the exact top-p mass threshold — no sort. The final
// pass Gumbel-argmaxes over {key >= threshold}, which samples the truncated
// softmax exactly. One 256-thread block per row; 4 passes over the vocab.
// (Replaces the torch sort path the reference's vLLM sampler falls back to.)
// ---------------------------------------------------------------------------
__device__ __forceinline__ unsigned int bf16_sort_key(short bits) {
  const unsigned short u = (unsigned short)bits;
  return (u & 0x8000u) ? (unsigned int)(unsigned short)~u
                       : (unsigned int)(u | 0x8000u);
}

__global__ __launch_bounds__(256) void topk_topp_sample_kernel(
    long* __restrict__ out,            // [S]
    const short* __restrict__ logits,  // [S, V] bf16
    const float* __restrict__ temperatures,
    const float* __restrict__ top_p,  // [S] (>=1 disables)
    const int* __restrict__ top_k,    // [S] (<=0 disables)
    const long* __restrict__ seeds,   // [S]
    const int V) {
  const int s = blockIdx.x;
  const short* row = logits + (long)s * V;
  const float inv_t = 1.f / fmaxf(temperatures[s], 1e-5f);
  const float p = top_p[s];
  const int k = top_k[s];
  const unsigned int seed = (unsigned int)(seeds[s] & 0xFFFFFFFF);
  const bool use_k = (k > 0) && (k < V);
  const bool use_p = p < 1.0f;
  const int tid = threadIdx.x;
  const int lane = tid & 63, wave = tid >> 6;

  __shared__ int h_cnt[256];
  __shared__ float h_mass[256];
  __shared__ float s_red[4];
  __shared__ int s_redi[4];
  __shared__ float sh_m;
  __shared__ int sh_bk, sh_bp, sh_need_k;
  __shared__ float sh_need_p;
  __shared__ unsigned int sh_thresh;

  // ---- pass A: row max (exp stability) ----
  float lmax = -1e38f;
  for (int i = tid; i < V; i += 256)
    lmax = fmaxf(lmax, bf16_bits_to_float(row[i]));
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    lmax = fmaxf(lmax, __shfl_xor(lmax, off, 64));
  if (lane == 0) s_red[wave] = lmax;
  __syncthreads();
  if (tid == 0) {
    float m = s_red[0];
    for (int w = 1; w < 4; ++w) m = fmaxf(m, s_red[w]);
    sh_m = m;
  }
  h_cnt[tid] = 0;
  h_mass[tid] = 0.f;
  __syncthreads();
  const float m = sh_m;

  // ---- pass B: high-byte histogram of (count, exp-mass) ----
  for (int i = tid; i < V; i += 256) {
    const short b = row[i];
    const unsigned int key = bf16_sort_key(b);
    const float e = __expf((bf16_bits_to_float(b) - m) * inv_t);
    atomicAdd(&h_cnt[key >> 8], 1);
    atomicAdd(&h_mass[key >> 8], e);
  }
  __syncthreads();
  if (tid == 0) {
    float Z = 0.f;
    for (int b = 0; b < 256; ++b) Z += h_mass[b];
    int bk = 0, need_k = 0x7FFFFFFF;
    if (use_k) {
      int c = 0;
      for (int b = 255; b >= 0; --b) {
        if (c + h_cnt[b] >= k) { bk = b; need_k = k - c; break; }
        c += h_cnt[b];
      }
    }
    int bp = 0;
    float need_p = 1e38f;
    if (use_p) {
      const float target = p * Z;
      float c = 0.f;
      for (int b = 255; b >= 0; --b) {
        if (c + h_mass[b] >= target) { bp = b; need_p = target - c; break; }
        c += h_mass[b];
      }
    }
    sh_bk = bk; sh_bp = bp; sh_need_k = need_k; sh_need_p = need_p;
  }
  __syncthreads();
  const int bk = sh_bk, bp = sh_bp;
  h_cnt[tid] = 0;
  h_mass[tid] = 0.f;
  __syncthreads();

  // ---- pass C: low-byte histograms inside the two boundary bins ----
  if (use_k || use_p) {
    for (int i = tid; i < V; i += 256) {
      const short b = row[i];
      const unsigned int key = bf16_sort_key(b);
      const int hi = key >> 8;
      if (use_k && hi == bk) atomicAdd(&h_cnt[key & 255], 1);
      if (use_p && hi == bp)
        atomicAdd(&h_mass[key & 255],
                  __expf((bf16_bits_to_float(b) - m) * inv_t));
    }
  }
  __syncthreads();
  if (tid == 0) {
    unsigned int tk = 0, tp = 0;
    if (use_k) {
      int c = 0, lo = 0;
      for (int l = 255; l >= 0; --l) {
        if (c + h_cnt[l] >= sh_need_k) { lo = l; break; }
        c += h_cnt[l];
      }
      tk = ((unsigned int)bk << 8) | lo;
    }
    if (use_p) {
      float c = 0.f;
      int lo = 0;
      for (int l = 255; l >= 0; --l) {
        if (c + h_mass[l] >= sh_need_p) { lo = l; break; }
        c += h_mass[l];
      }
      tp = ((unsigned int)bp << 8) | lo;
    }
    sh_thresh = tk > tp ? tk : tp;
  }
  __syncthreads();
  const unsigned int thr = sh_thresh;

  // ---- pass D: Gumbel-argmax over the surviving set ----
  float best = -1e38f;
  int best_idx = 0;
  for (int i = tid; i < V; i += 256) {
    const short b = row[i];
    if (bf16_sort_key(b) < thr) continue;
    const unsigned int h = hash2(seed, (unsigned int)i);
    const float u = ((float)h + 1.0f) * 2.3283064e-10f;
    const float val = bf16_bits_to_float(b) * inv_t - __logf(-__logf(u));
    if (val > best) { best = val; best_idx = i; }
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    float ov = __shfl_xor(best, off, 64);
    int oi = __shfl_xor(best_idx, off, 64);
    if (ov > best || (ov == best && oi < best_idx)) { best = ov; best_idx = oi; }
  }
  if (lane == 0) { s_red[wave] = best; s_redi[wave] = best_idx; }
  __syncthreads();
  if (tid == 0) {
    for (int w = 1; w < 4; ++w) {
      if (s_red[w] > best || (s_red[w] == best && s_redi[w] < best_idx)) {
        best = s_red[w];
        best_idx = s_redi[w];
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

hipError_t ks_topk_topp_sample(void* out, const void* logits,
                               const void* temperatures, const void* top_p,
                               const void* top_k, const void* seeds,
                               int num_seqs, int vocab, hipStream_t stream) {
  if (num_seqs == 0) return hipSuccess;
  hipLaunchKernelGGL(topk_topp_sample_kernel, dim3(num_seqs), dim3(256), 0,
                     stream, (long*)out, (const short*)logits,
                     (const float*)temperatures, (const float*)top_p,
                     (const int*)top_k, (const long*)seeds, vocab);
  HIP_CHECK_KERNEL();
  return hipSuccess;
}
}
