// Flash-style causal prefill attention (varlen, GQA) on MFMA.
//
// CDNA4 design (guide §5/§B): per workgroup 4 waves x 16 q-rows = 64 q rows
// of one (seq, head); K/V staged in LDS in 32-token tiles; QK^T and PV on
// v_mfma_f32_16x16x32_bf16; online softmax on the C-fragment rows; LDS
// XOR-swizzled (G4) to kill the D=128 row-major bank conflict; V stored
// transposed at stage time so the PV B-operand reads are contiguous.
//
// Replaces the vLLM flash prefill the reference delegates to
// (SURVEY.md §2.7 row 1). Numerics oracle: ops/torch_ref.flash_prefill_varlen.
#include "common.h"
#include "mfma_layouts.h"

#define NEG_INF (-1e30f)

namespace {

constexpr int KT = 32;        // kv tokens per tile
constexpr int QW = 16;        // q rows per wave
constexpr int NWAVES = 4;     // waves per workgroup (64 q rows)

// swizzled LDS index helpers (short units)
template <int D>
__device__ __forceinline__ int k_idx(int row, int col) {
  // [32][D] shorts; XOR 8-short chunks with row&7 (D=64: modulo row width)
  constexpr int MASK = (D / 8) - 1;
  const int chunk = (col >> 3) ^ ((row & 7) & MASK);
  return row * D + (chunk << 3) + (col & 7);
}
__device__ __forceinline__ int v_idx(int d, int tok) {
  // transposed [128 dims][32 tokens] shorts; XOR 8-short chunks with d&3
  return d * KT + (tok ^ ((d & 3) << 3));
}

template <int D, bool CAUSAL, bool WIN = false>  // D == 128
__global__ __launch_bounds__(256) void flash_prefill_kernel(
    short* __restrict__ out,      // [T, Hq, D] bf16
    const short* __restrict__ q,  // [T, Hq, D]
    const short* __restrict__ k,  // [T, Hkv, D]
    const short* __restrict__ v,  // [T, Hkv, D]
    const int* __restrict__ cu_seqlens,  // [S+1]
    const int Hq, const int Hkv, const float scale, const long sq,
    const long sk, const long sv, const int window) {
  const int head = blockIdx.x;
  const int tile = blockIdx.y;
  const int seq = blockIdx.z;
  const int q_start = cu_seqlens[seq];
  const int len = cu_seqlens[seq + 1] - q_start;
  const int tile_base = tile * (NWAVES * QW);
  if (tile_base >= len) return;
  const int group = Hq / Hkv;
  const int kv_head = head / group;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;

  __shared__ short k_tile[KT * D];        // 8 KB swizzled
  __shared__ short v_t[D * KT];           // 8 KB transposed+swizzled
  __shared__ short p_lds[NWAVES][QW * KT];  // 4 KB

  const int wq0 = tile_base + wave * QW;  // first q row of this wave
  const bool active = wq0 < len;

  // ---- load Q fragments (A operand), rows clamped to len-1 ----
  bf16x8_t a_q[D / 32];
  {
    const int qrow = min(wq0 + MFMA_RC_OF(lane), len - 1);
    const short* qp = q + (long)(q_start + qrow) * sq + (long)head * D;
#pragma unroll
    for (int c = 0; c < D / 32; ++c) {
      const int off = c * 32 + ((lane >> 4) << 3);
      a_q[c] = *reinterpret_cast<const bf16x8_t*>(qp + off);
    }
  }

  float m[4], l[4];
  f32x4_t o_acc[D / 16];
#pragma unroll
  for (int r = 0; r < 4; ++r) { m[r] = NEG_INF; l[r] = 0.f; }
#pragma unroll
  for (int c = 0; c < D / 16; ++c) o_acc[c] = f32x4_t{0.f, 0.f, 0.f, 0.f};

  // causal bound: last token needed by this workgroup (bidirectional: all)
  const int kv_limit = CAUSAL ? min(len, tile_base + NWAVES * QW) : len;
  const int ntiles = (kv_limit + KT - 1) / KT;

  for (int kt = 0; kt < ntiles; ++kt) {
    const int t0 = kt * KT;
    __syncthreads();
    // ---- stage K tile (swizzled) and V tile (transposed) ----
    for (int i = threadIdx.x; i < KT * (D / 8); i += 256) {
      const int r = i / (D / 8);          // token within tile
      const int c8 = (i % (D / 8)) * 8;   // first dim of this short8
      const int tok = t0 + r;
      short8_t val;
      if (tok < len) {
        val = *reinterpret_cast<const short8_t*>(
            k + (long)(q_start + tok) * sk + (long)kv_head * D + c8);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) val[j] = 0;
      }
      *reinterpret_cast<short8_t*>(&k_tile[k_idx<D>(r, c8)]) = val;
      short8_t vv;
      if (tok < len) {
        vv = *reinterpret_cast<const short8_t*>(
            v + (long)(q_start + tok) * sv + (long)kv_head * D + c8);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) vv[j] = 0;
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) v_t[v_idx(c8 + j, r)] = vv[j];
    }
    __syncthreads();
    if (!active) continue;
    // causal skip: this wave's rows are all below the tile's first token
    if (CAUSAL && t0 > wq0 + QW - 1) continue;

    // ---- QK^T: S[16 q][32 t] as 2 sub-tiles of 16 tokens ----
    f32x4_t s_frag[2] = {f32x4_t{0, 0, 0, 0}, f32x4_t{0, 0, 0, 0}};
#pragma unroll
    for (int n = 0; n < 2; ++n) {
      const int tok_row = n * 16 + MFMA_RC_OF(lane);
#pragma unroll
      for (int c = 0; c < D / 32; ++c) {
        const int col = c * 32 + ((lane >> 4) << 3);
        bf16x8_t bk =
            *reinterpret_cast<bf16x8_t*>(&k_tile[k_idx<D>(tok_row, col)]);
        s_frag[n] = mfma16x16x32(a_q[c], bk, s_frag[n]);
      }
    }
    // ---- mask + online softmax ----
    float p[2][4];
    float rescale[4];
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int qrow = wq0 + MFMA_C_ROW(lane, reg);
      float sv[2];
#pragma unroll
      for (int n = 0; n < 2; ++n) {
        const int tok = t0 + n * 16 + MFMA_C_COL(lane);
        const bool valid =
            (!CAUSAL || tok <= qrow) && (qrow < len) && (tok < len) &&
            (!WIN || tok > qrow - window);
        sv[n] = valid ? s_frag[n][reg] * scale : NEG_INF;
      }
      float rowmax = group_reduce_max<16>(fmaxf(sv[0], sv[1]));
      const float m_new = fmaxf(m[reg], rowmax);
      rescale[reg] = (m[reg] > NEG_INF && m_new > NEG_INF)
                         ? __expf(m[reg] - m_new)
                         : 0.f;
#pragma unroll
      for (int n = 0; n < 2; ++n)
        p[n][reg] = (sv[n] > NEG_INF) ? __expf(sv[n] - m_new) : 0.f;
      const float rowsum = group_reduce_sum<16>(p[0][reg] + p[1][reg]);
      l[reg] = l[reg] * rescale[reg] + rowsum;
      m[reg] = m_new;
    }
#pragma unroll
    for (int c = 0; c < D / 16; ++c) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) o_acc[c][reg] *= rescale[reg];
    }
    // ---- P -> LDS (C layout -> A layout relayout) ----
#pragma unroll
    for (int n = 0; n < 2; ++n) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int prow = MFMA_C_ROW(lane, reg);
        const int pcol = n * 16 + MFMA_C_COL(lane);
        p_lds[wave][prow * KT + pcol] = float_to_bf16_bits(p[n][reg]);
      }
    }
    asm volatile("s_waitcnt lgkmcnt(0)");
    __builtin_amdgcn_sched_barrier(0);  // keep reads after the waitcnt (G#18)
    // ---- PV: O[16 q][D] += P[16 q][32 t] @ V[32 t][D] ----
    bf16x8_t pa = *reinterpret_cast<bf16x8_t*>(
        &p_lds[wave][MFMA_RC_OF(lane) * KT + ((lane >> 4) << 3)]);
#pragma unroll
    for (int c = 0; c < D / 16; ++c) {
      const int d = c * 16 + MFMA_RC_OF(lane);
      bf16x8_t bv =
          *reinterpret_cast<bf16x8_t*>(&v_t[v_idx(d, (lane >> 4) << 3)]);
      o_acc[c] = mfma16x16x32(pa, bv, o_acc[c]);
    }
    asm volatile("s_waitcnt lgkmcnt(0)");
    __builtin_amdgcn_sched_barrier(0);  // p_lds overwritten next iter
  }

  if (!active) return;
  // ---- epilogue: normalize and store ----
  float inv_l[4];
#pragma unroll
  for (int reg = 0; reg < 4; ++reg)
    inv_l[reg] = (l[reg] > 0.f) ? 1.f / l[reg] : 0.f;
#pragma unroll
  for (int c = 0; c < D / 16; ++c) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int qrow = wq0 + MFMA_C_ROW(lane, reg);
      if (qrow < len) {
        out[((long)(q_start + qrow) * Hq + head) * D + c * 16 +
            MFMA_C_COL(lane)] = float_to_bf16_bits(o_acc[c][reg] * inv_l[reg]);
      }
    }
  }
}

}  // namespace

// ---------------------------------------------------------------------------
// v2: 32 q-rows per wave (128/wg) x 64-token KV tiles — 4x the MFMA work per
// barrier of v1 (guide ladder: bigger tiles amortize staging + softmax).
// D=128 only; v1 stays for D=64 and as the KS_PREFILL_V1 ablation.
// ---------------------------------------------------------------------------
namespace {

constexpr int KT2 = 64;   // kv tokens per tile
constexpr int QW2 = 32;   // q rows per wave
constexpr int PSTR = 72;  // p_lds row stride (64 + 8 shorts: 2-way banks)

template <int D, bool CAUSAL, bool AS = false, bool WIN = false>
__global__ __launch_bounds__(256) void flash_prefill_v2_kernel(
    short* __restrict__ out, const short* __restrict__ q,
    const short* __restrict__ k, const short* __restrict__ v,
    const int* __restrict__ cu_seqlens, const int Hq, const int Hkv,
    const float scale, const long sq, const long sk, const long sv,
    const int window) {
  const int head = blockIdx.x;
  const int tile = blockIdx.y;
  const int seq = blockIdx.z;
  const int q_start = cu_seqlens[seq];
  const int len = cu_seqlens[seq + 1] - q_start;
  const int tile_base = tile * (NWAVES * QW2);
  if (tile_base >= len) return;
  const int group = Hq / Hkv;
  const int kv_head = head / group;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;

  __shared__ short k_tile[KT2 * D];            // 16 KB swizzled
  __shared__ short v_t[D * KT2];               // 16 KB transposed+swizzled
  __shared__ short p_lds[NWAVES][QW2 * PSTR];  // 18 KB

  const int wq0 = tile_base + wave * QW2;
  const bool active = wq0 < len;

  // Q fragments: 2 row-tiles x 4 k-chunks (A operand)
  bf16x8_t a_q[2][D / 32];
#pragma unroll
  for (int rt = 0; rt < 2; ++rt) {
    const int qrow = min(wq0 + rt * 16 + MFMA_RC_OF(lane), len - 1);
    const short* qp = q + (long)(q_start + qrow) * sq + (long)head * D;
#pragma unroll
    for (int c = 0; c < D / 32; ++c) {
      a_q[rt][c] = *reinterpret_cast<const bf16x8_t*>(
          qp + c * 32 + ((lane >> 4) << 3));
    }
  }

  float m[2][4], l[2][4];
  f32x4_t o_acc[2][D / 16];
#pragma unroll
  for (int rt = 0; rt < 2; ++rt) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      m[rt][r] = NEG_INF;
      l[rt][r] = 0.f;
    }
#pragma unroll
    for (int c = 0; c < D / 16; ++c)
      o_acc[rt][c] = f32x4_t{0.f, 0.f, 0.f, 0.f};
  }

  const int kv_limit = CAUSAL ? min(len, tile_base + NWAVES * QW2) : len;
  const int ntiles = (kv_limit + KT2 - 1) / KT2;

  // T14 async-stage split (guide: +17% attn): the NEXT tile's K/V global
  // loads are ISSUED into registers during this tile's compute; the
  // ds_writes land after the barrier, so HBM latency hides under the
  // MFMA phase instead of serializing the staging loop.
  constexpr int CHUNKS = AS ? (KT2 * (D / 8) + 255) / 256 : 1;
  short8_t kreg[CHUNKS], vreg[CHUNKS];
  auto issue_tile = [&](int tt0) {
#pragma unroll
    for (int n = 0; n < CHUNKS; ++n) {
      const int i = threadIdx.x + n * 256;
      const int r = i / (D / 8);
      const int c8 = (i % (D / 8)) * 8;
      const int tok = tt0 + r;
      if (tok < len) {
        kreg[n] = *reinterpret_cast<const short8_t*>(
            k + (long)(q_start + tok) * sk + (long)kv_head * D + c8);
        vreg[n] = *reinterpret_cast<const short8_t*>(
            v + (long)(q_start + tok) * sv + (long)kv_head * D + c8);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          kreg[n][j] = 0;
          vreg[n][j] = 0;
        }
      }
    }
  };
  auto commit_tile = [&]() {
#pragma unroll
    for (int n = 0; n < CHUNKS; ++n) {
      const int i = threadIdx.x + n * 256;
      const int r = i / (D / 8);
      const int c8 = (i % (D / 8)) * 8;
      *reinterpret_cast<short8_t*>(&k_tile[k_idx<D>(r, c8)]) = kreg[n];
#pragma unroll
      for (int j = 0; j < 8; ++j)
        v_t[(c8 + j) * KT2 + (r ^ (((c8 + j) & 3) << 3))] = vreg[n][j];
    }
  };
  if constexpr (AS) issue_tile(0);

  for (int kt = 0; kt < ntiles; ++kt) {
    const int t0 = kt * KT2;
    __syncthreads();
    if constexpr (AS) {
      commit_tile();
      __syncthreads();
      if (kt + 1 < ntiles) issue_tile(t0 + KT2);
    } else {
      for (int i = threadIdx.x; i < KT2 * (D / 8); i += 256) {
        const int r = i / (D / 8);
        const int c8 = (i % (D / 8)) * 8;
        const int tok = t0 + r;
        short8_t val, vv;
        if (tok < len) {
          val = *reinterpret_cast<const short8_t*>(
              k + (long)(q_start + tok) * sk + (long)kv_head * D + c8);
          vv = *reinterpret_cast<const short8_t*>(
              v + (long)(q_start + tok) * sv + (long)kv_head * D + c8);
        } else {
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            val[j] = 0;
            vv[j] = 0;
          }
        }
        *reinterpret_cast<short8_t*>(&k_tile[k_idx<D>(r, c8)]) = val;
#pragma unroll
        for (int j = 0; j < 8; ++j) v_t[(c8 + j) * KT2 + (r ^ (((c8 + j) & 3) << 3))] = vv[j];
      }
      __syncthreads();
    }
    if (!active) continue;
    if (CAUSAL && t0 > wq0 + QW2 - 1) continue;

    // ---- per row-tile: QK^T over 4 token sub-tiles, online softmax ----
#pragma unroll
    for (int rt = 0; rt < 2; ++rt) {
      const int rbase = wq0 + rt * 16;
      if (CAUSAL && t0 > rbase + 15) continue;  // fully-masked row-tile
      f32x4_t s_frag[KT2 / 16];
#pragma unroll
      for (int n = 0; n < KT2 / 16; ++n) {
        s_frag[n] = f32x4_t{0.f, 0.f, 0.f, 0.f};
        const int tok_row = n * 16 + MFMA_RC_OF(lane);
#pragma unroll
        for (int c = 0; c < D / 32; ++c) {
          const int col = c * 32 + ((lane >> 4) << 3);
          bf16x8_t bk =
              *reinterpret_cast<bf16x8_t*>(&k_tile[k_idx<D>(tok_row, col)]);
          s_frag[n] = mfma16x16x32(a_q[rt][c], bk, s_frag[n]);
        }
      }
      float p[KT2 / 16][4];
      float rescale[4];
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int qrow = rbase + MFMA_C_ROW(lane, reg);
        float rowmax = NEG_INF;
        float sv_[KT2 / 16];
#pragma unroll
        for (int n = 0; n < KT2 / 16; ++n) {
          const int tok = t0 + n * 16 + MFMA_C_COL(lane);
          const bool valid =
              (!CAUSAL || tok <= qrow) && (qrow < len) && (tok < len) &&
              (!WIN || tok > qrow - window);
          sv_[n] = valid ? s_frag[n][reg] * scale : NEG_INF;
          rowmax = fmaxf(rowmax, sv_[n]);
        }
        rowmax = group_reduce_max<16>(rowmax);
        const float m_new = fmaxf(m[rt][reg], rowmax);
        rescale[reg] = (m[rt][reg] > NEG_INF && m_new > NEG_INF)
                           ? __expf(m[rt][reg] - m_new)
                           : 0.f;
        float psum = 0.f;
#pragma unroll
        for (int n = 0; n < KT2 / 16; ++n) {
          p[n][reg] = (sv_[n] > NEG_INF) ? __expf(sv_[n] - m_new) : 0.f;
          psum += p[n][reg];
        }
        psum = group_reduce_sum<16>(psum);
        l[rt][reg] = l[rt][reg] * rescale[reg] + psum;
        m[rt][reg] = m_new;
      }
#pragma unroll
      for (int c = 0; c < D / 16; ++c) {
#pragma unroll
        for (int reg = 0; reg < 4; ++reg) o_acc[rt][c][reg] *= rescale[reg];
      }
#pragma unroll
      for (int n = 0; n < KT2 / 16; ++n) {
#pragma unroll
        for (int reg = 0; reg < 4; ++reg) {
          p_lds[wave][(rt * 16 + MFMA_C_ROW(lane, reg)) * PSTR + n * 16 +
                      MFMA_C_COL(lane)] = float_to_bf16_bits(p[n][reg]);
        }
      }
    }
    asm volatile("s_waitcnt lgkmcnt(0)");
    __builtin_amdgcn_sched_barrier(0);
    // ---- PV: O[32 q][D] += P[32 q][64 t] @ V[64 t][D] ----
#pragma unroll
    for (int rt = 0; rt < 2; ++rt) {
      if (CAUSAL && t0 > wq0 + rt * 16 + 15) continue;
#pragma unroll
      for (int kk = 0; kk < KT2 / 32; ++kk) {
        bf16x8_t pa = *reinterpret_cast<bf16x8_t*>(
            &p_lds[wave][(rt * 16 + MFMA_RC_OF(lane)) * PSTR + kk * 32 +
                         ((lane >> 4) << 3)]);
#pragma unroll
        for (int c = 0; c < D / 16; ++c) {
          const int d = c * 16 + MFMA_RC_OF(lane);
          const int tok0 = kk * 32 + ((lane >> 4) << 3);
          bf16x8_t bv = *reinterpret_cast<bf16x8_t*>(
              &v_t[d * KT2 + (tok0 ^ ((d & 3) << 3))]);
          o_acc[rt][c] = mfma16x16x32(pa, bv, o_acc[rt][c]);
        }
      }
    }
    asm volatile("s_waitcnt lgkmcnt(0)");
    __builtin_amdgcn_sched_barrier(0);
  }

  if (!active) return;
#pragma unroll
  for (int rt = 0; rt < 2; ++rt) {
    float inv_l[4];
#pragma unroll
    for (int reg = 0; reg < 4; ++reg)
      inv_l[reg] = (l[rt][reg] > 0.f) ? 1.f / l[rt][reg] : 0.f;
#pragma unroll
    for (int c = 0; c < D / 16; ++c) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int qrow = wq0 + rt * 16 + MFMA_C_ROW(lane, reg);
        if (qrow < len) {
          out[((long)(q_start + qrow) * Hq + head) * D + c * 16 +
              MFMA_C_COL(lane)] =
              float_to_bf16_bits(o_acc[rt][c][reg] * inv_l[reg]);
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Paged-context prefill (chunked prefill): the chunk's q rows attend causally
// to the WHOLE context — previous chunks' KV plus this chunk's — read from
// the paged cache (the chunk's KV was written by reshape_and_cache first).
// Same MFMA/LDS structure as v2; staging gathers 64-token tiles as 4
// contiguous 16xD pages via the block table. Reference parity: the chunked
// prefill path of the vLLM backend the reference delegates to.
// Numerics oracle: ops/torch_ref.context_attention_varlen.
// ---------------------------------------------------------------------------
// CT: cache element type — short (bf16) or unsigned char (fp8 E4M3,
// converted to bf16 during LDS staging; MFMA math stays bf16)
template <int D, typename CT = short, bool WIN = false>  // D == 128
__global__ __launch_bounds__(256) void context_prefill_kernel(
    short* __restrict__ out,        // [Tq, Hq, D] bf16
    const short* __restrict__ q,    // [Tq, Hq, D] (row stride sq)
    const CT* __restrict__ k_cache,  // [NB, Hkv, 16, D]
    const CT* __restrict__ v_cache,
    const int* __restrict__ block_tables,  // [S, max_nb]
    const int* __restrict__ ctx_lens,      // [S] total context incl. chunk
    const int* __restrict__ cu_seqlens_q,  // [S+1]
    const int Hq, const int Hkv, const int max_nb, const float scale,
    const long sq, const int window) {
  const int head = blockIdx.x;
  const int tile = blockIdx.y;
  const int seq = blockIdx.z;
  const int q_start = cu_seqlens_q[seq];
  const int q_len = cu_seqlens_q[seq + 1] - q_start;
  const int ctx = ctx_lens[seq];
  const int start_pos = ctx - q_len;  // absolute position of chunk row 0
  const int tile_base = tile * (NWAVES * QW2);
  if (tile_base >= q_len) return;
  const int group = Hq / Hkv;
  const int kv_head = head / group;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int* bt = block_tables + (long)seq * max_nb;

  __shared__ short k_tile[KT2 * D];
  __shared__ short v_t[D * KT2];
  __shared__ short p_lds[NWAVES][QW2 * PSTR];

  const int wq0 = tile_base + wave * QW2;
  const bool active = wq0 < q_len;

  bf16x8_t a_q[2][D / 32];
#pragma unroll
  for (int rt = 0; rt < 2; ++rt) {
    const int qrow = min(wq0 + rt * 16 + MFMA_RC_OF(lane), q_len - 1);
    const short* qp = q + (long)(q_start + qrow) * sq + (long)head * D;
#pragma unroll
    for (int c = 0; c < D / 32; ++c) {
      a_q[rt][c] = *reinterpret_cast<const bf16x8_t*>(
          qp + c * 32 + ((lane >> 4) << 3));
    }
  }

  float m[2][4], l[2][4];
  f32x4_t o_acc[2][D / 16];
#pragma unroll
  for (int rt = 0; rt < 2; ++rt) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      m[rt][r] = NEG_INF;
      l[rt][r] = 0.f;
    }
#pragma unroll
    for (int c = 0; c < D / 16; ++c)
      o_acc[rt][c] = f32x4_t{0.f, 0.f, 0.f, 0.f};
  }

  // causal bound in ABSOLUTE positions: last key this workgroup's rows see
  const int kv_limit =
      min(ctx, start_pos + tile_base + NWAVES * QW2);
  const int ntiles = (kv_limit + KT2 - 1) / KT2;

  for (int kt = 0; kt < ntiles; ++kt) {
    const int t0 = kt * KT2;
    __syncthreads();
    // stage 64 context tokens = 4 pages of 16, gathered via the block table
    for (int i = threadIdx.x; i < KT2 * (D / 8); i += 256) {
      const int r = i / (D / 8);
      const int c8 = (i % (D / 8)) * 8;
      const int tok = t0 + r;
      short8_t val, vv;
      if (tok < ctx) {
        const long base =
            (((long)bt[tok >> 4] * Hkv + kv_head) * 16 + (tok & 15)) * D + c8;
        if constexpr (sizeof(CT) == 2) {
          val = *reinterpret_cast<const short8_t*>(k_cache + base);
          vv = *reinterpret_cast<const short8_t*>(v_cache + base);
        } else {
          // fp8: 8 bytes -> 8 bf16 via two packed converts per dword
          const unsigned int k2[2] = {
              reinterpret_cast<const unsigned int*>(k_cache + base)[0],
              reinterpret_cast<const unsigned int*>(k_cache + base)[1]};
          const unsigned int v2[2] = {
              reinterpret_cast<const unsigned int*>(v_cache + base)[0],
              reinterpret_cast<const unsigned int*>(v_cache + base)[1]};
#pragma unroll
          for (int d = 0; d < 2; ++d) {
            const float2_t klo =
                __builtin_amdgcn_cvt_pk_f32_fp8((int)k2[d], false);
            const float2_t khi =
                __builtin_amdgcn_cvt_pk_f32_fp8((int)k2[d], true);
            const float2_t vlo =
                __builtin_amdgcn_cvt_pk_f32_fp8((int)v2[d], false);
            const float2_t vhi =
                __builtin_amdgcn_cvt_pk_f32_fp8((int)v2[d], true);
            val[d * 4 + 0] = float_to_bf16_bits(klo[0]);
            val[d * 4 + 1] = float_to_bf16_bits(klo[1]);
            val[d * 4 + 2] = float_to_bf16_bits(khi[0]);
            val[d * 4 + 3] = float_to_bf16_bits(khi[1]);
            vv[d * 4 + 0] = float_to_bf16_bits(vlo[0]);
            vv[d * 4 + 1] = float_to_bf16_bits(vlo[1]);
            vv[d * 4 + 2] = float_to_bf16_bits(vhi[0]);
            vv[d * 4 + 3] = float_to_bf16_bits(vhi[1]);
          }
        }
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          val[j] = 0;
          vv[j] = 0;
        }
      }
      *reinterpret_cast<short8_t*>(&k_tile[k_idx<D>(r, c8)]) = val;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        v_t[(c8 + j) * KT2 + (r ^ (((c8 + j) & 3) << 3))] = vv[j];
    }
    __syncthreads();
    if (!active) continue;
    // wave-level causal skip: first key of tile past this wave's last q pos
    if (t0 > start_pos + wq0 + QW2 - 1) continue;

#pragma unroll
    for (int rt = 0; rt < 2; ++rt) {
      const int rbase = wq0 + rt * 16;
      if (t0 > start_pos + rbase + 15) continue;
      f32x4_t s_frag[KT2 / 16];
#pragma unroll
      for (int n = 0; n < KT2 / 16; ++n) {
        s_frag[n] = f32x4_t{0.f, 0.f, 0.f, 0.f};
        const int tok_row = n * 16 + MFMA_RC_OF(lane);
#pragma unroll
        for (int c = 0; c < D / 32; ++c) {
          const int col = c * 32 + ((lane >> 4) << 3);
          bf16x8_t bk =
              *reinterpret_cast<bf16x8_t*>(&k_tile[k_idx<D>(tok_row, col)]);
          s_frag[n] = mfma16x16x32(a_q[rt][c], bk, s_frag[n]);
        }
      }
      float p[KT2 / 16][4];
      float rescale[4];
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int qrow = rbase + MFMA_C_ROW(lane, reg);
        const int qpos = start_pos + qrow;
        float rowmax = NEG_INF;
        float sv_[KT2 / 16];
#pragma unroll
        for (int n = 0; n < KT2 / 16; ++n) {
          const int tok = t0 + n * 16 + MFMA_C_COL(lane);
          const bool valid = (tok <= qpos) && (qrow < q_len) && (tok < ctx) &&
                             (!WIN || tok > qpos - window);
          sv_[n] = valid ? s_frag[n][reg] * scale : NEG_INF;
          rowmax = fmaxf(rowmax, sv_[n]);
        }
        rowmax = group_reduce_max<16>(rowmax);
        const float m_new = fmaxf(m[rt][reg], rowmax);
        rescale[reg] = (m[rt][reg] > NEG_INF && m_new > NEG_INF)
                           ? __expf(m[rt][reg] - m_new)
                           : 0.f;
        float psum = 0.f;
#pragma unroll
        for (int n = 0; n < KT2 / 16; ++n) {
          p[n][reg] = (sv_[n] > NEG_INF) ? __expf(sv_[n] - m_new) : 0.f;
          psum += p[n][reg];
        }
        psum = group_reduce_sum<16>(psum);
        l[rt][reg] = l[rt][reg] * rescale[reg] + psum;
        m[rt][reg] = m_new;
      }
#pragma unroll
      for (int c = 0; c < D / 16; ++c) {
#pragma unroll
        for (int reg = 0; reg < 4; ++reg) o_acc[rt][c][reg] *= rescale[reg];
      }
#pragma unroll
      for (int n = 0; n < KT2 / 16; ++n) {
#pragma unroll
        for (int reg = 0; reg < 4; ++reg) {
          p_lds[wave][(rt * 16 + MFMA_C_ROW(lane, reg)) * PSTR + n * 16 +
                      MFMA_C_COL(lane)] = float_to_bf16_bits(p[n][reg]);
        }
      }
    }
    asm volatile("s_waitcnt lgkmcnt(0)");
    __builtin_amdgcn_sched_barrier(0);
#pragma unroll
    for (int rt = 0; rt < 2; ++rt) {
      if (t0 > start_pos + wq0 + rt * 16 + 15) continue;
#pragma unroll
      for (int kk = 0; kk < KT2 / 32; ++kk) {
        bf16x8_t pa = *reinterpret_cast<bf16x8_t*>(
            &p_lds[wave][(rt * 16 + MFMA_RC_OF(lane)) * PSTR + kk * 32 +
                         ((lane >> 4) << 3)]);
#pragma unroll
        for (int c = 0; c < D / 16; ++c) {
          const int d = c * 16 + MFMA_RC_OF(lane);
          const int tok0 = kk * 32 + ((lane >> 4) << 3);
          bf16x8_t bv = *reinterpret_cast<bf16x8_t*>(
              &v_t[d * KT2 + (tok0 ^ ((d & 3) << 3))]);
          o_acc[rt][c] = mfma16x16x32(pa, bv, o_acc[rt][c]);
        }
      }
    }
    asm volatile("s_waitcnt lgkmcnt(0)");
    __builtin_amdgcn_sched_barrier(0);
  }

  if (!active) return;
#pragma unroll
  for (int rt = 0; rt < 2; ++rt) {
    float inv_l[4];
#pragma unroll
    for (int reg = 0; reg < 4; ++reg)
      inv_l[reg] = (l[rt][reg] > 0.f) ? 1.f / l[rt][reg] : 0.f;
#pragma unroll
    for (int c = 0; c < D / 16; ++c) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int qrow = wq0 + rt * 16 + MFMA_C_ROW(lane, reg);
        if (qrow < q_len) {
          out[((long)(q_start + qrow) * Hq + head) * D + c * 16 +
              MFMA_C_COL(lane)] =
              float_to_bf16_bits(o_acc[rt][c][reg] * inv_l[reg]);
        }
      }
    }
  }
}

}  // namespace

extern "C" hipError_t ks_context_prefill_varlen(
    void* out, const void* q, const void* k_cache, const void* v_cache,
    const void* block_tables, const void* ctx_lens, const void* cu_seqlens_q,
    int num_seqs, int max_q_len, int Hq, int Hkv, int head_dim, int max_nb,
    float scale, long sq, int fp8_cache, int window, hipStream_t stream) {
  if (head_dim != 128) return hipErrorInvalidValue;
  if (Hq % Hkv != 0) return hipErrorInvalidValue;
  const int max_tiles = (max_q_len + NWAVES * QW2 - 1) / (NWAVES * QW2);
  if (max_tiles == 0 || num_seqs == 0) return hipSuccess;
  dim3 grid(Hq, max_tiles, num_seqs);
#define LAUNCH_CTX(CT, WW)                                                  \
  hipLaunchKernelGGL((context_prefill_kernel<128, CT, WW>), grid,           \
                     dim3(256), 0, stream, (short*)out, (const short*)q,    \
                     (const CT*)k_cache, (const CT*)v_cache,                \
                     (const int*)block_tables, (const int*)ctx_lens,        \
                     (const int*)cu_seqlens_q, Hq, Hkv, max_nb, scale, sq,  \
                     window)
  const bool winc = window > 0;
  if (fp8_cache) {
    if (winc) LAUNCH_CTX(unsigned char, true);
    else LAUNCH_CTX(unsigned char, false);
  } else {
    if (winc) LAUNCH_CTX(short, true);
    else LAUNCH_CTX(short, false);
  }
#undef LAUNCH_CTX
  HIP_CHECK_KERNEL();
  return hipSuccess;
}

extern "C" hipError_t ks_flash_prefill_varlen(
    void* out, const void* q, const void* k, const void* v,
    const void* cu_seqlens, int num_seqs, int max_seqlen, int Hq, int Hkv,
    int head_dim, float scale, long sq, long sk, long sv, int causal,
    int window, hipStream_t stream) {
  if (head_dim != 128 && head_dim != 64) return hipErrorInvalidValue;
  if (Hq % Hkv != 0) return hipErrorInvalidValue;
  const int max_tiles = (max_seqlen + NWAVES * QW - 1) / (NWAVES * QW);
  if (max_tiles == 0 || num_seqs == 0) return hipSuccess;
  dim3 grid(Hq, max_tiles, num_seqs);
  // v2 (32 q-rows/wave, 64-token tiles) for D=128; KS_PREFILL_V1=1 reverts
  static const bool use_v1 = [] {
    const char* e = getenv("KS_PREFILL_V1");
    return e != nullptr && e[0] == '1';
  }();
  if (head_dim == 128 && !use_v1) {
    const int max_tiles2 = (max_seqlen + NWAVES * QW2 - 1) / (NWAVES * QW2);
    dim3 grid2(Hq, max_tiles2, num_seqs);
    static const bool use_as = [] {  // T14 A/B: KS_PREFILL_AS=1
      const char* e = getenv("KS_PREFILL_AS");
      return e != nullptr && e[0] == '1';
    }();
#define LAUNCH_FP2(CC, AA, WW)                                              \
  hipLaunchKernelGGL((flash_prefill_v2_kernel<128, CC, AA, WW>), grid2,     \
                     dim3(256), 0, stream, (short*)out, (const short*)q,    \
                     (const short*)k, (const short*)v,                      \
                     (const int*)cu_seqlens, Hq, Hkv, scale, sq, sk, sv,    \
                     window)
    const bool win = window > 0;
    if (causal && use_as) {
      if (win) LAUNCH_FP2(true, true, true);
      else LAUNCH_FP2(true, true, false);
    } else if (causal) {
      if (win) LAUNCH_FP2(true, false, true);
      else LAUNCH_FP2(true, false, false);
    } else if (use_as) {
      if (win) LAUNCH_FP2(false, true, true);
      else LAUNCH_FP2(false, true, false);
    } else {
      if (win) LAUNCH_FP2(false, false, true);
      else LAUNCH_FP2(false, false, false);
    }
#undef LAUNCH_FP2
    HIP_CHECK_KERNEL();
    return hipSuccess;
  }
#define LAUNCH_FP(DD, CC, WW)                                               \
  hipLaunchKernelGGL((flash_prefill_kernel<DD, CC, WW>), grid, dim3(256),   \
                     0, stream, (short*)out, (const short*)q,               \
                     (const short*)k, (const short*)v,                      \
                     (const int*)cu_seqlens, Hq, Hkv, scale, sq, sk, sv,    \
                     window)
  const bool win1 = window > 0;
  if (head_dim == 128) {
    if (causal) { if (win1) LAUNCH_FP(128, true, true); else LAUNCH_FP(128, true, false); }
    else { if (win1) LAUNCH_FP(128, false, true); else LAUNCH_FP(128, false, false); }
  } else {
    if (causal) { if (win1) LAUNCH_FP(64, true, true); else LAUNCH_FP(64, true, false); }
    else { if (win1) LAUNCH_FP(64, false, true); else LAUNCH_FP(64, false, false); }
  }
#undef LAUNCH_FP
  HIP_CHECK_KERNEL();
  return hipSuccess;
}
