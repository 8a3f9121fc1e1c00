// Paged attention, decode (single query token per sequence).
//
// MI355X-first design (memory-bound KV read; CDNA guide Appendix B):
//  - workgroup = (seq, kv_head), 4 waves; each wave owns one q head of the
//    GQA group so the K/V tiles staged in LDS are read once per GROUP, not
//    once per q head (4x less HBM traffic at group=4).
//  - KV pages [block=16 tokens][D] staged via 16 B/lane vector loads into
//    LDS padded to 272 B rows (4-way bank aliasing worst case, G4).
//  - online softmax in registers; V-phase reads are bank-conflict-free
//    (lane -> consecutive 4 B within a row).
//  - split-context (flash-decode style) for small batches: partials
//    (acc, m, l) to workspace, fused reduction kernel.
//
// Replaces the vLLM paged-attention path the reference delegates to
// (SURVEY.md §2.7 row 1).
#include "common.h"
#include "mfma_layouts.h"

#define NEG_INF (-1e30f)

namespace {

constexpr int PAGE = 16;      // tokens per KV page
constexpr int NWAVES = 4;     // waves per workgroup

// D: head dim (64 or 128). ACC = D/64 output dims per lane.
// HPW: q heads per wave (GQA group = NWAVES*HPW covered per workgroup).
// OCC: min waves/EU hint (occupancy ablation KS_ATTN_OCC; 1 = compiler's
// choice, 127 VGPRs -> 4 waves/SIMD on gfx950)
// PB: broadcast the softmax probabilities through LDS (1 write + 4 b128
// reads) instead of 16 sequential ds_bpermute shuffles per page
// (ablation KS_ATTN_PB — probing whether the DS pipe bounds the loop)
// D2: QK dot via v_dot2_f32_bf16 on packed bf16 pairs (halves the dot
// VALU ops and the q-fragment registers; ablation KS_ATTN_D2)
typedef __bf16 bf16x2v_t __attribute__((ext_vector_type(2)));

// V4: wide V loads — each lane reads 16 B (8 dims) of one token row, 2
// tokens per lane per page (4 dwordx4 instead of 16 dword loads); the
// per-dim accumulator holds a 2-token partial and the cross-token sum is
// deferred to a single epilogue reduction (ablation KS_ATTN_V4)
template <int D, int HPW, int OCC = 1, bool PB = false, bool D2 = false,
          bool V4 = false, bool MQ = false>
__global__ __launch_bounds__(256, OCC) void paged_attention_kernel(
    short* __restrict__ out,            // [S, H, D] bf16
    const short* __restrict__ q,        // [S, H, D]
    const short* __restrict__ k_cache,  // [B, Hkv, PAGE, D]
    const short* __restrict__ v_cache,
    const int* __restrict__ block_tables,  // [S, max_blocks]
    const int* __restrict__ context_lens,  // [S]
    const float scale,
    const int num_kv_heads,
    const int group,        // q heads per kv head
    const int max_blocks,
    const long q_row_stride,  // elements between q token rows
    // split-context: partial output when n_splits > 1
    const int n_splits,
    float* __restrict__ part_out,  // [S, H, n_splits, D]
    float* __restrict__ part_ml,   // [S, H, n_splits, 2] (m, l)
    const int window               // sliding window (0 = full attention)
) {
  constexpr int ACC = D / 64;
  const int kv_head = blockIdx.x;
  const int seq = blockIdx.y;
  const int split = blockIdx.z;
  const int ctx = context_lens[seq];
  if (ctx <= 0) return;
  const int num_heads = num_kv_heads * group;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int tok = lane >> 2;   // token within page handled by this lane
  const int part = lane & 3;   // quarter of D handled in the QK phase
  __shared__ float p_bc[NWAVES][PAGE];  // PB variant: per-wave p broadcast

  const int nblocks = (ctx + PAGE - 1) / PAGE;
  // split-context range
  // sliding window: only pages holding the last `window` tokens are read
  const int wstart = (window > 0 && ctx > window) ? ctx - window : 0;
  const int first_blk = wstart / PAGE;
  const int blocks_per_split =
      (nblocks - first_blk + n_splits - 1) / n_splits;
  const int blk_lo = first_blk + split * blocks_per_split;
  const int blk_hi = min(nblocks, blk_lo + blocks_per_split);
  if (blk_lo >= blk_hi) {
    // record empty partial so the reducer can skip it
    if (n_splits > 1 && threadIdx.x < (unsigned)group) {
      const int h = kv_head * group + threadIdx.x;
      float* ml = part_ml + (((long)seq * num_heads + h) * n_splits + split) * 2;
      ml[0] = NEG_INF;
      ml[1] = 0.f;
    }
    return;
  }

  // q fragments for this wave's heads: dims [part*D/4, (part+1)*D/4) f32
  constexpr int QFRAG = D / 4;  // dims per part
  float q_frag[HPW][D2 ? 1 : QFRAG];
  bf16x2v_t q_pk[HPW][D2 ? QFRAG / 2 : 1];  // packed-bf16 variant (D2)
  int heads[HPW];
  bool hact[HPW];
  bool any_active = false;
  const int nw = blockDim.x >> 6;  // active waves (4, or 2 for L1-relief)
#pragma unroll
  for (int h = 0; h < HPW; ++h) {
    const int head = kv_head * group + wave + h * nw;
    heads[h] = head;
    hact[h] = (wave + h * nw) < group;
    any_active |= hact[h];
    if (hact[h]) {
      const short* qp =
          q + (long)seq * q_row_stride + (long)head * D + part * QFRAG;
      if constexpr (D2) {
        const bf16x2v_t* qp2 = reinterpret_cast<const bf16x2v_t*>(qp);
#pragma unroll
        for (int j = 0; j < QFRAG / 2; ++j) q_pk[h][j] = qp2[j];
      } else {
#pragma unroll
        for (int j = 0; j < QFRAG; ++j)
          q_frag[h][j] = bf16_bits_to_float(qp[j]);
      }
    }
  }

  // MQ: the wave's q row as MFMA A-fragments (lane supplies k-chunk
  // (lane>>4)*8 of every 32-dim mfma step; row index is lane&15 but all
  // lanes load the head's own q — rows 1-15 of C are ignored)
  bf16x8_t q_mf[MQ ? D / 32 : 1];
  if constexpr (MQ) {
    const short* qb = q + (long)seq * q_row_stride + (long)heads[0] * D;
#pragma unroll
    for (int ki = 0; ki < D / 32; ++ki)
      q_mf[ki] = *reinterpret_cast<const bf16x8_t*>(
          qb + ki * 32 + ((lane >> 4) << 3));
  }

  float m[HPW], l[HPW];
  float acc[HPW][ACC];
  float acc8[V4 ? 8 : 1];  // V4: 8-dim 2-token partial accumulator
#pragma unroll
  for (int h = 0; h < HPW; ++h) {
    m[h] = NEG_INF;
    l[h] = 0.f;
#pragma unroll
    for (int a = 0; a < ACC; ++a) acc[h][a] = 0.f;
  }
#pragma unroll
  for (int a = 0; a < (V4 ? 8 : 1); ++a) acc8[a] = 0.f;

  if (!any_active) return;
  // Direct-global K/V reads, no LDS, no barriers: each page is read once per
  // workgroup group-wise (4 waves share it through L1/L2); barrier-free
  // iterations let the compiler keep many loads in flight across pages
  // (CDNA guide common-mistake #7: don't stage what the cache covers).
  const int bt_base = (int)((long)seq * max_blocks);
  for (int bi = blk_lo; bi < blk_hi; ++bi) {
    const int block_id = block_tables[bt_base + bi];
    const short* page =
        k_cache + (((long)block_id * num_kv_heads + kv_head) * PAGE) * D;
    // ---- QK: lane computes a quarter-dot for its token (coalesced: the
    // wave's 64 lanes cover the full 4 KB page); K is read ONCE for all
    // HPW heads of this wave ----
    const short8_t* kp =
        reinterpret_cast<const short8_t*>(page + tok * D + part * QFRAG);
    const int gtok = bi * PAGE + tok;
    const bool tok_valid = gtok < ctx && gtok >= wstart;
    const short* vpage =
        v_cache + (((long)block_id * num_kv_heads + kv_head) * PAGE) * D;

    if constexpr (MQ && HPW == 1 && V4) {
      // ---- MFMA QK (round-2 experiment): the page's 16 tokens form the
      // B columns of four 16x16x32 MFMA steps (K^T fragments are
      // contiguous 16-B reads: lane reads dims (lane>>4)*8 of token
      // lane&15), freeing the VALU dot + 4-lane reduce entirely. Scores
      // land in C row 0 = acc reg 0 of lanes 0-15 (token = lane). PV
      // stays on the V4 wide-load path. ----
      short8_t vwide4[4];
      const int vdim = (lane & 15) * 8;
      const int vtok4 = lane >> 4;
#pragma unroll
      for (int q4 = 0; q4 < 4; ++q4)
        vwide4[q4] = *reinterpret_cast<const short8_t*>(
            vpage + (q4 * 4 + vtok4) * D + vdim);
      f32x4_t cfrag = f32x4_t{0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int ki = 0; ki < D / 32; ++ki) {
        const bf16x8_t bk = *reinterpret_cast<const bf16x8_t*>(
            page + (lane & 15) * D + ki * 32 + ((lane >> 4) << 3));
        cfrag = mfma16x16x32(q_mf[ki], bk, cfrag);
      }
      const int mytok = bi * PAGE + (lane & 15);
      const bool my_valid =
          lane < 16 && mytok < ctx && mytok >= wstart;
      float s = my_valid ? cfrag[0] * scale : NEG_INF;
      float tmax = s;
#pragma unroll
      for (int off = 1; off < 64; off <<= 1)
        tmax = fmaxf(tmax, __shfl_xor(tmax, off, 64));
      const float m_new = fmaxf(m[0], tmax);
      const float rescale = __expf(m[0] - m_new);
      const float p = (s > NEG_INF) ? __expf(s - m_new) : 0.f;
      float psum = p;
#pragma unroll
      for (int off = 1; off < 64; off <<= 1)
        psum += __shfl_xor(psum, off, 64);
      l[0] = l[0] * rescale + psum;
      m[0] = m_new;
      if (lane < 16) p_bc[wave][lane] = p;
      asm volatile("s_waitcnt lgkmcnt(0)");
#pragma unroll
      for (int a = 0; a < 8; ++a) acc8[a] *= rescale;
#pragma unroll
      for (int q4 = 0; q4 < 4; ++q4) {
        const float pt = p_bc[wave][q4 * 4 + vtok4];
#pragma unroll
        for (int a = 0; a < 8; ++a)
          acc8[a] += pt * bf16_bits_to_float(vwide4[q4][a]);
      }
      continue;
    }

    if constexpr (HPW == 1) {
      // fast path: both the K fragments and the V row words for the page
      // are ISSUED as one burst (8 KB/wave in flight) before any use, and
      // the softmax update is branchless (NEG_INF is finite, so the
      // fully-masked-page case degenerates to a no-op: rescale=exp(0)=1,
      // p=0) — nothing blocks the scheduler from overlapping pages.
      short8_t kreg[QFRAG / 8];
#pragma unroll
      for (int c = 0; c < QFRAG / 8; ++c) kreg[c] = kp[c];
      unsigned int vreg[PAGE];
      short8_t vwide[V4 ? 4 : 1];  // V4: 8 dims (16 B) of one token row/load
      if constexpr (V4) {
        // 16 lanes span a 128-dim row; lane covers dims [8*(lane%16), +8)
        // of tokens lane/16, 4+lane/16, 8+lane/16, 12+lane/16
        const int vdim = (lane & 15) * 8;
        const int vtok = lane >> 4;
#pragma unroll
        for (int q4 = 0; q4 < 4; ++q4)
          vwide[q4] = *reinterpret_cast<const short8_t*>(
              vpage + (q4 * 4 + vtok) * D + vdim);
      } else {
#pragma unroll
        for (int t = 0; t < PAGE; ++t) {
          if constexpr (ACC == 2) {
            vreg[t] = *reinterpret_cast<const unsigned int*>(
                vpage + t * D + lane * ACC);
          } else {
            vreg[t] = (unsigned short)*(vpage + t * D + lane);
          }
        }
      }
      float s = 0.f;
      if constexpr (D2) {
#pragma unroll
        for (int c = 0; c < QFRAG / 8; ++c) {
          const bf16x2v_t* k2 = reinterpret_cast<const bf16x2v_t*>(&kreg[c]);
#pragma unroll
          for (int j = 0; j < 4; ++j)
            s = __builtin_amdgcn_fdot2_f32_bf16(
                k2[j], q_pk[0][c * 4 + j], s, false);
        }
      } else {
#pragma unroll
        for (int c = 0; c < QFRAG / 8; ++c) {
#pragma unroll
          for (int j = 0; j < 8; ++j)
            s += q_frag[0][c * 8 + j] * bf16_bits_to_float(kreg[c][j]);
        }
      }
      s = group_reduce_sum<4>(s);
      s = tok_valid ? s * scale : NEG_INF;
      // s/p are uniform within each 4-lane token group: reducing across
      // the 16 groups needs only the xor offsets {4,8,16,32} (4 DS ops,
      // not the 6 of a full wave reduce)
      float tmax = s;
#pragma unroll
      for (int off = 4; off < 64; off <<= 1)
        tmax = fmaxf(tmax, __shfl_xor(tmax, off, 64));
      const float m_new = fmaxf(m[0], tmax);
      const float rescale = __expf(m[0] - m_new);
      const float p = (s > NEG_INF) ? __expf(s - m_new) : 0.f;
      // xor offsets {4,8,16,32} never mix the low 2 lane bits, so each
      // lane sums exactly one lane per token group: no overcount
      float psum = p;
#pragma unroll
      for (int off = 4; off < 64; off <<= 1)
        psum += __shfl_xor(psum, off, 64);
      l[0] = l[0] * rescale + psum;
#pragma unroll
      for (int a = 0; a < ACC; ++a) acc[0][a] *= rescale;
      m[0] = m_new;
      if constexpr (V4) {
        // V4 PV: lane multiplies its two token rows by their p (read from
        // the broadcast) and accumulates 8-dim partials; rescale applies
        // to partials exactly like the full sum
        if (part == 0) p_bc[wave][tok] = p;
        asm volatile("s_waitcnt lgkmcnt(0)");
        const int vtok = lane >> 4;
#pragma unroll
        for (int a = 0; a < 8; ++a) acc8[a] *= rescale;
#pragma unroll
        for (int q4 = 0; q4 < 4; ++q4) {
          const float pt = p_bc[wave][q4 * 4 + vtok];
#pragma unroll
          for (int a = 0; a < 8; ++a)
            acc8[a] += pt * bf16_bits_to_float(vwide[q4][a]);
        }
      } else if constexpr (PB) {
        // one LDS write per token group, then 4 x b128 reads give every
        // lane all 16 p values (same-wave, no barrier; lgkm wait only)
        if (part == 0) p_bc[wave][tok] = p;
        asm volatile("s_waitcnt lgkmcnt(0)");
        float4_t pv[PAGE / 4];
#pragma unroll
        for (int c = 0; c < PAGE / 4; ++c)
          pv[c] = reinterpret_cast<const float4_t*>(p_bc[wave])[c];
#pragma unroll
        for (int t = 0; t < PAGE; ++t) {
          const float pt = pv[t / 4][t % 4];
          if constexpr (ACC == 2) {
            acc[0][0] += pt * bf16_bits_to_float((short)(vreg[t] & 0xFFFF));
            acc[0][1] += pt * bf16_bits_to_float((short)(vreg[t] >> 16));
          } else {
            acc[0][0] += pt * bf16_bits_to_float((short)vreg[t]);
          }
        }
      } else {
#pragma unroll
        for (int t = 0; t < PAGE; ++t) {
          const float pt = __shfl(p, t * 4, 64);
          if constexpr (ACC == 2) {
            acc[0][0] += pt * bf16_bits_to_float((short)(vreg[t] & 0xFFFF));
            acc[0][1] += pt * bf16_bits_to_float((short)(vreg[t] >> 16));
          } else {
            acc[0][0] += pt * bf16_bits_to_float((short)vreg[t]);
          }
        }
      }
      continue;
    }

    // HPW >= 2: K/V buffered once per page and reused across heads
    float kbuf[QFRAG];
#pragma unroll
    for (int c = 0; c < QFRAG / 8; ++c) {
      short8_t kv8 = kp[c];
#pragma unroll
      for (int j = 0; j < 8; ++j)
        kbuf[c * 8 + j] = bf16_bits_to_float(kv8[j]);
    }
    float2_t vbuf[PAGE];
#pragma unroll
    for (int t = 0; t < PAGE; ++t) {
      const short* vrow = vpage + t * D + lane * ACC;
      if constexpr (ACC == 2) {
        const unsigned int packed =
            *reinterpret_cast<const unsigned int*>(vrow);
        vbuf[t][0] = bf16_bits_to_float((short)(packed & 0xFFFF));
        vbuf[t][1] = bf16_bits_to_float((short)(packed >> 16));
      } else {
        vbuf[t][0] = bf16_bits_to_float(vrow[0]);
        vbuf[t][1] = 0.f;
      }
    }
#pragma unroll
    for (int h = 0; h < HPW; ++h) {
      if (!hact[h]) continue;
      float s = 0.f;
#pragma unroll
      for (int j = 0; j < QFRAG; ++j) s += q_frag[h][j] * kbuf[j];
      s = group_reduce_sum<4>(s);  // full dot in all 4 lanes of the token
      s = tok_valid ? s * scale : NEG_INF;

      // ---- online softmax update (4-step reduces: s/p are uniform
      // within each 4-lane token group) ----
      float tmax = s;
#pragma unroll
      for (int off = 4; off < 64; off <<= 1)
        tmax = fmaxf(tmax, __shfl_xor(tmax, off, 64));
      if (tmax > NEG_INF) {
        const float m_new = fmaxf(m[h], tmax);
        const float rescale = (m[h] > NEG_INF) ? __expf(m[h] - m_new) : 0.f;
        const float p = (s > NEG_INF) ? __expf(s - m_new) : 0.f;
        float psum = p;
#pragma unroll
        for (int off = 4; off < 64; off <<= 1)
          psum += __shfl_xor(psum, off, 64);
        l[h] = l[h] * rescale + psum;
#pragma unroll
        for (int a = 0; a < ACC; ++a) acc[h][a] *= rescale;
        m[h] = m_new;

        // ---- PV via LDS p-broadcast (1 write + 4 b128 reads instead of
        // 16 ds_bpermute) ----
        if (part == 0) p_bc[wave][tok] = p;
        asm volatile("s_waitcnt lgkmcnt(0)");
        float4_t pv4[PAGE / 4];
#pragma unroll
        for (int c = 0; c < PAGE / 4; ++c)
          pv4[c] = reinterpret_cast<const float4_t*>(p_bc[wave])[c];
#pragma unroll
        for (int t = 0; t < PAGE; ++t) {
          const float pt = pv4[t / 4][t % 4];
#pragma unroll
          for (int a = 0; a < ACC; ++a) acc[h][a] += pt * vbuf[t][a];
        }
      }
    }
  }

#pragma unroll
  for (int h = 0; h < HPW; ++h) {
    if (!hact[h]) continue;
    const int head = heads[h];
    if constexpr (V4) {
      // cross token-slice reduction (deferred from the per-page PV): lanes
      // sharing lane%16 hold 4-token partials of the same 8 dims
#pragma unroll
      for (int a = 0; a < 8; ++a) {
        acc8[a] += __shfl_xor(acc8[a], 16, 64);
        acc8[a] += __shfl_xor(acc8[a], 32, 64);
      }
      const int vdim = (lane & 15) * 8;
      if (n_splits == 1) {
        const float inv_l = (l[h] > 0.f) ? 1.f / l[h] : 0.f;
        if (lane < 16) {
          short* op = out + ((long)seq * num_heads + head) * D + vdim;
#pragma unroll
          for (int a = 0; a < 8; ++a)
            op[a] = float_to_bf16_bits(acc8[a] * inv_l);
        }
      } else {
        if (lane < 16) {
          float* po =
              part_out +
              ((((long)seq * num_heads + head) * n_splits + split)) * D +
              vdim;
#pragma unroll
          for (int a = 0; a < 8; ++a) po[a] = acc8[a];
        }
        if (lane == 0) {
          float* ml = part_ml +
                      (((long)seq * num_heads + head) * n_splits + split) * 2;
          ml[0] = m[h];
          ml[1] = l[h];
        }
      }
      continue;
    }
    if (n_splits == 1) {
      const float inv_l = (l[h] > 0.f) ? 1.f / l[h] : 0.f;
      short* op = out + ((long)seq * num_heads + head) * D + lane * ACC;
#pragma unroll
      for (int a = 0; a < ACC; ++a)
        op[a] = float_to_bf16_bits(acc[h][a] * inv_l);
    } else {
      float* po =
          part_out +
          ((((long)seq * num_heads + head) * n_splits + split)) * D +
          lane * ACC;
#pragma unroll
      for (int a = 0; a < ACC; ++a) po[a] = acc[h][a];
      if (lane == 0) {
        float* ml =
            part_ml + (((long)seq * num_heads + head) * n_splits + split) * 2;
        ml[0] = m[h];
        ml[1] = l[h];
      }
    }
  }
}


// ---------------------------------------------------------------------------
// fp8 (OCP E4M3) KV-cache decode attention: the HPW=1 fast path with byte
// pages — half the KV bytes of bf16, converted pairwise in-register with
// v_cvt_pk_f32_fp8. D=128 (ACC=2), GQA group <= 4. Split-context shares the
// bf16 reduce kernel (partials are fp32 either way).
// ---------------------------------------------------------------------------
typedef unsigned int uint4_t __attribute__((ext_vector_type(4)));

// CVTB: QK chain via gfx950 scaled converts — fp8 pairs go straight to
// bf16 (cvt_scalef32_pk_bf16_fp8, exact: e4m3 fits bf16's mantissa) and
// dot against packed-bf16 q with fdot2_f32_bf16, replacing the
// cvt_pk_f32 + 4xFMA chain (48 -> 32 VALU per lane per page on QK).
template <int D, bool CVTB = false>  // D == 128
__global__ __launch_bounds__(256) void paged_attention_fp8_kernel(
    short* __restrict__ out, const short* __restrict__ q,
    const unsigned char* __restrict__ k_cache,
    const unsigned char* __restrict__ v_cache,
    const int* __restrict__ block_tables, const int* __restrict__ context_lens,
    const float scale, const int num_kv_heads, const int group,
    const int max_blocks, const long q_row_stride, const int n_splits,
    float* __restrict__ part_out, float* __restrict__ part_ml,
    const int window) {
  constexpr int ACC = D / 64;
  constexpr int QFRAG = D / 4;
  const int kv_head = blockIdx.x;
  const int seq = blockIdx.y;
  const int split = blockIdx.z;
  const int ctx = context_lens[seq];
  if (ctx <= 0) return;
  const int num_heads = num_kv_heads * group;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int tok = lane >> 2;
  const int part = lane & 3;

  const int nblocks = (ctx + PAGE - 1) / PAGE;
  // sliding window: only pages holding the last `window` tokens are read
  const int wstart = (window > 0 && ctx > window) ? ctx - window : 0;
  const int first_blk = wstart / PAGE;
  const int blocks_per_split =
      (nblocks - first_blk + n_splits - 1) / n_splits;
  const int blk_lo = first_blk + split * blocks_per_split;
  const int blk_hi = min(nblocks, blk_lo + blocks_per_split);
  const int head = kv_head * group + wave;
  const bool active = wave < group;
  if (blk_lo >= blk_hi) {
    if (n_splits > 1 && threadIdx.x < (unsigned)group) {
      const int h = kv_head * group + threadIdx.x;
      float* ml = part_ml + (((long)seq * num_heads + h) * n_splits + split) * 2;
      ml[0] = NEG_INF;
      ml[1] = 0.f;
    }
    return;
  }
  if (!active) return;

  __shared__ float p_bc8[4][PAGE];
  float q_frag[CVTB ? 1 : QFRAG];
  bf16x2v_t q_pk8[CVTB ? QFRAG / 2 : 1];
  {
    const short* qp =
        q + (long)seq * q_row_stride + (long)head * D + part * QFRAG;
    if constexpr (CVTB) {
      const bf16x2v_t* qp2 = reinterpret_cast<const bf16x2v_t*>(qp);
#pragma unroll
      for (int j = 0; j < QFRAG / 2; ++j) q_pk8[j] = qp2[j];
    } else {
#pragma unroll
      for (int j = 0; j < QFRAG; ++j) q_frag[j] = bf16_bits_to_float(qp[j]);
    }
  }
  float m = NEG_INF, l = 0.f;
  // wide-V scheme (same as the bf16 V4 path): 8 lanes span a 128-B fp8
  // row; lane covers dims [16*(lane%8), +16) of tokens lane/8 and
  // 8+lane/8; accumulator keeps 2-token partials, summed in the epilogue
  float acc[16];
#pragma unroll
  for (int a = 0; a < 16; ++a) acc[a] = 0.f;
  const int vdim = (lane & 7) * 16;
  const int vtok8 = lane >> 3;

  const int bt_base = (int)((long)seq * max_blocks);
  for (int bi = blk_lo; bi < blk_hi; ++bi) {
    const int block_id = block_tables[bt_base + bi];
    const unsigned char* page =
        k_cache + (((long)block_id * num_kv_heads + kv_head) * PAGE) * D;
    const unsigned char* vpage =
        v_cache + (((long)block_id * num_kv_heads + kv_head) * PAGE) * D;
    const int gtok = bi * PAGE + tok;
    const bool tok_valid = gtok < ctx && gtok >= wstart;
    // burst-issue: K quarter-row (32 B) + V 16-B row chunks (2 tokens)
    uint4_t kreg[QFRAG / 16];
#pragma unroll
    for (int c = 0; c < QFRAG / 16; ++c)
      kreg[c] = reinterpret_cast<const uint4_t*>(
          page + tok * D + part * QFRAG)[c];
    uint4_t vwide[2];
#pragma unroll
    for (int q8 = 0; q8 < 2; ++q8)
      vwide[q8] = *reinterpret_cast<const uint4_t*>(
          vpage + (q8 * 8 + vtok8) * D + vdim);
    float s = 0.f;
#pragma unroll
    for (int c = 0; c < QFRAG / 16; ++c) {
#pragma unroll
      for (int w = 0; w < 4; ++w) {
        const int ui = (int)kreg[c][w];
        const int j = c * 16 + w * 4;
        if constexpr (CVTB) {
          const bf16x2v_t lo =
              __builtin_amdgcn_cvt_scalef32_pk_bf16_fp8(ui, 1.0f, false);
          const bf16x2v_t hi =
              __builtin_amdgcn_cvt_scalef32_pk_bf16_fp8(ui, 1.0f, true);
          s = __builtin_amdgcn_fdot2_f32_bf16(lo, q_pk8[j / 2], s, false);
          s = __builtin_amdgcn_fdot2_f32_bf16(hi, q_pk8[j / 2 + 1], s,
                                              false);
        } else {
          const float2_t lo = __builtin_amdgcn_cvt_pk_f32_fp8(ui, false);
          const float2_t hi = __builtin_amdgcn_cvt_pk_f32_fp8(ui, true);
          s += q_frag[j] * lo[0] + q_frag[j + 1] * lo[1] +
               q_frag[j + 2] * hi[0] + q_frag[j + 3] * hi[1];
        }
      }
    }
    s = group_reduce_sum<4>(s);
    s = tok_valid ? s * scale : NEG_INF;
    const float tmax = wave_reduce_max(s);
    const float m_new = fmaxf(m, tmax);
    const float rescale = __expf(m - m_new);
    const float p = (s > NEG_INF) ? __expf(s - m_new) : 0.f;
    const float psum = wave_reduce_sum(part == 0 ? p : 0.f);
    l = l * rescale + psum;
#pragma unroll
    for (int a = 0; a < 16; ++a) acc[a] *= rescale;
    m = m_new;
    if (part == 0) p_bc8[wave][tok] = p;
    asm volatile("s_waitcnt lgkmcnt(0)");
#pragma unroll
    for (int q8 = 0; q8 < 2; ++q8) {
      const float pt = p_bc8[wave][q8 * 8 + vtok8];
#pragma unroll
      for (int w = 0; w < 4; ++w) {
        const int ui = (int)vwide[q8][w];
        const float2_t lo = __builtin_amdgcn_cvt_pk_f32_fp8(ui, false);
        const float2_t hi = __builtin_amdgcn_cvt_pk_f32_fp8(ui, true);
        acc[w * 4 + 0] += pt * lo[0];
        acc[w * 4 + 1] += pt * lo[1];
        acc[w * 4 + 2] += pt * hi[0];
        acc[w * 4 + 3] += pt * hi[1];
      }
    }
  }

  // sum the 2-token partials: lanes sharing lane%8 (xor 8,16,32)
#pragma unroll
  for (int a = 0; a < 16; ++a) {
#pragma unroll
    for (int off = 8; off < 64; off <<= 1)
      acc[a] += __shfl_xor(acc[a], off, 64);
  }
  if (n_splits == 1) {
    const float inv_l = (l > 0.f) ? 1.f / l : 0.f;
    if (lane < 8) {
      short* op = out + ((long)seq * num_heads + head) * D + vdim;
#pragma unroll
      for (int a = 0; a < 16; ++a)
        op[a] = float_to_bf16_bits(acc[a] * inv_l);
    }
  } else {
    if (lane < 8) {
      float* po = part_out +
                  ((((long)seq * num_heads + head) * n_splits + split)) * D +
                  vdim;
#pragma unroll
      for (int a = 0; a < 16; ++a) po[a] = acc[a];
    }
    if (lane == 0) {
      float* ml =
          part_ml + (((long)seq * num_heads + head) * n_splits + split) * 2;
      ml[0] = m;
      ml[1] = l;
    }
  }
}

// ---------------------------------------------------------------------------
// 2-page unrolled variant of the HPW=1 fast path: both pages' K+V bursts
// (16 KB/wave) are issued before either page's compute, doubling the
// bytes-in-flight per wave of the latency-bound loop.
// ---------------------------------------------------------------------------
template <int D>
__global__ __launch_bounds__(256) void paged_attention_u2_kernel(
    short* __restrict__ out, const short* __restrict__ q,
    const short* __restrict__ k_cache, const short* __restrict__ v_cache,
    const int* __restrict__ block_tables, const int* __restrict__ context_lens,
    const float scale, const int num_kv_heads, const int group,
    const int max_blocks, const long q_row_stride, const int n_splits,
    float* __restrict__ part_out, float* __restrict__ part_ml,
    const int window) {
  constexpr int ACC = D / 64;
  constexpr int QFRAG = D / 4;
  const int kv_head = blockIdx.x;
  const int seq = blockIdx.y;
  const int split = blockIdx.z;
  const int ctx = context_lens[seq];
  if (ctx <= 0) return;
  const int num_heads = num_kv_heads * group;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int tok = lane >> 2;
  const int part = lane & 3;

  const int nblocks = (ctx + PAGE - 1) / PAGE;
  // sliding window: only pages holding the last `window` tokens are read
  const int wstart = (window > 0 && ctx > window) ? ctx - window : 0;
  const int first_blk = wstart / PAGE;
  const int blocks_per_split =
      (nblocks - first_blk + n_splits - 1) / n_splits;
  const int blk_lo = first_blk + split * blocks_per_split;
  const int blk_hi = min(nblocks, blk_lo + blocks_per_split);
  if (blk_lo >= blk_hi) {
    if (n_splits > 1 && threadIdx.x < (unsigned)group) {
      const int h = kv_head * group + threadIdx.x;
      float* ml = part_ml + (((long)seq * num_heads + h) * n_splits + split) * 2;
      ml[0] = NEG_INF;
      ml[1] = 0.f;
    }
    return;
  }
  const int head = kv_head * group + wave;
  if (wave >= group) return;

  __shared__ float p_bc8[4][PAGE];
  float q_frag[QFRAG];
  {
    const short* qp =
        q + (long)seq * q_row_stride + (long)head * D + part * QFRAG;
#pragma unroll
    for (int j = 0; j < QFRAG; ++j) q_frag[j] = bf16_bits_to_float(qp[j]);
  }
  float m = NEG_INF, l = 0.f;
  // wide-V scheme (same as the bf16 V4 path): 8 lanes span a 128-B fp8
  // row; lane covers dims [16*(lane%8), +16) of tokens lane/8 and
  // 8+lane/8; accumulator keeps 2-token partials, summed in the epilogue
  float acc[16];
#pragma unroll
  for (int a = 0; a < 16; ++a) acc[a] = 0.f;
  const int vdim = (lane & 7) * 16;
  const int vtok8 = lane >> 3;

  const int bt_base = (int)((long)seq * max_blocks);

  auto load_page = [&](int bi, short8_t* kreg, unsigned int* vreg) {
    const int block_id = block_tables[bt_base + bi];
    const long pbase = (((long)block_id * num_kv_heads + kv_head) * PAGE) * D;
    const short8_t* kp =
        reinterpret_cast<const short8_t*>(k_cache + pbase + tok * D + part * QFRAG);
#pragma unroll
    for (int c = 0; c < QFRAG / 8; ++c) kreg[c] = kp[c];
    const short* vpage = v_cache + pbase;
#pragma unroll
    for (int t = 0; t < PAGE; ++t) {
      if constexpr (ACC == 2) {
        vreg[t] =
            *reinterpret_cast<const unsigned int*>(vpage + t * D + lane * ACC);
      } else {
        vreg[t] = (unsigned short)*(vpage + t * D + lane);
      }
    }
  };

  auto compute_page = [&](int bi, const short8_t* kreg,
                          const unsigned int* vreg) {
    float s = 0.f;
#pragma unroll
    for (int c = 0; c < QFRAG / 8; ++c) {
#pragma unroll
      for (int j = 0; j < 8; ++j)
        s += q_frag[c * 8 + j] * bf16_bits_to_float(kreg[c][j]);
    }
    s = group_reduce_sum<4>(s);
    const int gtok = bi * PAGE + tok;
    s = (gtok < ctx && gtok >= wstart) ? s * scale : NEG_INF;
    const float tmax = wave_reduce_max(s);
    const float m_new = fmaxf(m, tmax);
    const float rescale = __expf(m - m_new);
    const float p = (s > NEG_INF) ? __expf(s - m_new) : 0.f;
    const float psum = wave_reduce_sum(part == 0 ? p : 0.f);
    l = l * rescale + psum;
#pragma unroll
    for (int a = 0; a < ACC; ++a) acc[a] *= rescale;
    m = m_new;
#pragma unroll
    for (int t = 0; t < PAGE; ++t) {
      const float pt = __shfl(p, t * 4, 64);
      if constexpr (ACC == 2) {
        acc[0] += pt * bf16_bits_to_float((short)(vreg[t] & 0xFFFF));
        acc[1] += pt * bf16_bits_to_float((short)(vreg[t] >> 16));
      } else {
        acc[0] += pt * bf16_bits_to_float((short)vreg[t]);
      }
    }
  };

  short8_t kA[QFRAG / 8], kB[QFRAG / 8];
  unsigned int vA[PAGE], vB[PAGE];
  int bi = blk_lo;
  for (; bi + 1 < blk_hi; bi += 2) {
    load_page(bi, kA, vA);
    load_page(bi + 1, kB, vB);
    compute_page(bi, kA, vA);
    compute_page(bi + 1, kB, vB);
  }
  if (bi < blk_hi) {
    load_page(bi, kA, vA);
    compute_page(bi, kA, vA);
  }

  if (n_splits == 1) {
    const float inv_l = (l > 0.f) ? 1.f / l : 0.f;
    short* op = out + ((long)seq * num_heads + head) * D + lane * ACC;
#pragma unroll
    for (int a = 0; a < ACC; ++a) op[a] = float_to_bf16_bits(acc[a] * inv_l);
  } else {
    float* po = part_out +
                (((long)seq * num_heads + head) * n_splits + split) * D +
                lane * ACC;
#pragma unroll
    for (int a = 0; a < ACC; ++a) po[a] = acc[a];
    if (lane == 0) {
      float* ml =
          part_ml + (((long)seq * num_heads + head) * n_splits + split) * 2;
      ml[0] = m;
      ml[1] = l;
    }
  }
}

// ---------------------------------------------------------------------------
// Wave-split variant: the 4 waves partition the PAGES (each page is read by
// exactly one wave — no cross-wave L1 re-reads), every wave computes ALL
// GROUP heads for its pages, and the four per-wave online-softmax partials
// are merged through LDS at the end. Q is kept as bf16 fragments
// (GROUP x 16 VGPR) and converted in the dot.
// ---------------------------------------------------------------------------
template <int D, int GROUP>
__global__ __launch_bounds__(256) void paged_attention_ws_kernel(
    short* __restrict__ out, const short* __restrict__ q,
    const short* __restrict__ k_cache, const short* __restrict__ v_cache,
    const int* __restrict__ block_tables, const int* __restrict__ context_lens,
    const float scale, const int num_kv_heads, const int max_blocks,
    const long q_row_stride, const int n_splits,
    float* __restrict__ part_out, float* __restrict__ part_ml,
    const int window) {
  constexpr int ACC = D / 64;
  constexpr int QFRAG = D / 4;
  const int kv_head = blockIdx.x;
  const int seq = blockIdx.y;
  const int split = blockIdx.z;
  const int ctx = context_lens[seq];
  if (ctx <= 0) return;
  const int num_heads = num_kv_heads * GROUP;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int tok = lane >> 2;
  const int part = lane & 3;

  const int nblocks = (ctx + PAGE - 1) / PAGE;
  // sliding window: only pages holding the last `window` tokens are read
  const int wstart = (window > 0 && ctx > window) ? ctx - window : 0;
  const int first_blk = wstart / PAGE;
  const int blocks_per_split =
      (nblocks - first_blk + n_splits - 1) / n_splits;
  const int blk_lo = first_blk + split * blocks_per_split;
  const int blk_hi = min(nblocks, blk_lo + blocks_per_split);

  __shared__ float lds_ml[NWAVES][GROUP][2];
  __shared__ float lds_acc[NWAVES][GROUP][64 * ACC];

  if (blk_lo >= blk_hi) {
    if (n_splits > 1 && threadIdx.x < (unsigned)GROUP) {
      const int h = kv_head * GROUP + threadIdx.x;
      float* ml = part_ml + (((long)seq * num_heads + h) * n_splits + split) * 2;
      ml[0] = NEG_INF;
      ml[1] = 0.f;
    }
    return;
  }

  // q rows stay in L1 (re-read per page) instead of registers: GROUP x 16
  // VGPRs of persistent fragments would cap occupancy at 1-2 waves/SIMD
  const short8_t* qrows[GROUP];
#pragma unroll
  for (int g = 0; g < GROUP; ++g) {
    qrows[g] = reinterpret_cast<const short8_t*>(
        q + (long)seq * q_row_stride + (long)(kv_head * GROUP + g) * D +
        part * QFRAG);
  }

  float m[GROUP], l[GROUP], acc[GROUP][ACC];
#pragma unroll
  for (int g = 0; g < GROUP; ++g) {
    m[g] = NEG_INF;
    l[g] = 0.f;
#pragma unroll
    for (int a = 0; a < ACC; ++a) acc[g][a] = 0.f;
  }

  const int bt_base = (int)((long)seq * max_blocks);
  for (int bi = blk_lo + wave; bi < blk_hi; bi += NWAVES) {
    const int block_id = block_tables[bt_base + bi];
    const long pbase = (((long)block_id * num_kv_heads + kv_head) * PAGE) * D;
    const short8_t* kp =
        reinterpret_cast<const short8_t*>(k_cache + pbase + tok * D + part * QFRAG);
    short8_t kreg[QFRAG / 8];
#pragma unroll
    for (int c = 0; c < QFRAG / 8; ++c) kreg[c] = kp[c];
    const short* vpage = v_cache + pbase;
    unsigned int vreg[PAGE];
#pragma unroll
    for (int t = 0; t < PAGE; ++t) {
      if constexpr (ACC == 2) {
        vreg[t] =
            *reinterpret_cast<const unsigned int*>(vpage + t * D + lane * ACC);
      } else {
        vreg[t] = (unsigned short)*(vpage + t * D + lane);
      }
    }
    const int gtok = bi * PAGE + tok;
    const bool tok_valid = gtok < ctx && gtok >= wstart;
#pragma unroll 1
    for (int g = 0; g < GROUP; ++g) {
      float s = 0.f;
#pragma unroll
      for (int c = 0; c < QFRAG / 8; ++c) {
        const short8_t qv = qrows[g][c];
#pragma unroll
        for (int j = 0; j < 8; ++j)
          s += bf16_bits_to_float(qv[j]) * bf16_bits_to_float(kreg[c][j]);
      }
      s = group_reduce_sum<4>(s);
      s = tok_valid ? s * scale : NEG_INF;
      const float tmax = wave_reduce_max(s);
      const float m_new = fmaxf(m[g], tmax);
      const float rescale = __expf(m[g] - m_new);
      const float p = (s > NEG_INF) ? __expf(s - m_new) : 0.f;
      const float psum = wave_reduce_sum(part == 0 ? p : 0.f);
      l[g] = l[g] * rescale + psum;
#pragma unroll
      for (int a = 0; a < ACC; ++a) acc[g][a] *= rescale;
      m[g] = m_new;
#pragma unroll
      for (int t = 0; t < PAGE; ++t) {
        const float pt = __shfl(p, t * 4, 64);
        if constexpr (ACC == 2) {
          acc[g][0] += pt * bf16_bits_to_float((short)(vreg[t] & 0xFFFF));
          acc[g][1] += pt * bf16_bits_to_float((short)(vreg[t] >> 16));
        } else {
          acc[g][0] += pt * bf16_bits_to_float((short)vreg[t]);
        }
      }
    }
  }

  // ---- cross-wave merge via LDS ----
#pragma unroll
  for (int g = 0; g < GROUP; ++g) {
    if (lane == 0) {
      lds_ml[wave][g][0] = m[g];
      lds_ml[wave][g][1] = l[g];
    }
#pragma unroll
    for (int a = 0; a < ACC; ++a)
      lds_acc[wave][g][lane * ACC + a] = acc[g][a];
  }
  __syncthreads();
  // wave w merges heads {w, w+4, ...}
  for (int g = wave; g < GROUP; g += NWAVES) {
    float m_all = NEG_INF;
#pragma unroll
    for (int w = 0; w < NWAVES; ++w) m_all = fmaxf(m_all, lds_ml[w][g][0]);
    float l_all = 0.f;
    float a_all[ACC];
#pragma unroll
    for (int a = 0; a < ACC; ++a) a_all[a] = 0.f;
#pragma unroll
    for (int w = 0; w < NWAVES; ++w) {
      const float wgt = __expf(lds_ml[w][g][0] - m_all);
      l_all += wgt * lds_ml[w][g][1];
#pragma unroll
      for (int a = 0; a < ACC; ++a)
        a_all[a] += wgt * lds_acc[w][g][lane * ACC + a];
    }
    const int head = kv_head * GROUP + g;
    if (n_splits == 1) {
      const float inv_l = (l_all > 0.f) ? 1.f / l_all : 0.f;
      short* op = out + ((long)seq * num_heads + head) * D + lane * ACC;
#pragma unroll
      for (int a = 0; a < ACC; ++a)
        op[a] = float_to_bf16_bits(a_all[a] * inv_l);
    } else {
      float* po = part_out +
                  (((long)seq * num_heads + head) * n_splits + split) * D +
                  lane * ACC;
#pragma unroll
      for (int a = 0; a < ACC; ++a) po[a] = a_all[a];
      if (lane == 0) {
        float* ml =
            part_ml + (((long)seq * num_heads + head) * n_splits + split) * 2;
        ml[0] = m_all;
        ml[1] = l_all;
      }
    }
  }
}

// Combine split-context partials: one wave per (seq, head); lane owns D/64
// dims across all splits.
template <int D>
__global__ void paged_attention_reduce_kernel(
    short* __restrict__ out,             // [S, H, D]
    const float* __restrict__ part_out,  // [S, H, n_splits, D]
    const float* __restrict__ part_ml,   // [S, H, n_splits, 2]
    const long total_sh, const int n_splits) {
  constexpr int ACC = D / 64;
  const long sh = blockIdx.x * (long)(blockDim.x >> 6) + (threadIdx.x >> 6);
  const int lane = threadIdx.x & 63;
  if (sh >= total_sh) return;
  const float* ml_base = part_ml + sh * n_splits * 2;
  float m = NEG_INF;
  for (int sp = 0; sp < n_splits; ++sp) m = fmaxf(m, ml_base[sp * 2]);
  float l = 0.f;
  float acc[ACC];
#pragma unroll
  for (int a = 0; a < ACC; ++a) acc[a] = 0.f;
  for (int sp = 0; sp < n_splits; ++sp) {
    const float ms = ml_base[sp * 2];
    const float ls = ml_base[sp * 2 + 1];
    if (ls <= 0.f || ms <= NEG_INF) continue;
    const float w = __expf(ms - m);
    l += ls * w;
    const float* po = part_out + (sh * n_splits + sp) * D + lane * ACC;
#pragma unroll
    for (int a = 0; a < ACC; ++a) acc[a] += w * po[a];
  }
  const float inv_l = (l > 0.f) ? 1.f / l : 0.f;
  short* op = out + sh * D + lane * ACC;
#pragma unroll
  for (int a = 0; a < ACC; ++a) op[a] = float_to_bf16_bits(acc[a] * inv_l);
}

}  // namespace


namespace {
// shared split-context reduction launch (used by every kernel variant)
inline hipError_t launch_split_reduce(void* out, void* part_out,
                                      void* part_ml, int num_seqs,
                                      int num_heads, int head_dim,
                                      int n_splits, hipStream_t stream) {
  if (n_splits <= 1) return hipSuccess;
  const long sh = (long)num_seqs * num_heads;
  const int wpb = 4;
  dim3 rgrid((unsigned)((sh + wpb - 1) / wpb));
  if (head_dim == 128) {
    hipLaunchKernelGGL((paged_attention_reduce_kernel<128>), rgrid,
                       dim3(wpb * 64), 0, stream, (short*)out,
                       (const float*)part_out, (const float*)part_ml, sh,
                       n_splits);
  } else {
    hipLaunchKernelGGL((paged_attention_reduce_kernel<64>), rgrid,
                       dim3(wpb * 64), 0, stream, (short*)out,
                       (const float*)part_out, (const float*)part_ml, sh,
                       n_splits);
  }
  HIP_CHECK_KERNEL();
  return hipSuccess;
}
}  // namespace

extern "C" hipError_t ks_paged_attention_decode_fp8(
    void* out, const void* q, const void* k_cache, const void* v_cache,
    const void* block_tables, const void* context_lens, float scale,
    int num_seqs, int num_heads, int num_kv_heads, int head_dim,
    int max_blocks, int block_size, long q_row_stride, int n_splits,
    void* part_out, void* part_ml, int window, hipStream_t stream) {
  if (block_size != PAGE || head_dim != 128) return hipErrorInvalidValue;
  const int group = num_heads / num_kv_heads;
  if (group > NWAVES) return hipErrorInvalidValue;  // fp8: GQA group <= 4
  if (n_splits < 1) n_splits = 1;
  dim3 grid(num_kv_heads, num_seqs, n_splits);
  // scaled-convert QK chain (fp8->bf16 + fdot2) is the default;
  // KS_FP8_CVTB=0 reverts to the cvt_pk_f32 + FMA chain for A/B.
  static const bool cvtb = [] {
    const char* e = getenv("KS_FP8_CVTB");
    return e == nullptr || e[0] != '0';
  }();
  if (cvtb) {
    hipLaunchKernelGGL((paged_attention_fp8_kernel<128, true>), grid,
                       dim3(256), 0, stream, (short*)out, (const short*)q,
                       (const unsigned char*)k_cache,
                       (const unsigned char*)v_cache,
                       (const int*)block_tables, (const int*)context_lens,
                       scale, num_kv_heads, group, max_blocks, q_row_stride,
                       n_splits, (float*)part_out, (float*)part_ml, window);
  } else {
    hipLaunchKernelGGL((paged_attention_fp8_kernel<128, false>), grid,
                       dim3(256), 0, stream, (short*)out, (const short*)q,
                       (const unsigned char*)k_cache,
                       (const unsigned char*)v_cache,
                       (const int*)block_tables, (const int*)context_lens,
                       scale, num_kv_heads, group, max_blocks, q_row_stride,
                       n_splits, (float*)part_out, (float*)part_ml, window);
  }
  HIP_CHECK_KERNEL();
  return launch_split_reduce(out, part_out, part_ml, num_seqs, num_heads,
                             head_dim, n_splits, stream);
}

extern "C" hipError_t ks_paged_attention_decode(
    void* out, const void* q, const void* k_cache, const void* v_cache,
    const void* block_tables, const void* context_lens, float scale,
    int num_seqs, int num_heads, int num_kv_heads, int head_dim,
    int max_blocks, int block_size, long q_row_stride, int n_splits,
    void* part_out, void* part_ml, int window, hipStream_t stream) {
  if (block_size != PAGE) return hipErrorInvalidValue;
  const int group = num_heads / num_kv_heads;
  const int hpw = (group + NWAVES - 1) / NWAVES;  // q heads per wave
  if (hpw > 2) return hipErrorInvalidValue;  // groups up to 8 supported
  if (n_splits < 1) n_splits = 1;
  dim3 grid(num_kv_heads, num_seqs, n_splits);
  dim3 block(256);

  // wave-split variant (each page read by one wave). Measured SLOWER than
  // the head-split kernel on MI355X (profiles/attn_decode_bench.md):
  // the per-head serial chains outweigh the 4x L1-reuse saving. Kept as an
  // opt-in ablation (KS_ATTN_WS=1).
  static const bool use_ws = [] {
    const char* e = getenv("KS_ATTN_WS");
    return e != nullptr && e[0] == '1';
  }();
  const bool pow2_group =
      group == 1 || group == 2 || group == 4 || group == 8;
  if (use_ws && pow2_group && head_dim == 128) {
#define LAUNCH_WS(GG)                                                       \
  hipLaunchKernelGGL((paged_attention_ws_kernel<128, GG>), grid, block, 0,  \
                     stream, (short*)out, (const short*)q,                  \
                     (const short*)k_cache, (const short*)v_cache,          \
                     (const int*)block_tables, (const int*)context_lens,    \
                     scale, num_kv_heads, max_blocks, q_row_stride,         \
                     n_splits, (float*)part_out, (float*)part_ml, window)
    if (group == 1) LAUNCH_WS(1);
    else if (group == 2) LAUNCH_WS(2);
    else if (group == 4) LAUNCH_WS(4);
    else LAUNCH_WS(8);
#undef LAUNCH_WS
    HIP_CHECK_KERNEL();
    return launch_split_reduce(out, part_out, part_ml, num_seqs, num_heads,
                               head_dim, n_splits, stream);
  }

  // 2-wave x 2-head mode (A/B: KS_ATTN_H2=1): halves the per-CU L1
  // traffic from duplicated page reads at the cost of HPW=2 registers
  static const bool use_h2 = [] {
    const char* e = getenv("KS_ATTN_H2");
    return e != nullptr && e[0] == '1';
  }();
  if (use_h2 && group == 4 && head_dim == 128) {
    hipLaunchKernelGGL((paged_attention_kernel<128, 2>), grid, dim3(128), 0,
                       stream, (short*)out, (const short*)q,
                       (const short*)k_cache, (const short*)v_cache,
                       (const int*)block_tables, (const int*)context_lens,
                       scale, num_kv_heads, group, max_blocks, q_row_stride,
                       n_splits, (float*)part_out, (float*)part_ml, window);
    HIP_CHECK_KERNEL();
    return launch_split_reduce(out, part_out, part_ml, num_seqs, num_heads,
                               head_dim, n_splits, stream);
  }

  // MFMA-QK experiment (A/B: KS_ATTN_MQ=1): QK on the matrix pipe,
  // V4 wide-load PV
  static const bool use_mq = [] {
    const char* e = getenv("KS_ATTN_MQ");
    return e != nullptr && e[0] == '1';
  }();
  if (use_mq && hpw == 1 && head_dim == 128) {
    hipLaunchKernelGGL(
        (paged_attention_kernel<128, 1, 1, true, false, true, true>), grid,
        block, 0, stream, (short*)out, (const short*)q,
        (const short*)k_cache, (const short*)v_cache,
        (const int*)block_tables, (const int*)context_lens, scale,
        num_kv_heads, group, max_blocks, q_row_stride, n_splits,
        (float*)part_out, (float*)part_ml, window);
    HIP_CHECK_KERNEL();
    return launch_split_reduce(out, part_out, part_ml, num_seqs, num_heads,
                               head_dim, n_splits, stream);
  }

  // 2-page unrolled fast path (A/B: KS_ATTN_U2=0 disables)
  static const bool use_u2 = [] {  // measured slower; opt-in ablation
    const char* e = getenv("KS_ATTN_U2");
    return e != nullptr && e[0] == '1';
  }();
  if (use_u2 && hpw == 1 && head_dim == 128) {
    hipLaunchKernelGGL((paged_attention_u2_kernel<128>), grid, block, 0,
                       stream, (short*)out, (const short*)q,
                       (const short*)k_cache, (const short*)v_cache,
                       (const int*)block_tables, (const int*)context_lens,
                       scale, num_kv_heads, group, max_blocks, q_row_stride,
                       n_splits, (float*)part_out, (float*)part_ml, window);
    HIP_CHECK_KERNEL();
    return launch_split_reduce(out, part_out, part_ml, num_seqs, num_heads,
                               head_dim, n_splits, stream);
  }
  // occupancy ablation: KS_ATTN_OCC=5/6 forces a tighter VGPR budget so
  // more waves are resident to hide the scattered-page load latency
  static const int occ = [] {
    const char* e = getenv("KS_ATTN_OCC");
    return e ? atoi(e) : 0;
  }();
  // LDS p-broadcast is the measured default (+14-19% at large batch:
  // 3.34->3.80 TB/s @ S=512, 2.87->3.42 @ ctx=2048 — fewer DS ops per
  // page + 96 VGPRs/5 waves). KS_ATTN_PB=0 reverts to the shuffle path.
  static const bool use_pb = [] {
    const char* e = getenv("KS_ATTN_PB");
    return e == nullptr || e[0] != '0';
  }();
  // packed-bf16 dot wins in the latency regime (+10% at S=64, +5% at
  // S=8 split-context) and is noise-level at large batch where the PV
  // phase dominates; dispatch it for small batches. KS_ATTN_D2=1 forces
  // it everywhere, =0 disables.
  // wide-V-load variant wins the throughput regime (+9% at S=256/512,
  // +23% at long context) but drops occupancy (110 VGPR) and loses ~8%
  // in the latency regime — auto-dispatch by batch size.
  // KS_ATTN_V4=1 forces it, =0 disables.
  static const int v4_mode = [] {
    const char* e = getenv("KS_ATTN_V4");
    return e ? (e[0] == '1' ? 1 : 0) : -1;  // -1 = auto
  }();
  const bool use_v4 =
      v4_mode == 1 || (v4_mode == -1 && num_seqs > 128);
  static const bool v4_d2 = [] {  // A/B: packed-bf16 dot inside V4
    const char* e = getenv("KS_ATTN_V4D2");
    return e != nullptr && e[0] == '1';
  }();
  if (use_v4 && hpw == 1 && head_dim == 128) {
    if (v4_d2) {
      hipLaunchKernelGGL(
          (paged_attention_kernel<128, 1, 1, true, true, true>), grid, block,
          0, stream, (short*)out, (const short*)q, (const short*)k_cache,
          (const short*)v_cache, (const int*)block_tables,
          (const int*)context_lens, scale, num_kv_heads, group, max_blocks,
          q_row_stride, n_splits, (float*)part_out, (float*)part_ml, window);
    } else {
      hipLaunchKernelGGL(
          (paged_attention_kernel<128, 1, 1, true, false, true>), grid,
          block, 0, stream, (short*)out, (const short*)q,
          (const short*)k_cache, (const short*)v_cache,
          (const int*)block_tables, (const int*)context_lens, scale,
          num_kv_heads, group, max_blocks, q_row_stride, n_splits,
          (float*)part_out, (float*)part_ml, window);
    }
    HIP_CHECK_KERNEL();
    return launch_split_reduce(out, part_out, part_ml, num_seqs, num_heads,
                               head_dim, n_splits, stream);
  }
  static const int d2_mode = [] {
    const char* e = getenv("KS_ATTN_D2");
    return e ? (e[0] == '1' ? 1 : 0) : -1;  // -1 = auto
  }();
  const bool use_d2 =
      d2_mode == 1 || (d2_mode == -1 && num_seqs <= 128);
  if (use_d2 && hpw == 1 && head_dim == 128) {
    hipLaunchKernelGGL((paged_attention_kernel<128, 1, 1, true, true>), grid,
                       block, 0, stream, (short*)out, (const short*)q,
                       (const short*)k_cache, (const short*)v_cache,
                       (const int*)block_tables, (const int*)context_lens,
                       scale, num_kv_heads, group, max_blocks, q_row_stride,
                       n_splits, (float*)part_out, (float*)part_ml, window);
    HIP_CHECK_KERNEL();
    return launch_split_reduce(out, part_out, part_ml, num_seqs, num_heads,
                               head_dim, n_splits, stream);
  }
  if (use_pb && hpw == 1 && head_dim == 128) {
    hipLaunchKernelGGL((paged_attention_kernel<128, 1, 1, true>), grid, block,
                       0, stream, (short*)out, (const short*)q,
                       (const short*)k_cache, (const short*)v_cache,
                       (const int*)block_tables, (const int*)context_lens,
                       scale, num_kv_heads, group, max_blocks, q_row_stride,
                       n_splits, (float*)part_out, (float*)part_ml, window);
    HIP_CHECK_KERNEL();
    return launch_split_reduce(out, part_out, part_ml, num_seqs, num_heads,
                               head_dim, n_splits, stream);
  }
  if (occ >= 5 && hpw == 1 && head_dim == 128) {
    if (occ >= 6) {
      hipLaunchKernelGGL((paged_attention_kernel<128, 1, 6>), grid, block, 0,
                         stream, (short*)out, (const short*)q,
                         (const short*)k_cache, (const short*)v_cache,
                         (const int*)block_tables, (const int*)context_lens,
                         scale, num_kv_heads, group, max_blocks, q_row_stride,
                         n_splits, (float*)part_out, (float*)part_ml, window);
    } else {
      hipLaunchKernelGGL((paged_attention_kernel<128, 1, 5>), grid, block, 0,
                         stream, (short*)out, (const short*)q,
                         (const short*)k_cache, (const short*)v_cache,
                         (const int*)block_tables, (const int*)context_lens,
                         scale, num_kv_heads, group, max_blocks, q_row_stride,
                         n_splits, (float*)part_out, (float*)part_ml, window);
    }
    HIP_CHECK_KERNEL();
    return launch_split_reduce(out, part_out, part_ml, num_seqs, num_heads,
                               head_dim, n_splits, stream);
  }
#define LAUNCH_PA(DD, HH)                                                  \
  hipLaunchKernelGGL((paged_attention_kernel<DD, HH>), grid, block, 0,     \
                     stream, (short*)out, (const short*)q,                 \
                     (const short*)k_cache, (const short*)v_cache,         \
                     (const int*)block_tables, (const int*)context_lens,   \
                     scale, num_kv_heads, group, max_blocks, q_row_stride, \
                     n_splits, (float*)part_out, (float*)part_ml, window)
  if (head_dim == 128) {
    if (hpw == 1) LAUNCH_PA(128, 1);
    else LAUNCH_PA(128, 2);
  } else if (head_dim == 64) {
    if (hpw == 1) LAUNCH_PA(64, 1);
    else LAUNCH_PA(64, 2);
  } else {
    return hipErrorInvalidValue;
  }
#undef LAUNCH_PA
  HIP_CHECK_KERNEL();
  return launch_split_reduce(out, part_out, part_ml, num_seqs, num_heads,
                             head_dim, n_splits, stream);
}
