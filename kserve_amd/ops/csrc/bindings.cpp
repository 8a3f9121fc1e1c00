// Python bindings for the CDNA4 kernels (torch extension, ROCm-native:
// c10::hip stream APIs, no CUDA-compat shims).
#include <torch/extension.h>

#include <c10/hip/HIPStream.h>

#include <hip/hip_runtime.h>

// launchers from the .hip translation units
extern "C" {
hipError_t ks_rms_norm(void*, const void*, const void*, float, int, int,
                       hipStream_t);
hipError_t ks_fused_add_rms_norm(void*, void*, const void*, float, int, int,
                                 hipStream_t);
hipError_t ks_silu_and_mul(void*, const void*, int, int, hipStream_t);
hipError_t ks_rotary_embedding(void*, void*, const void*, const void*, int,
                               int, int, int, long, long, hipStream_t);
hipError_t ks_reshape_and_cache(const void*, const void*, void*, void*,
                                const void*, int, int, int, int, long, long,
                                hipStream_t);
hipError_t ks_paged_attention_decode(void*, const void*, const void*,
                                     const void*, const void*, const void*,
                                     float, int, int, int, int, int, int,
                                     long, int, void*, void*, int,
                                     hipStream_t);
hipError_t ks_flash_prefill_varlen(void*, const void*, const void*,
                                   const void*, const void*, int, int, int,
                                   int, int, float, long, long, long, int,
                                   int, hipStream_t);
hipError_t ks_context_prefill_varlen(void*, const void*, const void*,
                                     const void*, const void*, const void*,
                                     const void*, int, int, int, int, int,
                                     int, float, long, int, int,
                                     hipStream_t);
hipError_t ks_paged_attention_decode_fp8(void*, const void*, const void*,
                                         const void*, const void*,
                                         const void*, float, int, int, int,
                                         int, int, int, long, int, void*,
                                         void*, int, hipStream_t);
hipError_t ks_reshape_and_cache_fp8(const void*, const void*, void*, void*,
                                    const void*, int, int, int, int, long,
                                    long, hipStream_t);
hipError_t ks_layer_norm(void*, const void*, const void*, const void*, float,
                         int, int, hipStream_t);
hipError_t ks_fused_add_layer_norm(void*, const void*, const void*,
                                   const void*, const void*, float, int, int,
                                   hipStream_t);
hipError_t ks_gelu(void*, const void*, long, hipStream_t);
hipError_t ks_greedy_sample(void*, const void*, int, int, hipStream_t);
hipError_t ks_gumbel_sample(void*, const void*, const void*, const void*,
                            const void*, int, int, hipStream_t);
hipError_t ks_topk_topp_sample(void*, const void*, const void*, const void*,
                               const void*, const void*, int, int,
                               hipStream_t);
hipError_t ks_mfma_probe(void*, const void*, const void*, hipStream_t);
hipError_t ks_skinny_gemm(void*, void*, const void*, const void*, int, int,
                          int, long, hipStream_t);
hipError_t ks_gemm8(void*, const void*, const void*, int, int, int, int,
                    hipStream_t);
}

namespace {

hipStream_t current_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

void check_hip(hipError_t err, const char* op) {
  TORCH_CHECK(err == hipSuccess, op, " failed: ", hipGetErrorString(err));
}

#define CHECK_BF16_CONTIG(t)                                       \
  TORCH_CHECK((t).scalar_type() == at::kBFloat16, #t " must be bf16"); \
  TORCH_CHECK((t).is_contiguous(), #t " must be contiguous");      \
  TORCH_CHECK((t).is_cuda(), #t " must be on GPU")

// [T, heads, D] where rows may be strided (qkv split views) but the
// (head, dim) block of each row is dense
#define CHECK_KV_CACHE(t)                                               \
  TORCH_CHECK((t).scalar_type() == at::kBFloat16 ||                       \
                  (t).scalar_type() == at::kFloat8_e4m3fn,                \
              #t " must be bf16 or fp8_e4m3");                            \
  TORCH_CHECK((t).is_contiguous(), #t " must be contiguous");             \
  TORCH_CHECK((t).is_cuda(), #t " must be on GPU")

#define CHECK_BF16_ROWS(t)                                              \
  TORCH_CHECK((t).scalar_type() == at::kBFloat16, #t " must be bf16");  \
  TORCH_CHECK((t).is_cuda(), #t " must be on GPU");                     \
  TORCH_CHECK((t).dim() == 3 && (t).stride(2) == 1 &&                   \
                  (t).stride(1) == (t).size(2),                         \
              #t " inner dims must be dense")

void rms_norm(at::Tensor& out, at::Tensor& input, at::Tensor& weight,
              double eps) {
  CHECK_BF16_CONTIG(out);
  CHECK_BF16_CONTIG(input);
  CHECK_BF16_CONTIG(weight);
  int hidden = input.size(-1);
  int rows = input.numel() / hidden;
  check_hip(ks_rms_norm(out.data_ptr(), input.data_ptr(), weight.data_ptr(),
                        (float)eps, rows, hidden, current_stream()),
            "rms_norm");
}

void fused_add_rms_norm(at::Tensor& x, at::Tensor& residual, at::Tensor& weight,
                        double eps) {
  CHECK_BF16_CONTIG(x);
  CHECK_BF16_CONTIG(residual);
  int hidden = x.size(-1);
  int rows = x.numel() / hidden;
  check_hip(
      ks_fused_add_rms_norm(x.data_ptr(), residual.data_ptr(),
                            weight.data_ptr(), (float)eps, rows, hidden,
                            current_stream()),
      "fused_add_rms_norm");
}

void silu_and_mul(at::Tensor& out, at::Tensor& input) {
  CHECK_BF16_CONTIG(out);
  CHECK_BF16_CONTIG(input);
  int d = out.size(-1);
  TORCH_CHECK(input.size(-1) == 2 * d, "input last dim must be 2*out");
  int rows = out.numel() / d;
  check_hip(ks_silu_and_mul(out.data_ptr(), input.data_ptr(), rows, d,
                            current_stream()),
            "silu_and_mul");
}

void rotary_embedding(at::Tensor& positions, at::Tensor& q, at::Tensor& k,
                      at::Tensor& cos_sin_cache) {
  CHECK_BF16_ROWS(q);
  CHECK_BF16_ROWS(k);
  TORCH_CHECK(positions.scalar_type() == at::kLong, "positions must be int64");
  TORCH_CHECK(cos_sin_cache.scalar_type() == at::kFloat,
              "cos_sin_cache must be fp32");
  int T = q.size(0);
  int Hq = q.size(1);
  int Hk = k.size(1);
  int D = q.size(2);
  check_hip(ks_rotary_embedding(q.data_ptr(), k.data_ptr(),
                                positions.data_ptr(), cos_sin_cache.data_ptr(),
                                T, Hq, Hk, D, (long)q.stride(0),
                                (long)k.stride(0), current_stream()),
            "rotary_embedding");
}

void reshape_and_cache(at::Tensor& k, at::Tensor& v, at::Tensor& k_cache,
                       at::Tensor& v_cache, at::Tensor& slot_mapping) {
  CHECK_BF16_ROWS(k);
  CHECK_BF16_ROWS(v);
  CHECK_KV_CACHE(k_cache);
  CHECK_KV_CACHE(v_cache);
  TORCH_CHECK(slot_mapping.scalar_type() == at::kInt, "slot_mapping int32");
  int T = k.size(0);
  int Hkv = k.size(1);
  int D = k.size(2);
  int block_size = k_cache.size(2);
  if (k_cache.scalar_type() == at::kFloat8_e4m3fn) {
    check_hip(ks_reshape_and_cache_fp8(k.data_ptr(), v.data_ptr(),
                                       k_cache.data_ptr(), v_cache.data_ptr(),
                                       slot_mapping.data_ptr(), T, Hkv, D,
                                       block_size, (long)k.stride(0),
                                       (long)v.stride(0), current_stream()),
              "reshape_and_cache_fp8");
    return;
  }
  check_hip(ks_reshape_and_cache(k.data_ptr(), v.data_ptr(),
                                 k_cache.data_ptr(), v_cache.data_ptr(),
                                 slot_mapping.data_ptr(), T, Hkv, D,
                                 block_size, (long)k.stride(0),
                                 (long)v.stride(0), current_stream()),
            "reshape_and_cache");
}

void paged_attention_decode(at::Tensor& out, at::Tensor& q,
                            at::Tensor& k_cache, at::Tensor& v_cache,
                            at::Tensor& block_tables, at::Tensor& context_lens,
                            double scale, int64_t window) {
  CHECK_BF16_CONTIG(out);
  CHECK_BF16_ROWS(q);
  CHECK_KV_CACHE(k_cache);
  CHECK_KV_CACHE(v_cache);
  TORCH_CHECK(block_tables.scalar_type() == at::kInt, "block_tables int32");
  TORCH_CHECK(context_lens.scalar_type() == at::kInt, "context_lens int32");
  const bool fp8 = k_cache.scalar_type() == at::kFloat8_e4m3fn;
  int S = q.size(0);
  int H = q.size(1);
  int D = q.size(2);
  int Hkv = k_cache.size(1);
  int block_size = k_cache.size(2);
  int max_blocks = block_tables.size(1);
  // split-context heuristic, measured A/B 2026-09-12 (gpurun_out/
  // attn_split_*.log): splitting HURTS at S>=256 (3.69->3.43 TB/s, the
  // fp32 partials + reduce outweigh extra waves) and for the group=8
  // HPW=2 shapes (2.05->1.81), but the 512-workgroup group<=4 regime
  // gains (S=64: 3.75->3.96 at 4 splits) and tiny batches need deep
  // splits (S=8: 0.71 unsplit vs 1.6-1.7 at 8-16). Rule: group<=4 AND
  // base<=512 -> clamp(2048/base, 2, 8); otherwise split only below
  // 512 workgroups (the round-1 rule).
  int n_splits = 1;
  long base_wgs = (long)S * Hkv;
  const int grp = H / Hkv;
  static const int split_override = [] {
    const char* e = getenv("KS_ATTN_SPLIT");
    return e ? atoi(e) : 0;  // 0 = heuristic
  }();
  if (split_override > 0) {
    n_splits = split_override;
  } else if (grp <= 4 && base_wgs <= 512) {
    n_splits = (int)(2048 / base_wgs);
    if (n_splits < 2) n_splits = 2;
    if (n_splits > 8) n_splits = 8;
  } else if (base_wgs < 512) {
    n_splits = (int)((512 + base_wgs - 1) / base_wgs);
    if (n_splits > 16) n_splits = 16;
  }
  if (n_splits > 1) {
    int max_split_blocks = (max_blocks + n_splits - 1) / n_splits;
    if (max_split_blocks < 1) n_splits = 1;
  }
  at::Tensor part_out, part_ml;
  void *po = nullptr, *pml = nullptr;
  if (n_splits > 1) {
    auto opts = at::TensorOptions().dtype(at::kFloat).device(q.device());
    part_out = at::empty({S, H, n_splits, D}, opts);
    part_ml = at::empty({S, H, n_splits, 2}, opts);
    po = part_out.data_ptr();
    pml = part_ml.data_ptr();
  }
  if (fp8) {
    TORCH_CHECK(H / Hkv <= 4 && D == 128,
                "fp8 KV decode supports D=128 and GQA group <= 4");
    check_hip(ks_paged_attention_decode_fp8(
                  out.data_ptr(), q.data_ptr(), k_cache.data_ptr(),
                  v_cache.data_ptr(), block_tables.data_ptr(),
                  context_lens.data_ptr(), (float)scale, S, H, Hkv, D,
                  max_blocks, block_size, (long)q.stride(0), n_splits, po,
                  pml, (int)window, current_stream()),
              "paged_attention_decode_fp8");
    return;
  }
  check_hip(ks_paged_attention_decode(
                out.data_ptr(), q.data_ptr(), k_cache.data_ptr(),
                v_cache.data_ptr(), block_tables.data_ptr(),
                context_lens.data_ptr(), (float)scale, S, H, Hkv, D,
                max_blocks, block_size, (long)q.stride(0), n_splits, po, pml,
                (int)window, current_stream()),
            "paged_attention_decode");
}

void flash_prefill_varlen(at::Tensor& out, at::Tensor& q, at::Tensor& k,
                          at::Tensor& v, at::Tensor& cu_seqlens,
                          int64_t max_seqlen, double scale,
                          bool causal = true, int64_t window = 0) {
  CHECK_BF16_CONTIG(out);
  CHECK_BF16_ROWS(q);
  CHECK_BF16_ROWS(k);
  CHECK_BF16_ROWS(v);
  TORCH_CHECK(cu_seqlens.scalar_type() == at::kInt, "cu_seqlens int32");
  int num_seqs = cu_seqlens.size(0) - 1;
  int Hq = q.size(1);
  int Hkv = k.size(1);
  int D = q.size(2);
  check_hip(ks_flash_prefill_varlen(out.data_ptr(), q.data_ptr(), k.data_ptr(),
                                    v.data_ptr(), cu_seqlens.data_ptr(),
                                    num_seqs, (int)max_seqlen, Hq, Hkv, D,
                                    (float)scale, (long)q.stride(0),
                                    (long)k.stride(0), (long)v.stride(0),
                                    causal ? 1 : 0, (int)window,
                                    current_stream()),
            "flash_prefill_varlen");
}

void context_prefill_varlen(at::Tensor& out, at::Tensor& q,
                            at::Tensor& k_cache, at::Tensor& v_cache,
                            at::Tensor& block_tables, at::Tensor& ctx_lens,
                            at::Tensor& cu_seqlens_q, int64_t max_q_len,
                            double scale, int64_t window) {
  CHECK_BF16_CONTIG(out);
  CHECK_BF16_ROWS(q);
  CHECK_KV_CACHE(k_cache);
  CHECK_KV_CACHE(v_cache);
  TORCH_CHECK(block_tables.scalar_type() == at::kInt, "block_tables int32");
  TORCH_CHECK(block_tables.is_contiguous(), "block_tables contiguous");
  TORCH_CHECK(ctx_lens.scalar_type() == at::kInt, "ctx_lens int32");
  TORCH_CHECK(cu_seqlens_q.scalar_type() == at::kInt, "cu_seqlens_q int32");
  TORCH_CHECK(k_cache.size(2) == 16, "block_size must be 16");
  int num_seqs = cu_seqlens_q.size(0) - 1;
  int Hq = q.size(1);
  int Hkv = k_cache.size(1);
  int D = q.size(2);
  int fp8 = k_cache.scalar_type() == at::kFloat8_e4m3fn ? 1 : 0;
  check_hip(ks_context_prefill_varlen(
                out.data_ptr(), q.data_ptr(), k_cache.data_ptr(),
                v_cache.data_ptr(), block_tables.data_ptr(),
                ctx_lens.data_ptr(), cu_seqlens_q.data_ptr(), num_seqs,
                (int)max_q_len, Hq, Hkv, D, (int)block_tables.size(1),
                (float)scale, (long)q.stride(0), fp8, (int)window,
                current_stream()),
            "context_prefill_varlen");
}

void layer_norm(at::Tensor& out, at::Tensor& input, at::Tensor& weight,
                at::Tensor& bias, double eps) {
  CHECK_BF16_CONTIG(out);
  CHECK_BF16_CONTIG(input);
  int hidden = input.size(-1);
  int rows = input.numel() / hidden;
  check_hip(ks_layer_norm(out.data_ptr(), input.data_ptr(), weight.data_ptr(),
                          bias.data_ptr(), (float)eps, rows, hidden,
                          current_stream()),
            "layer_norm");
}

void fused_add_layer_norm(at::Tensor& out, at::Tensor& input,
                          at::Tensor& residual, at::Tensor& weight,
                          at::Tensor& bias, double eps) {
  CHECK_BF16_CONTIG(out);
  CHECK_BF16_CONTIG(input);
  CHECK_BF16_CONTIG(residual);
  int hidden = input.size(-1);
  int rows = input.numel() / hidden;
  check_hip(ks_fused_add_layer_norm(out.data_ptr(), input.data_ptr(),
                                    residual.data_ptr(), weight.data_ptr(),
                                    bias.data_ptr(), (float)eps, rows, hidden,
                                    current_stream()),
            "fused_add_layer_norm");
}

void gelu(at::Tensor& out, at::Tensor& input) {
  CHECK_BF16_CONTIG(out);
  CHECK_BF16_CONTIG(input);
  check_hip(ks_gelu(out.data_ptr(), input.data_ptr(), input.numel(),
                    current_stream()),
            "gelu");
}

void greedy_sample(at::Tensor& out, at::Tensor& logits) {
  CHECK_BF16_CONTIG(logits);
  TORCH_CHECK(out.scalar_type() == at::kLong, "out int64");
  check_hip(ks_greedy_sample(out.data_ptr(), logits.data_ptr(),
                             logits.size(0), logits.size(1),
                             current_stream()),
            "greedy_sample");
}

void gumbel_sample(at::Tensor& out, at::Tensor& logits,
                   at::Tensor& temperatures, at::Tensor& top_k,
                   at::Tensor& seeds) {
  CHECK_BF16_CONTIG(logits);
  TORCH_CHECK(out.scalar_type() == at::kLong, "out int64");
  TORCH_CHECK(temperatures.scalar_type() == at::kFloat, "temps fp32");
  TORCH_CHECK(seeds.scalar_type() == at::kLong, "seeds int64");
  check_hip(ks_gumbel_sample(out.data_ptr(), logits.data_ptr(),
                             temperatures.data_ptr(), top_k.data_ptr(),
                             seeds.data_ptr(), logits.size(0), logits.size(1),
                             current_stream()),
            "gumbel_sample");
}

void topk_topp_sample(at::Tensor& out, at::Tensor& logits,
                      at::Tensor& temperatures, at::Tensor& top_p,
                      at::Tensor& top_k, at::Tensor& seeds) {
  CHECK_BF16_CONTIG(logits);
  TORCH_CHECK(out.scalar_type() == at::kLong, "out int64");
  TORCH_CHECK(temperatures.scalar_type() == at::kFloat, "temps fp32");
  TORCH_CHECK(top_p.scalar_type() == at::kFloat, "top_p fp32");
  TORCH_CHECK(top_k.scalar_type() == at::kInt, "top_k int32");
  TORCH_CHECK(seeds.scalar_type() == at::kLong, "seeds int64");
  check_hip(ks_topk_topp_sample(out.data_ptr(), logits.data_ptr(),
                                temperatures.data_ptr(), top_p.data_ptr(),
                                top_k.data_ptr(), seeds.data_ptr(),
                                logits.size(0), logits.size(1),
                                current_stream()),
            "topk_topp_sample");
}

void skinny_gemm(at::Tensor& out, at::Tensor& x, at::Tensor& w) {
  // out [N, M] bf16 = x [N, K] @ w [M, K]^T  (decode shapes, N <= 256)
  CHECK_BF16_CONTIG(out);
  CHECK_BF16_CONTIG(w);
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 && x.is_cuda() &&
                  x.stride(1) == 1,
              "x must be bf16 row-dense");
  int N = x.size(0);
  int K = x.size(1);
  int M = w.size(0);
  TORCH_CHECK(w.size(1) == K, "shape mismatch");
  at::Tensor ws;
  void* wsp = nullptr;
  // mirror the kernel's K-split decision to size the partials workspace
  int nsplit = 1;
  int feat_wgs = M / 64;
  while (feat_wgs * nsplit < 768 && nsplit < 8 && (K / (nsplit * 2)) >= 64)
    nsplit *= 2;
  if (nsplit > 1) {
    // bf16 partials (each a full-precision-accumulated K/nsplit dot)
    ws = at::empty({(long)nsplit * N * M},
                   at::TensorOptions().dtype(at::kBFloat16).device(x.device()));
    wsp = ws.data_ptr();
  }
  check_hip(ks_skinny_gemm(out.data_ptr(), wsp, x.data_ptr(), w.data_ptr(),
                           M, K, N, (long)x.stride(0), current_stream()),
            "skinny_gemm");
}

void gemm8(at::Tensor& d, at::Tensor& a, at::Tensor& w, int64_t swizzle) {
  // EXPERIMENTAL (see gemm8.hip header): not used by ops.linear dispatch
  CHECK_BF16_CONTIG(d);
  CHECK_BF16_CONTIG(a);
  CHECK_BF16_CONTIG(w);
  int M = a.size(0), K = a.size(1), N = w.size(0);
  TORCH_CHECK(w.size(1) == K && d.size(0) == M && d.size(1) == N,
              "shape mismatch");
  check_hip(ks_gemm8(d.data_ptr(), a.data_ptr(), w.data_ptr(), M, N, K,
                     (int)swizzle, current_stream()),
            "gemm8");
}

void mfma_probe(at::Tensor& c, at::Tensor& a, at::Tensor& b) {
  check_hip(ks_mfma_probe(c.data_ptr(), a.data_ptr(), b.data_ptr(),
                          current_stream()),
            "mfma_probe");
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rms_norm", &rms_norm, "RMSNorm (bf16, CDNA4)");
  m.def("fused_add_rms_norm", &fused_add_rms_norm,
        "fused residual-add + RMSNorm, in-place");
  m.def("silu_and_mul", &silu_and_mul, "SwiGLU activation");
  m.def("rotary_embedding", &rotary_embedding, "neox RoPE in-place");
  m.def("reshape_and_cache", &reshape_and_cache, "KV page scatter");
  m.def("paged_attention_decode", &paged_attention_decode,
        "paged decode attention (GQA, split-context)");
  m.def("flash_prefill_varlen", &flash_prefill_varlen,
        "MFMA flash attention (varlen GQA, causal or bidirectional)",
        pybind11::arg("out"), pybind11::arg("q"), pybind11::arg("k"),
        pybind11::arg("v"), pybind11::arg("cu_seqlens"),
        pybind11::arg("max_seqlen"), pybind11::arg("scale"),
        pybind11::arg("causal") = true, pybind11::arg("window") = 0);
  m.def("context_prefill_varlen", &context_prefill_varlen,
        "MFMA prefill attention against the paged KV cache (chunked prefill)");
  m.def("layer_norm", &layer_norm, "LayerNorm (bf16)");
  m.def("fused_add_layer_norm", &fused_add_layer_norm,
        "residual-add + LayerNorm");
  m.def("gelu", &gelu, "erf GELU");
  m.def("greedy_sample", &greedy_sample, "argmax sampling");
  m.def("gumbel_sample", &gumbel_sample, "Gumbel-max temperature sampling");
  m.def("topk_topp_sample", &topk_topp_sample,
        "fused top-k/top-p + Gumbel-max sampler (radix-histogram select)");
  m.def("skinny_gemm", &skinny_gemm, "decode GEMM (N<=256, MFMA streaming)");
  m.def("gemm8", &gemm8,
        "EXPERIMENTAL 8-phase 256x256 MFMA GEMM (round-2 candidate)");
  m.def("mfma_probe", &mfma_probe, "MFMA layout probe (tests)");
}
