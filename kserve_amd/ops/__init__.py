"""Custom op dispatch: CDNA4 HIP kernels on GPU, torch reference on CPU.

Policy (round-end harness contract): GPU tensors MUST run the native
extension — if ``kserve_amd_C`` is missing on a CUDA/ROCm device the op
raises ``NoNativeExtension`` instead of silently falling back to eager
torch. CPU tensors use the fp32 torch reference (tests, engine logic).
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

from kserve_amd.errors import NoNativeExtension
from kserve_amd.ops import torch_ref

_C = None
_IMPORT_ERROR: Optional[str] = None
try:
    import kserve_amd_C as _C  # built in-tree by setup.py (travels to GPU box)
except ImportError as e:  # pragma: no cover - exercised on GPU box
    _IMPORT_ERROR = str(e)


def has_native() -> bool:
    return _C is not None


def _native(op: str):
    if _C is None:
        raise NoNativeExtension(op, f"(import error: {_IMPORT_ERROR})")
    return _C


def rms_norm(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    if x.is_cuda:
        out = torch.empty_like(x)
        _native("rms_norm").rms_norm(out, x, weight, eps)
        return out
    return torch_ref.rms_norm(x, weight, eps)


def fused_add_rms_norm(
    x: torch.Tensor, residual: torch.Tensor, weight: torch.Tensor, eps: float
) -> Tuple[torch.Tensor, torch.Tensor]:
    """In-place on GPU: x <- rmsnorm(x+residual), residual <- x+residual."""
    if x.is_cuda:
        _native("fused_add_rms_norm").fused_add_rms_norm(x, residual, weight, eps)
        return x, residual
    return torch_ref.fused_add_rms_norm(x, residual, weight, eps)


def silu_and_mul(x: torch.Tensor) -> torch.Tensor:
    if x.is_cuda:
        d = x.shape[-1] // 2
        out = torch.empty(
            (*x.shape[:-1], d), dtype=x.dtype, device=x.device
        )
        _native("silu_and_mul").silu_and_mul(out, x)
        return out
    return torch_ref.silu_and_mul(x)


def rotary_embedding(
    positions: torch.Tensor,
    q: torch.Tensor,
    k: torch.Tensor,
    cos_sin_cache: torch.Tensor,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Neox-style RoPE. GPU path mutates q/k in place and returns them."""
    if q.is_cuda:
        _native("rotary_embedding").rotary_embedding(positions, q, k, cos_sin_cache)
        return q, k
    return torch_ref.rotary_embedding(positions, q, k, cos_sin_cache)


def reshape_and_cache(
    k: torch.Tensor,
    v: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    slot_mapping: torch.Tensor,
) -> None:
    if k.is_cuda:
        _native("reshape_and_cache").reshape_and_cache(
            k, v, k_cache, v_cache, slot_mapping
        )
        return
    torch_ref.reshape_and_cache(k, v, k_cache, v_cache, slot_mapping)


def paged_attention_decode(
    q: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,
    context_lens: torch.Tensor,
    scale: float,
    out: Optional[torch.Tensor] = None,
    window: int = 0,
) -> torch.Tensor:
    if q.is_cuda:
        if out is None:
            out = torch.empty_like(q)
        _native("paged_attention_decode").paged_attention_decode(
            out, q, k_cache, v_cache, block_tables, context_lens, scale,
            window,
        )
        return out
    return torch_ref.paged_attention_decode(
        q, k_cache, v_cache, block_tables, context_lens, scale,
        window=window,
    )


def flash_prefill_varlen(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    cu_seqlens: torch.Tensor,
    max_seqlen: int,
    scale: float,
    window: int = 0,
) -> torch.Tensor:
    if q.is_cuda:
        out = torch.empty_like(q)
        _native("flash_prefill").flash_prefill_varlen(
            out, q, k, v, cu_seqlens, int(max_seqlen), scale,
            causal=True, window=window,
        )
        return out
    return torch_ref.flash_prefill_varlen(
        q, k, v, cu_seqlens, scale, causal=True, window=window
    )


def context_attention_varlen(
    q: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,
    cu_seqlens_q: torch.Tensor,
    context_lens: torch.Tensor,
    max_q_len: int,
    scale: float,
    window: int = 0,
) -> torch.Tensor:
    """Prefill attention against the paged cache (chunked prefill): the
    chunk's KV must already be written to the cache; queries attend causally
    to the whole per-seq context (bounded below by the sliding window when
    window > 0)."""
    if q.is_cuda:
        out = torch.empty_like(q)
        _native("context_prefill").context_prefill_varlen(
            out, q, k_cache, v_cache, block_tables, context_lens,
            cu_seqlens_q, int(max_q_len), scale, window
        )
        return out
    return torch_ref.context_attention_varlen(
        q, k_cache, v_cache, block_tables, cu_seqlens_q, context_lens, scale,
        window=window,
    )


def greedy_sample(logits: torch.Tensor) -> torch.Tensor:
    if logits.is_cuda:
        out = torch.empty(
            logits.shape[0], dtype=torch.int64, device=logits.device
        )
        _native("greedy_sample").greedy_sample(out, logits)
        return out
    return torch_ref.greedy_sample(logits)


def random_sample(
    logits: torch.Tensor,
    temperatures: torch.Tensor,
    top_p: torch.Tensor,
    top_k: torch.Tensor,
    seeds: Optional[torch.Tensor] = None,
    generator: Optional[torch.Generator] = None,
    min_p: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """Temperature/top-k/top-p/min-p sampling.

    GPU: one fused HIP kernel. Plain temperature sampling is Gumbel-max;
    top-k/top-p use a bf16 radix-histogram select (exact k-th value / mass
    threshold, no sort) followed by Gumbel-max over the surviving set.
    min_p (rare) routes to the fp32 sort path.
    """
    has_min_p = min_p is not None and bool((min_p > 0).any())
    if logits.is_cuda and seeds is not None and not has_min_p:
        out = torch.empty(
            logits.shape[0], dtype=torch.int64, device=logits.device
        )
        needs_trunc = bool((top_p < 1.0).any()) or bool((top_k > 0).any())
        if needs_trunc:
            _native("random_sample").topk_topp_sample(
                out, logits, temperatures, top_p, top_k.to(torch.int32), seeds
            )
        else:
            _native("random_sample").gumbel_sample(
                out, logits, temperatures, top_k.to(torch.int32), seeds
            )
        return out
    # min-p (or CPU): fp32 sort path
    return torch_ref.random_sample(
        logits, temperatures, top_p, top_k, generator=generator, min_p=min_p
    )


def linear(x: torch.Tensor, weight: torch.Tensor, bias: Optional[torch.Tensor] = None) -> torch.Tensor:
    """GEMM dispatch: custom MFMA skinny-GEMM for decode-sized batches
    (weight-streaming regime where hipBLASLt underperforms ~4x on gfx950 —
    see tools/gemm_bench.py), hipBLASLt via F.linear otherwise."""
    import torch.nn.functional as F

    # measured crossover (tools/gemm_bench.py on MI355X): the custom kernel
    # beats hipBLASLt only for small token batches on qkv/o-sized shapes;
    # hipBLASLt keeps the wide-M / large-K / large-N shapes.
    if (
        _C is not None
        and x.is_cuda
        and bias is None
        and x.dim() == 2
        and 0 < x.shape[0] <= 64
        and weight.shape[0] % 64 == 0
        and weight.shape[0] <= 8192
        and weight.shape[1] % 64 == 0
        and weight.shape[1] <= 8192
        and x.stride(1) == 1
    ):
        out = torch.empty(
            (x.shape[0], weight.shape[0]), dtype=x.dtype, device=x.device
        )
        _C.skinny_gemm(out, x, weight)
        return out
    return F.linear(x, weight, bias)


def layer_norm(x, weight, bias, eps: float):
    if x.is_cuda:
        out = torch.empty_like(x)
        _native("layer_norm").layer_norm(out, x, weight, bias, eps)
        return out
    return torch_ref.layer_norm(x, weight, bias, eps)


def fused_add_layer_norm(x, residual, weight, bias, eps: float):
    if x.is_cuda:
        out = torch.empty_like(x)
        _native("fused_add_layer_norm").fused_add_layer_norm(
            out, x, residual, weight, bias, eps
        )
        return out
    return torch_ref.fused_add_layer_norm(x, residual, weight, bias, eps)


def gelu(x):
    if x.is_cuda:
        out = torch.empty_like(x)
        _native("gelu").gelu(out, x)
        return out
    return torch_ref.gelu(x)


def flash_attn_varlen(q, k, v, cu_seqlens, max_seqlen, scale, causal=True):
    """Bidirectional-capable variant (encoder models)."""
    if q.is_cuda:
        out = torch.empty_like(q)
        _native("flash_prefill").flash_prefill_varlen(
            out, q, k, v, cu_seqlens, int(max_seqlen), scale, causal
        )
        return out
    return torch_ref.flash_prefill_varlen(q, k, v, cu_seqlens, scale, causal=causal)


def topk_topp_sample_into(
    out: torch.Tensor,
    logits: torch.Tensor,
    temperatures: torch.Tensor,
    top_p: torch.Tensor,
    top_k: torch.Tensor,
    seeds: torch.Tensor,
) -> torch.Tensor:
    """Destination-passing form of the fused sampler for hipGraph capture
    (static out buffer, per-row device seeds; see model_runner
    _ensure_sampled_graph). GPU-only."""
    _native("random_sample").topk_topp_sample(
        out, logits, temperatures, top_p,
        top_k if top_k.dtype == torch.int32 else top_k.to(torch.int32),
        seeds,
    )
    return out
