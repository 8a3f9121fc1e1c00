"""Plain-PyTorch reference implementations of every custom op.

These are the numerics oracle for the HIP kernels (GPU tests compare the
CDNA4 kernels against these run in fp32) and the CPU execution path for
engine logic tests. They are NOT used on GPU tensors in production — the
dispatcher fails loudly instead (errors.NoNativeExtension).
"""

from __future__ import annotations

import math
from typing import Optional, Tuple

import torch


def rms_norm(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    dtype = x.dtype
    xf = x.float()
    var = xf.pow(2).mean(dim=-1, keepdim=True)
    out = xf * torch.rsqrt(var + eps)
    return (out * weight.float()).to(dtype)


def fused_add_rms_norm(
    x: torch.Tensor, residual: torch.Tensor, weight: torch.Tensor, eps: float
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Returns (normed, new_residual) where new_residual = x + residual."""
    new_residual = (x.float() + residual.float()).to(x.dtype)
    return rms_norm(new_residual, weight, eps), new_residual


def silu_and_mul(x: torch.Tensor) -> torch.Tensor:
    """x: [..., 2*d] (gate | up) -> silu(gate) * up."""
    d = x.shape[-1] // 2
    gate, up = x[..., :d], x[..., d:]
    return (torch.nn.functional.silu(gate.float()) * up.float()).to(x.dtype)


def _llama3_scale_inv_freq(inv_freq, rope_scaling: dict):
    """Llama-3.1 rope scaling (HF 'rope_type: llama3'): low-frequency
    components are slowed by `factor`, high-frequency ones kept, with a
    smooth interpolation between the two wavelength thresholds."""
    import math

    factor = rope_scaling.get("factor", 8.0)
    lo = rope_scaling.get("low_freq_factor", 1.0)
    hi = rope_scaling.get("high_freq_factor", 4.0)
    orig = rope_scaling.get("original_max_position_embeddings", 8192)
    low_wl = orig / lo
    high_wl = orig / hi
    out = inv_freq.clone()
    for i, f in enumerate(inv_freq.tolist()):
        wl = 2 * math.pi / f
        if wl < high_wl:
            continue  # high frequency: unscaled
        if wl > low_wl:
            out[i] = f / factor  # low frequency: fully scaled
        else:
            smooth = (orig / wl - lo) / (hi - lo)
            out[i] = (1 - smooth) * f / factor + smooth * f
    return out


def make_cos_sin_cache(
    head_dim: int,
    max_positions: int,
    theta: float = 10000.0,
    dtype: torch.dtype = torch.float32,
    device="cpu",
    rope_scaling: Optional[dict] = None,
) -> torch.Tensor:
    """[max_positions, head_dim] with cos in the first half, sin in the second
    (host-precomputed per CDNA guide Appendix B: no on-device trig).
    rope_scaling: HF dict; 'llama3' rope_type applies 3.1-style frequency
    scaling."""
    inv_freq = 1.0 / (
        theta ** (torch.arange(0, head_dim, 2, dtype=torch.float64) / head_dim)
    )
    if rope_scaling:
        rtype = rope_scaling.get("rope_type", rope_scaling.get("type"))
        if rtype == "llama3":
            inv_freq = _llama3_scale_inv_freq(inv_freq, rope_scaling)
        elif rtype == "linear":
            inv_freq = inv_freq / rope_scaling.get("factor", 1.0)
    t = torch.arange(max_positions, dtype=torch.float64)
    freqs = torch.outer(t, inv_freq)  # [P, head_dim/2]
    cache = torch.cat([freqs.cos(), freqs.sin()], dim=-1).to(dtype).to(device)
    return cache


def rotary_embedding(
    positions: torch.Tensor,  # [num_tokens]
    q: torch.Tensor,  # [num_tokens, num_heads, head_dim]
    k: torch.Tensor,  # [num_tokens, num_kv_heads, head_dim]
    cos_sin_cache: torch.Tensor,  # [max_pos, head_dim]
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Neox-style (rotate half) RoPE, out-of-place reference."""
    head_dim = q.shape[-1]
    half = head_dim // 2
    cs = cos_sin_cache[positions].float()  # [T, head_dim]
    cos = cs[:, :half].unsqueeze(1)  # [T, 1, half]
    sin = cs[:, half:].unsqueeze(1)

    def rot(x):
        xf = x.float()
        x1, x2 = xf[..., :half], xf[..., half:]
        return torch.cat([x1 * cos - x2 * sin, x2 * cos + x1 * sin], dim=-1).to(x.dtype)

    return rot(q), rot(k)


def reshape_and_cache(
    k: torch.Tensor,  # [num_tokens, num_kv_heads, head_dim]
    v: torch.Tensor,
    k_cache: torch.Tensor,  # [num_blocks, num_kv_heads, block_size, head_dim]
    v_cache: torch.Tensor,
    slot_mapping: torch.Tensor,  # [num_tokens] int64/int32
) -> None:
    block_size = k_cache.shape[2]
    slots = slot_mapping.long()
    blk = slots // block_size
    off = slots % block_size
    # advanced indexing: k_cache[blk[i], h, off[i], :] = k[i, h, :]
    # (fp8 caches need an explicit cast: index_put refuses mixed dtypes)
    k_cache[blk, :, off] = k.to(k_cache.dtype)
    v_cache[blk, :, off] = v.to(v_cache.dtype)


def paged_attention_decode(
    q: torch.Tensor,  # [num_seqs, num_heads, head_dim]
    k_cache: torch.Tensor,  # [num_blocks, num_kv_heads, block_size, head_dim]
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,  # [num_seqs, max_blocks] int32 (padded)
    context_lens: torch.Tensor,  # [num_seqs] int32 — includes current token
    scale: float,
    window: int = 0,
) -> torch.Tensor:
    num_seqs, num_heads, head_dim = q.shape
    num_kv_heads = k_cache.shape[1]
    block_size = k_cache.shape[2]
    group = num_heads // num_kv_heads
    out = torch.empty_like(q)
    for s in range(num_seqs):
        ctx = int(context_lens[s])
        nb = -(-ctx // block_size)
        blocks = block_tables[s, :nb].long()
        # [nb, H_kv, B, D] -> [H_kv, nb*B, D]
        keys = k_cache[blocks].permute(1, 0, 2, 3).reshape(num_kv_heads, -1, head_dim)[
            :, :ctx
        ]
        vals = v_cache[blocks].permute(1, 0, 2, 3).reshape(num_kv_heads, -1, head_dim)[
            :, :ctx
        ]
        qh = q[s].float()  # [H, D]
        scores = torch.einsum(
            "hd,htd->ht", qh, keys.float().repeat_interleave(group, dim=0)
        ) * scale
        if window > 0 and ctx > window:
            scores[:, : ctx - window] = float("-inf")
        probs = torch.softmax(scores, dim=-1)
        o = torch.einsum(
            "ht,htd->hd", probs, vals.float().repeat_interleave(group, dim=0)
        )
        out[s] = o.to(q.dtype)
    return out


def flash_prefill_varlen(
    q: torch.Tensor,  # [total_tokens, num_heads, head_dim]
    k: torch.Tensor,  # [total_tokens, num_kv_heads, head_dim]
    v: torch.Tensor,
    cu_seqlens: torch.Tensor,  # [num_seqs+1] int32
    scale: float,
    causal: bool = True,
    window: int = 0,
) -> torch.Tensor:
    num_heads = q.shape[1]
    num_kv_heads = k.shape[1]
    group = num_heads // num_kv_heads
    out = torch.empty_like(q)
    for s in range(cu_seqlens.numel() - 1):
        a, b = int(cu_seqlens[s]), int(cu_seqlens[s + 1])
        qs = q[a:b].float().transpose(0, 1)  # [H, T, D]
        ks = k[a:b].float().transpose(0, 1).repeat_interleave(group, dim=0)
        vs = v[a:b].float().transpose(0, 1).repeat_interleave(group, dim=0)
        scores = torch.einsum("htd,hsd->hts", qs, ks) * scale
        if causal:
            T = b - a
            mask = torch.triu(
                torch.ones(T, T, dtype=torch.bool, device=q.device), diagonal=1
            )
            if window > 0:
                # sliding window: token i sees only (i-window, i]
                mask |= torch.tril(
                    torch.ones(T, T, dtype=torch.bool, device=q.device),
                    diagonal=-window,
                )
            scores.masked_fill_(mask, float("-inf"))
        probs = torch.softmax(scores, dim=-1)
        o = torch.einsum("hts,hsd->htd", probs, vs)
        out[a:b] = o.transpose(0, 1).to(q.dtype)
    return out


def context_attention_varlen(
    q: torch.Tensor,  # [total_new_tokens, num_heads, head_dim]
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,  # [num_seqs, max_blocks]
    cu_seqlens_q: torch.Tensor,  # [num_seqs+1] new-token offsets
    context_lens: torch.Tensor,  # [num_seqs] total context (incl. new tokens)
    scale: float,
    window: int = 0,
) -> torch.Tensor:
    """Prefill attention against the paged cache (supports chunked prefill:
    new tokens attend to all cached tokens before them, causally)."""
    num_heads = q.shape[1]
    num_kv_heads = k_cache.shape[1]
    head_dim = q.shape[2]
    block_size = k_cache.shape[2]
    group = num_heads // num_kv_heads
    out = torch.empty_like(q)
    for s in range(cu_seqlens_q.numel() - 1):
        a, b = int(cu_seqlens_q[s]), int(cu_seqlens_q[s + 1])
        n_new = b - a
        ctx = int(context_lens[s])
        nb = -(-ctx // block_size)
        blocks = block_tables[s, :nb].long()
        keys = k_cache[blocks].permute(1, 0, 2, 3).reshape(num_kv_heads, -1, head_dim)[
            :, :ctx
        ].float().repeat_interleave(group, dim=0)
        vals = v_cache[blocks].permute(1, 0, 2, 3).reshape(num_kv_heads, -1, head_dim)[
            :, :ctx
        ].float().repeat_interleave(group, dim=0)
        qs = q[a:b].float().transpose(0, 1)  # [H, n_new, D]
        scores = torch.einsum("htd,hsd->hts", qs, keys) * scale
        # causal: new token i (absolute pos ctx - n_new + i) sees pos <= its own
        start = ctx - n_new
        pos_q = torch.arange(start, ctx, device=q.device).unsqueeze(1)
        pos_k = torch.arange(ctx, device=q.device).unsqueeze(0)
        bad = pos_k > pos_q
        if window > 0:
            bad |= pos_k <= pos_q - window
        scores.masked_fill_(bad.unsqueeze(0), float("-inf"))
        probs = torch.softmax(scores, dim=-1)
        o = torch.einsum("hts,hsd->htd", probs, vals)
        out[a:b] = o.transpose(0, 1).to(q.dtype)
    return out


def greedy_sample(logits: torch.Tensor) -> torch.Tensor:
    return logits.argmax(dim=-1).to(torch.int64)


def random_sample(
    logits: torch.Tensor,  # [num_seqs, vocab]
    temperatures: torch.Tensor,  # [num_seqs]
    top_p: torch.Tensor,  # [num_seqs]
    top_k: torch.Tensor,  # [num_seqs] (-1 disables)
    generator: Optional[torch.Generator] = None,
    min_p: Optional[torch.Tensor] = None,  # [num_seqs] (0 disables)
) -> torch.Tensor:
    """Temperature + top-k + top-p + min-p sampling, fp32 reference."""
    lf = logits.float()
    temps = temperatures.clamp_min(1e-5).unsqueeze(-1)
    lf = lf / temps
    vocab = lf.shape[-1]
    # top-k mask
    k_eff = torch.where(top_k > 0, top_k, torch.full_like(top_k, vocab))
    sorted_logits, sorted_idx = lf.sort(dim=-1, descending=True)
    ranks = torch.arange(vocab, device=lf.device).unsqueeze(0)
    mask = ranks >= k_eff.unsqueeze(-1)
    # top-p mask on sorted probs
    probs_sorted = torch.softmax(sorted_logits, dim=-1)
    cumprobs = probs_sorted.cumsum(dim=-1)
    mask |= (cumprobs - probs_sorted) > top_p.unsqueeze(-1)
    if min_p is not None:
        # min-p: keep tokens with prob >= min_p * p_max
        mask |= probs_sorted < (min_p.unsqueeze(-1) * probs_sorted[:, :1])
    sorted_logits = sorted_logits.masked_fill(mask, float("-inf"))
    # degenerate rows (everything masked, e.g. an exhausted guided-decoding
    # machine with EOS disabled): pick rank 0 deterministically instead of
    # feeding NaNs to multinomial — callers discard the token
    dead = torch.isinf(sorted_logits).all(dim=-1)
    if bool(dead.any()):
        sorted_logits[dead, 0] = 0.0
    probs = torch.softmax(sorted_logits, dim=-1)
    choice = torch.multinomial(probs, 1, generator=generator).squeeze(-1)
    return sorted_idx.gather(-1, choice.unsqueeze(-1)).squeeze(-1)


def layer_norm(
    x: torch.Tensor, weight: torch.Tensor, bias: torch.Tensor, eps: float
) -> torch.Tensor:
    return torch.nn.functional.layer_norm(
        x.float(), (x.shape[-1],), weight.float(), bias.float(), eps
    ).to(x.dtype)


def fused_add_layer_norm(
    x: torch.Tensor,
    residual: torch.Tensor,
    weight: torch.Tensor,
    bias: torch.Tensor,
    eps: float,
) -> torch.Tensor:
    return layer_norm((x.float() + residual.float()).to(x.dtype), weight, bias, eps)


def gelu(x: torch.Tensor) -> torch.Tensor:
    return torch.nn.functional.gelu(x.float()).to(x.dtype)
