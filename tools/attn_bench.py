#!/usr/bin/env python3
"""Decode-attention kernel microbench: achieved KV-read TB/s vs the ~6.3 TB/s
HBM ceiling, across bench-relevant shapes. Run on GPU."""

import math
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from kserve_amd import ops

assert torch.cuda.is_available()
dev = "cuda:0"
torch.manual_seed(0)

SHAPES = [
    # (S, H, Hkv, ctx)
    (256, 32, 8, 576),   # bench batch 256
    (512, 32, 8, 576),
    (64, 32, 8, 576),
    (8, 32, 8, 576),     # latency mode (split-context)
    (256, 32, 8, 2048),  # long context
    (64, 64, 8, 576),    # 70b-ish TP=1 heads
]

D, bs = 128, 16
print(f"{'S':>4} {'H':>3} {'Hkv':>3} {'ctx':>5} {'us':>8} {'TB/s':>6}")
for S, H, Hkv, ctx in SHAPES:
    nb = (ctx + bs - 1) // bs
    B = S * nb + 1
    kc = torch.randn(B, Hkv, bs, D, dtype=torch.bfloat16, device=dev)
    vc = torch.randn(B, Hkv, bs, D, dtype=torch.bfloat16, device=dev)
    bt = torch.arange(1, S * nb + 1, dtype=torch.int32, device=dev).reshape(S, nb)
    ctx_t = torch.full((S,), ctx, dtype=torch.int32, device=dev)
    q = torch.randn(S, H, D, dtype=torch.bfloat16, device=dev)
    scale = 1.0 / math.sqrt(D)
    out = torch.empty_like(q)
    for _ in range(5):
        ops.paged_attention_decode(q, kc, vc, bt, ctx_t, scale, out=out)
    torch.cuda.synchronize()
    t0 = torch.cuda.Event(enable_timing=True)
    t1 = torch.cuda.Event(enable_timing=True)
    t0.record()
    iters = 30
    for _ in range(iters):
        ops.paged_attention_decode(q, kc, vc, bt, ctx_t, scale, out=out)
    t1.record()
    torch.cuda.synchronize()
    us = t0.elapsed_time(t1) / iters * 1000
    kv_bytes = S * ctx * Hkv * D * 2 * 2  # K+V read once per kv-head group
    tbs = kv_bytes / (us * 1e-6) / 1e12
    print(f"{S:4d} {H:3d} {Hkv:3d} {ctx:5d} {us:8.1f} {tbs:6.2f}")

# fp8 KV cache variant (half the bytes; tok/s-equivalent speedup at large
# batch where decode attention dominates)
print("\nfp8 E4M3 KV cache:")
print(f"{'S':>4} {'H':>3} {'Hkv':>3} {'ctx':>5} {'us':>8} {'TB/s':>6} {'vs bf16':>8}")
for S, H, Hkv, ctx in SHAPES[:3] + SHAPES[4:5]:
    nb = (ctx + bs - 1) // bs
    B = S * nb + 1
    kc = torch.randn(B, Hkv, bs, D, dtype=torch.bfloat16, device=dev).to(
        torch.float8_e4m3fn
    )
    vc = torch.randn(B, Hkv, bs, D, dtype=torch.bfloat16, device=dev).to(
        torch.float8_e4m3fn
    )
    bt = torch.arange(1, S * nb + 1, dtype=torch.int32, device=dev).reshape(S, nb)
    ctx_t = torch.full((S,), ctx, dtype=torch.int32, device=dev)
    q = torch.randn(S, H, D, dtype=torch.bfloat16, device=dev)
    scale = 1.0 / math.sqrt(D)
    out = torch.empty_like(q)
    for _ in range(5):
        ops.paged_attention_decode(q, kc, vc, bt, ctx_t, scale, out=out)
    torch.cuda.synchronize()
    t0 = torch.cuda.Event(enable_timing=True)
    t1 = torch.cuda.Event(enable_timing=True)
    t0.record()
    iters = 30
    for _ in range(iters):
        ops.paged_attention_decode(q, kc, vc, bt, ctx_t, scale, out=out)
    t1.record()
    torch.cuda.synchronize()
    us = t0.elapsed_time(t1) / iters * 1000
    kv_bytes = S * ctx * Hkv * D * 2 * 1  # fp8: 1 byte
    tbs = kv_bytes / (us * 1e-6) / 1e12
    print(f"{S:4d} {H:3d} {Hkv:3d} {ctx:5d} {us:8.1f} {tbs:6.2f}")
