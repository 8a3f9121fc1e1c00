#!/usr/bin/env python3
"""Flash prefill attention microbench: TFLOP/s at bench shapes."""

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
    # (num_seqs, seqlen, Hq, Hkv)
    (32, 512, 32, 8),   # one 16K-token prefill step of the bench
    (8, 2048, 32, 8),
    (2, 8192, 32, 8),
    (32, 512, 64, 8),   # 70b heads
]
D = 128
print(f"{'S':>4} {'len':>5} {'Hq':>3} {'us':>9} {'TFLOP/s':>8}")
for S, L, Hq, Hkv in SHAPES:
    total = S * L
    cu = torch.arange(0, (S + 1) * L, L, dtype=torch.int32, device=dev)
    q = torch.randn(total, Hq, D, dtype=torch.bfloat16, device=dev)
    k = torch.randn(total, Hkv, D, dtype=torch.bfloat16, device=dev)
    v = torch.randn(total, Hkv, D, dtype=torch.bfloat16, device=dev)
    scale = 1.0 / math.sqrt(D)
    for _ in range(3):
        out = ops.flash_prefill_varlen(q, k, v, cu, L, scale)
    torch.cuda.synchronize()
    t0 = torch.cuda.Event(enable_timing=True)
    t1 = torch.cuda.Event(enable_timing=True)
    t0.record()
    iters = 20
    for _ in range(iters):
        out = ops.flash_prefill_varlen(q, k, v, cu, L, scale)
    t1.record()
    torch.cuda.synchronize()
    us = t0.elapsed_time(t1) / iters * 1000
    # causal: 2 matmuls (QK+PV) x L^2/2 x D x Hq per seq
    flops = S * 2 * 2 * (L * L / 2) * D * Hq
    tf = flops / (us * 1e-6) / 1e12
    print(f"{S:4d} {L:5d} {Hq:3d} {us:9.1f} {tf:8.1f}")
