#!/usr/bin/env python3
"""Decode-attention PMC target: repeated V4-path launches at the headline
shape (S=256, ctx=576) for rocprofv3 --pmc."""
import math
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from kserve_amd import ops

dev = "cuda:0"
torch.manual_seed(0)
S, H, Hkv, ctx, D, bs = 256, 32, 8, 576, 128, 16
nb = (ctx + bs - 1) // bs
B = S * nb + 1
kc = torch.randn(B, Hkv, bs, D, dtype=torch.bfloat16, device=dev)
vc = torch.randn(B, Hkv, bs, D, dtype=torch.bfloat16, device=dev)
bt = torch.arange(1, S * nb + 1, dtype=torch.int32, device=dev).reshape(S, nb)
ct = torch.full((S,), ctx, dtype=torch.int32, device=dev)
q = torch.randn(S, H, D, dtype=torch.bfloat16, device=dev)
out = torch.empty_like(q)
for _ in range(int(os.environ.get("KS_ATTN_PMC_ITERS", "10"))):
    ops.paged_attention_decode(q, kc, vc, bt, ct, 1.0 / math.sqrt(D), out=out)
torch.cuda.synchronize()
print("done")
