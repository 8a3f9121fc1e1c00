#!/usr/bin/env python3
"""Single-shape gemm8 run for PMC capture (rocprofv3 --pmc wraps this)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import kserve_amd_C  # noqa: E402

dev = "cuda:0"
M = N = K = 4096
a = torch.randn(M, K, dtype=torch.bfloat16, device=dev)
w = torch.randn(N, K, dtype=torch.bfloat16, device=dev)
d = torch.empty(M, N, dtype=torch.bfloat16, device=dev)
sw = int(os.environ.get("KS_GEMM8_SW", "1"))
for _ in range(5):
    kserve_amd_C.gemm8(d, a, w, sw)
torch.cuda.synchronize()
print("done")
