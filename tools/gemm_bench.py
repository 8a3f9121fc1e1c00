#!/usr/bin/env python3
"""Microbench the decode-path GEMM shapes: achieved TB/s (weight traffic) per
shape, to judge hipBLASLt (tuned/untuned) against the HBM roofline.

Run on GPU: python tools/gemm_bench.py [--tune]
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

parser = argparse.ArgumentParser()
parser.add_argument("--tune", action="store_true")
parser.add_argument("--iters", type=int, default=50)
args = parser.parse_args()

if args.tune:
    os.environ["PYTORCH_TUNABLEOP_ENABLED"] = "1"
    os.environ["PYTORCH_TUNABLEOP_TUNING"] = "1"
    os.environ["PYTORCH_TUNABLEOP_MAX_TUNING_DURATION_MS"] = "200"
    os.environ["PYTORCH_TUNABLEOP_FILENAME"] = "gpurun_out/tunableop_decode.csv"
else:
    base = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                        "profiles", "tunableop_gfx950.csv")
    if os.path.exists(base.replace(".csv", "0.csv")):
        os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
        os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "0")
        os.environ.setdefault("PYTORCH_TUNABLEOP_FILENAME", base)

import torch
import torch.nn.functional as F

assert torch.cuda.is_available()
dev = "cuda:0"

# decode GEMMs for llama-3-8b at token batch N (plus prefill-ish 16384)
SHAPES = [
    # (name, N tokens, K in, M out)
    ("qkv", 256, 4096, 6144),
    ("o", 256, 4096, 4096),
    ("gate_up", 256, 4096, 28672),
    ("down", 256, 14336, 4096),
    ("lm_head", 256, 4096, 128256),
    ("qkv", 64, 4096, 6144),
    ("gate_up", 64, 4096, 28672),
    ("down", 64, 14336, 4096),
    ("lm_head", 64, 4096, 128256),
    ("qkv_prefill", 16384, 4096, 6144),
    ("gate_up_prefill", 16384, 4096, 28672),
]

import kserve_amd_C

def bench_custom(x, w, iters):
    out = torch.empty(x.shape[0], w.shape[0], dtype=torch.bfloat16, device=dev)
    for _ in range(10):
        kserve_amd_C.skinny_gemm(out, x, w)
    torch.cuda.synchronize()
    t0 = torch.cuda.Event(enable_timing=True); t1 = torch.cuda.Event(enable_timing=True)
    t0.record()
    for _ in range(iters):
        kserve_amd_C.skinny_gemm(out, x, w)
    t1.record(); torch.cuda.synchronize()
    return t0.elapsed_time(t1) / iters * 1000

print(f"{'shape':18s} {'N':>6s} {'K':>6s} {'M':>7s} {'us':>9s} {'TB/s(w)':>8s} {'TFLOP/s':>8s} {'cust_us':>8s} {'cust_TB/s':>9s}")
for name, N, K, M in SHAPES:
    x = torch.randn(N, K, dtype=torch.bfloat16, device=dev)
    w = torch.randn(M, K, dtype=torch.bfloat16, device=dev)
    for _ in range(10):
        y = F.linear(x, w)
    torch.cuda.synchronize()
    t0 = torch.cuda.Event(enable_timing=True)
    t1 = torch.cuda.Event(enable_timing=True)
    t0.record()
    for _ in range(args.iters):
        y = F.linear(x, w)
    t1.record()
    torch.cuda.synchronize()
    us = t0.elapsed_time(t1) / args.iters * 1000
    wbytes = M * K * 2
    tbs = wbytes / (us * 1e-6) / 1e12
    tf = 2 * N * K * M / (us * 1e-6) / 1e12
    cus, ctbs = float("nan"), float("nan")
    if N <= 256 and M % 64 == 0 and K % 64 == 0:
        cus = bench_custom(x, w, args.iters)
        ctbs = wbytes / (cus * 1e-6) / 1e12
    print(f"{name:18s} {N:6d} {K:6d} {M:7d} {us:9.1f} {tbs:8.2f} {tf:8.1f} {cus:8.1f} {ctbs:9.2f}")
