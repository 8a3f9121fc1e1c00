#!/usr/bin/env python3
"""8-phase GEMM bench vs hipBLASLt (EXPERIMENTAL kernel; see gemm8.hip).
Round-2 entry point: validate numerics (KS_GEMM8=1 pytest -k gemm8) then
run this for TF comparisons at the decode shapes + guide squares."""

import os
import sys
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import kserve_amd_C  # noqa: E402

assert torch.cuda.is_available()
dev = "cuda:0"
torch.manual_seed(0)

SHAPES = [
    (4096, 4096, 4096),   # guide reference square (expect ~1.5 PF w/ swizzle)
    (1536, 6144, 4096),   # qkv @ concurrency 1536
    (1536, 4096, 4096),   # o_proj
    (1536, 28672, 4096),  # gate_up
    (1536, 4096, 14336),  # down (K not mult of 256 is fine: BK=64)
]

def time_fn(fn, iters=20):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = torch.cuda.Event(enable_timing=True); t1 = torch.cuda.Event(enable_timing=True)
    t0.record()
    for _ in range(iters):
        fn()
    t1.record(); torch.cuda.synchronize()
    return t0.elapsed_time(t1) / iters * 1e3  # us

print(f"{'M':>6} {'N':>6} {'K':>6} {'blaslt TF':>10} {'g8 TF':>8} {'sw1 TF':>8} {'sw2 TF':>8}")
for M, N, K in SHAPES:
    a = torch.randn(M, K, dtype=torch.bfloat16, device=dev)
    w = torch.randn(N, K, dtype=torch.bfloat16, device=dev)
    d = torch.empty(M, N, dtype=torch.bfloat16, device=dev)
    fl = 2.0 * M * N * K
    t_ref = time_fn(lambda: torch.nn.functional.linear(a, w))
    row = [fl / (t_ref * 1e-6) / 1e12]
    for sw in (0, 1, 2):
        if M % 256 or N % 256 or K % 64:
            row.append(float("nan")); continue
        try:
            t = time_fn(lambda: kserve_amd_C.gemm8(d, a, w, sw))
            row.append(fl / (t * 1e-6) / 1e12)
        except Exception as e:
            print("  gemm8 failed:", e)
            row.append(float("nan"))
    print(f"{M:6d} {N:6d} {K:6d} {row[0]:10.1f} {row[1]:8.1f} {row[2]:8.1f} {row[3]:8.1f}")


# m233-style structural decomposition (KS_GEMM8_ABLATE): results WRONG,
# timing only — where do the non-MFMA cycles go?
if os.environ.get("KS_GEMM8_DECOMP") == "1":
    M = N = K = 4096
    a = torch.randn(M, K, dtype=torch.bfloat16, device=dev)
    w = torch.randn(N, K, dtype=torch.bfloat16, device=dev)
    d = torch.empty(M, N, dtype=torch.bfloat16, device=dev)
    fl = 2.0 * M * N * K
    import subprocess
    names = {1: "no-stage (ds_read+MFMA+bar)",
             2: "no-dsread (stage+MFMA+bar)",
             3: "MFMA-only (+bar)"}
    print("decomposition @4096^3 (sw1, sched5; separate processes):")
    for abl, label in names.items():
        out = subprocess.run(
            ["python", "-c", f"""
import os, sys
os.environ['KS_GEMM8_ABLATE'] = '{abl}'
sys.path.insert(0, {repr(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))})
import torch, kserve_amd_C
dev = 'cuda:0'
M = N = K = 4096
a = torch.randn(M, K, dtype=torch.bfloat16, device=dev)
w = torch.randn(N, K, dtype=torch.bfloat16, device=dev)
d = torch.empty(M, N, dtype=torch.bfloat16, device=dev)
for _ in range(3): kserve_amd_C.gemm8(d, a, w, 1)
torch.cuda.synchronize()
t0 = torch.cuda.Event(enable_timing=True); t1 = torch.cuda.Event(enable_timing=True)
t0.record()
for _ in range(20): kserve_amd_C.gemm8(d, a, w, 1)
t1.record(); torch.cuda.synchronize()
print(t0.elapsed_time(t1) / 20 * 1e3)
"""],
            capture_output=True, text=True)
        try:
            us = float(out.stdout.strip().splitlines()[-1])
            print(f"  ablate {abl} {label:32s}: {fl / (us * 1e-6) / 1e12:7.1f} TF")
        except Exception:
            print(f"  ablate {abl} failed: {out.stderr[-200:]}")
