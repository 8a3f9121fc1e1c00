"""Static instruction-shape checks (no GPU needed): hipcc -S each hot
kernel for gfx950 and assert the code generator produced the structures
the design depends on — MFMA on the matrix cores, global_load_lds
staging, counted waitcnts. Catches silent regressions (e.g. a refactor
that drops MFMA to VALU loops or re-introduces scalar staging)."""

import os
import re
import subprocess

import pytest

CSRC = os.path.join(os.path.dirname(__file__), "..", "kserve_amd", "ops", "csrc")
HIPCC = "/opt/rocm/bin/hipcc"

pytestmark = pytest.mark.skipif(
    not os.path.exists(HIPCC), reason="hipcc not available"
)


def compile_to_asm(tmp_path, name):
    out = tmp_path / f"{name}.s"
    subprocess.run(
        [
            HIPCC, "-S", os.path.join(CSRC, f"{name}.hip"), "-o", str(out),
            "--offload-arch=gfx950", "-O3", "-std=c++17", "-fPIC",
            "-ffast-math", "--cuda-device-only",
        ],
        check=True,
        capture_output=True,
        timeout=300,
    )
    return out.read_text()


def kernel_bodies(asm):
    """mangled-name -> body text"""
    parts = re.split(r"\n([_A-Za-z0-9$.]+):\s*;+\s*@\1", asm)
    out = {}
    for i in range(1, len(parts) - 1, 2):
        out[parts[i]] = parts[i + 1]
    return out


class TestPrefillAsm:
    def test_prefill_kernels_use_mfma(self, tmp_path):
        asm = compile_to_asm(tmp_path, "attention_prefill")
        bodies = kernel_bodies(asm)
        v2 = [b for n, b in bodies.items() if "flash_prefill_v2" in n]
        assert v2, "v2 kernel missing from codegen"
        for body in v2:
            n_mfma = len(re.findall(r"v_mfma_f32_16x16x32_bf16", body))
            assert n_mfma >= 24, f"prefill v2 lost its MFMA tiling ({n_mfma})"
        ctx = [b for n, b in bodies.items() if "context_prefill" in n]
        assert ctx, "paged-context kernel missing"
        for body in ctx:
            assert re.search(r"v_mfma_f32_16x16x32_bf16", body)


class TestDecodeAsm:
    def test_decode_uses_wide_loads_not_scalar(self, tmp_path):
        asm = compile_to_asm(tmp_path, "attention_decode")
        # the fast paths must issue 128-bit global loads (burst staging)
        assert "global_load_dwordx4" in asm
        # fp8 kernel must use the packed hardware converts, not bit math
        assert "v_cvt_pk_f32_fp8" in asm

    def test_gemm8_matches_template_shape(self, tmp_path):
        """The experimental 8-phase GEMM's emitted structure must match the
        template: 16 MFMA x 4 phases, global_load_lds staging, setprio
        around the MFMA bursts, counted (non-zero-only-at-boundary) vmcnt."""
        asm = compile_to_asm(tmp_path, "gemm8")
        bodies = kernel_bodies(asm)
        # canonical variants only: SCHED 0/1, ABLATE 0 (mangled
        # gemm8_kernelILi<sw>ELi<sched>ELi0EE); sched 3/4 drop a barrier
        # and ablate bodies intentionally drop instructions
        g8 = {
            n: b for n, b in bodies.items()
            if re.search(r"gemm8_kernelILi[0-2]ELi[01]ELi0EE", n)
        }
        assert len(g8) == 6, f"expected 6 instantiations, got {list(g8)}"
        for name, body in g8.items():
            counts = {
                "mfma": len(re.findall(r"v_mfma_f32_16x16x32_bf16", body)),
                "glds": len(re.findall(r"global_load_lds", body)),
                "setprio": len(re.findall(r"s_setprio", body)),
                "ds_read": len(re.findall(r"ds_read_b128", body)),
                "ds_write": len(re.findall(r"ds_write", body)),
            }
            assert counts["mfma"] == 64, (name, counts)   # 4 phases x 16
            assert counts["glds"] == 16, (name, counts)   # 8 prologue + 8 loop
            assert counts["setprio"] == 8, (name, counts)
            assert counts["ds_read"] == 48, (name, counts)  # 12 x 4 phases
            assert counts["ds_write"] == 0, (name, counts)  # staging is DMA

    def test_gemm8_half_granular_counted_vmcnt(self, tmp_path):
        """SCHED=2 half-granular schedule: one 16 KiB unit staged per
        phase, a single counted vmcnt(4) per K-tile in the main loop
        (the guide's T3+T4 discipline — never drain to 0 mid-loop)."""
        asm = compile_to_asm(tmp_path, "gemm8")
        bodies = kernel_bodies(asm)
        hg = {n: b for n, b in bodies.items()
              if "gemm8_hg_kernel" in n and "stub" not in n
              and "v_mfma" in b}
        assert len(hg) == 3, f"expected 3 hg instantiations, got {list(hg)}"
        for name, body in hg.items():
            counts = {
                "mfma": len(re.findall(r"v_mfma_f32_16x16x32_bf16", body)),
                "glds": len(re.findall(r"global_load_lds", body)),
                "vmcnt4": len(re.findall(r"s_waitcnt vmcnt\(4\)", body)),
                "vmcnt0": len(re.findall(r"s_waitcnt vmcnt\(0\)", body)),
                "ds_write": len(re.findall(r"ds_write", body)),
            }
            assert counts["mfma"] == 64, (name, counts)
            # 12 prologue (6 units) + 8 main loop (4 units x 2 loads)
            assert counts["glds"] == 20, (name, counts)
            assert counts["vmcnt4"] >= 1, (name, counts)   # counted waits
            assert counts["vmcnt0"] <= 2, (name, counts)   # epilogue only
            assert counts["ds_write"] == 0, (name, counts)


class TestSamplerAsm:
    def test_topk_sampler_has_no_scratch(self, tmp_path):
        """The radix-select sampler must stay in registers/LDS (scratch
        spills would crater the 4-pass vocab scan)."""
        asm = compile_to_asm(tmp_path, "sampling")
        assert "scratch_store" not in asm, "sampler spilled to scratch"
