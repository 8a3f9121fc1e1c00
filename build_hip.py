#!/usr/bin/env python3
"""Build the kserve_amd_C native extension in-tree for gfx950.

Kernels (.hip) are compiled by hipcc directly (pure CDNA4 HIP, no torch
coupling, no hipify); bindings.cpp compiles against torch's ROCm headers and
links the objects into kserve_amd_C.so at the repo root — the .so travels to
GPU boxes with the source snapshot.

Usage: python build_hip.py [--force]
"""

import os
import subprocess
import sys

REPO = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(REPO, "kserve_amd", "ops", "csrc")
BUILD = os.path.join(REPO, "build", "hip")
OUT_SO = os.path.join(REPO, "kserve_amd_C.so")

HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")

KERNEL_SOURCES = [
    "rmsnorm.hip",
    "layernorm.hip",
    "activation.hip",
    "rope.hip",
    "kv_cache.hip",
    "attention_decode.hip",
    "attention_prefill.hip",
    "sampling.hip",
    "skinny_gemm.hip",
    "gemm8.hip",
    "mfma_probe.hip",
]


def run(cmd):
    print("+", " ".join(cmd), flush=True)
    subprocess.check_call(cmd)


def newer(src, dst):
    return not os.path.exists(dst) or os.path.getmtime(src) > os.path.getmtime(dst)


def build(force: bool = False) -> str:
    import torch
    import sysconfig

    os.makedirs(BUILD, exist_ok=True)
    torch_dir = os.path.dirname(torch.__file__)
    torch_inc = [
        os.path.join(torch_dir, "include"),
        os.path.join(torch_dir, "include", "torch", "csrc", "api", "include"),
    ]
    py_inc = sysconfig.get_paths()["include"]

    common_h = os.path.join(CSRC, "common.h")
    layouts_h = os.path.join(CSRC, "mfma_layouts.h")
    objs = []
    for src in KERNEL_SOURCES:
        src_path = os.path.join(CSRC, src)
        obj = os.path.join(BUILD, src.replace(".hip", ".o"))
        objs.append(obj)
        if force or newer(src_path, obj) or newer(common_h, obj) or newer(layouts_h, obj):
            run([
                HIPCC, "-c", src_path, "-o", obj,
                f"--offload-arch={ARCH}", "-O3", "-std=c++17", "-fPIC",
                "-ffast-math",
            ])

    bind_src = os.path.join(CSRC, "bindings.cpp")
    bind_obj = os.path.join(BUILD, "bindings.o")
    if force or newer(bind_src, bind_obj):
        cmd = [
            HIPCC, "-c", bind_src, "-o", bind_obj,
            "-O2", "-std=c++17", "-fPIC",
            "-D_GLIBCXX_USE_CXX11_ABI=1",
            "-DTORCH_EXTENSION_NAME=kserve_amd_C",
            "-DUSE_ROCM", "-D__HIP_PLATFORM_AMD__",
            "-DTORCH_API_INCLUDE_EXTENSION_H",
            f"-I{py_inc}",
        ]
        cmd += [f"-I{d}" for d in torch_inc]
        run(cmd)
    objs.append(bind_obj)

    if force or any(newer(o, OUT_SO) for o in objs):
        run([
            HIPCC, "-shared", "-o", OUT_SO, *objs,
            f"-L{os.path.join(torch_dir, 'lib')}",
            "-ltorch", "-ltorch_cpu", "-ltorch_python", "-lc10",
            "-ltorch_hip", "-lc10_hip", "-lamdhip64",
            f"-Wl,-rpath,{os.path.join(torch_dir, 'lib')}",
        ])
    print(f"built {OUT_SO}")
    return OUT_SO


if __name__ == "__main__":
    build(force="--force" in sys.argv)
