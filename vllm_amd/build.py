"""In-tree build of the vllm_amd HIP/CDNA4 extension (gfx950 only).

Drives hipcc directly (no JIT cache — the .so lands next to the package
so the gpurun snapshot carries it). Usage:

    python -m vllm_amd.build           # build vllm_amd/_C.so
    python -m vllm_amd.build --check   # build only if sources newer
"""

from __future__ import annotations

import os
import subprocess
import sys
from pathlib import Path

import torch

PKG_DIR = Path(__file__).resolve().parent
CSRC = PKG_DIR / "csrc"
OUT_SO = PKG_DIR / "_C.so"

SOURCES = [
    CSRC / "elementwise.hip",
    CSRC / "comms.hip",
    CSRC / "moe.hip",
    CSRC / "mla.hip",
    CSRC / "quant_fp8.hip",
    CSRC / "attention_decode.hip",
    CSRC / "attention_prefill.hip",
    CSRC / "gemm_hipblaslt.cpp",
    CSRC / "bindings.cpp",
]

HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def _torch_flags() -> tuple[list[str], list[str]]:
    torch_dir = Path(torch.__file__).parent
    inc = [
        f"-I{torch_dir}/include",
        f"-I{torch_dir}/include/torch/csrc/api/include",
    ]
    abi = "1" if torch.compiled_with_cxx11_abi() else "0"
    cxx = [
        "-std=c++17",
        "-O3",
        f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
        "-DUSE_ROCM",
        "-D__HIP_PLATFORM_AMD__",
        "-fPIC",
        f"--offload-arch={ARCH}",
        "-fno-gpu-rdc",
        "-Wno-unused-result",
    ]
    link = [
        f"-L{torch_dir}/lib",
        "-ltorch",
        "-ltorch_hip",
        "-lc10",
        "-lc10_hip",
        "-lamdhip64",
        "-lhipblaslt",
        "-L/opt/rocm/lib",
        "-Wl,-rpath,/opt/rocm/lib",
        f"-Wl,-rpath,{torch_dir}/lib",
    ]
    return inc + cxx, link


def needs_build() -> bool:
    if not OUT_SO.exists():
        return True
    so_mtime = OUT_SO.stat().st_mtime
    deps = list(SOURCES) + [CSRC / "common.h", Path(__file__)]
    return any(p.stat().st_mtime > so_mtime for p in deps)


def build(verbose: bool = True) -> Path:
    flags, link = _torch_flags()
    objs = []
    build_dir = CSRC / ".build"
    build_dir.mkdir(exist_ok=True)
    for src in SOURCES:
        obj = build_dir / (src.stem + ".o")
        cmd = [HIPCC, "-c", str(src), "-o", str(obj)] + flags
        if src.suffix == ".hip":
            cmd.insert(1, "-x")
            cmd.insert(2, "hip")
        if verbose:
            print("[vllm_amd.build]", " ".join(cmd), file=sys.stderr)
        subprocess.run(cmd, check=True)
        objs.append(str(obj))
    cmd = [HIPCC, "-shared", "-o", str(OUT_SO)] + objs + link
    if verbose:
        print("[vllm_amd.build]", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)
    return OUT_SO


def ensure_built(verbose: bool = False) -> Path:
    if needs_build():
        build(verbose=verbose)
    return OUT_SO


if __name__ == "__main__":
    if "--check" in sys.argv:
        ensure_built(verbose=True)
    else:
        build(verbose=True)
    print(f"built {OUT_SO}")
