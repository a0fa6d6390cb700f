"""In-tree build of the gpustack_amd HIP extension for gfx950 (MI355X).

Drives hipcc directly (no hipify, no CUDA compat layer): the .hip sources
are native HIP/CDNA4 and the binding file links against torch-ROCm's native
c10/hip API. The resulting _hip_ops.so lives inside the package so it
travels with repo snapshots (and stays out of git via .gitignore).
"""
from __future__ import annotations

import os
import subprocess
import sys
import sysconfig
from pathlib import Path

PKG_DIR = Path(__file__).resolve().parent
CSRC = PKG_DIR / "csrc"
SO_PATH = PKG_DIR / "_hip_ops.so"
ARCH = os.environ.get("GPUSTACK_AMD_ARCH", "gfx950")

HIP_SOURCES = ["pointwise.hip", "attn_decode.hip", "attn_prefill.hip", "skinny_gemm.hip", "gemm8.hip", "gemm8_lab.hip", "moe_gemm.hip", "w4_gemm.hip", "mla_decode.hip"]
CPP_SOURCES = ["bindings.cpp"]


def _torch_paths():
    import torch

    troot = Path(torch.__file__).resolve().parent
    includes = [
        str(troot / "include"),
        str(troot / "include" / "torch" / "csrc" / "api" / "include"),
        sysconfig.get_paths()["include"],
        str(CSRC),
    ]
    libdir = str(troot / "lib")
    abi = "1" if torch._C._GLIBCXX_USE_CXX11_ABI else "0"
    return includes, libdir, abi


def _common_flags(abi: str):
    return [
        "-O3",
        "-std=c++17",
        "-fPIC",
        f"--offload-arch={ARCH}",
        f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
        "-DUSE_ROCM=1",
        "-D__HIP_PLATFORM_AMD__=1",
        "-DHIPBLAS_V2",
        "-DHIP_ENABLE_WARP_SYNC_BUILTINS=1",
        "-DTORCH_EXTENSION_NAME=_hip_ops",
        "-Wno-unused-result",
        "-fvisibility=hidden",
    ]


def _run(cmd: list[str]) -> None:
    proc = subprocess.run(cmd, capture_output=True, text=True)
    if proc.returncode != 0:
        raise RuntimeError(
            f"build command failed ({' '.join(cmd[:3])} ...):\n{proc.stderr[-8000:]}"
        )


def build(verbose: bool = False, force: bool = False) -> Path:
    includes, libdir, abi = _torch_paths()
    inc_flags = [f"-I{p}" for p in includes]
    flags = _common_flags(abi)

    objs = []
    newest_src = 0.0
    for src in HIP_SOURCES + CPP_SOURCES + ["common.h"]:
        p = CSRC / src
        if p.exists():
            newest_src = max(newest_src, p.stat().st_mtime)
    if SO_PATH.exists() and not force and SO_PATH.stat().st_mtime >= newest_src:
        return SO_PATH

    build_dir = CSRC / ".build"
    build_dir.mkdir(exist_ok=True)
    for src in HIP_SOURCES + CPP_SOURCES:
        obj = build_dir / (Path(src).stem + ".o")
        srcp = CSRC / src
        if (
            not force
            and obj.exists()
            and obj.stat().st_mtime >= srcp.stat().st_mtime
            and obj.stat().st_mtime >= (CSRC / "common.h").stat().st_mtime
        ):
            objs.append(str(obj))
            continue
        cmd = ["hipcc", *flags, *inc_flags, "-c", str(srcp), "-o", str(obj)]
        if src.endswith(".cpp"):
            cmd.insert(1, "-x")
            cmd.insert(2, "hip")  # binding file still needs hip runtime types
        if verbose:
            print(" ".join(cmd), file=sys.stderr)
        _run(cmd)
        objs.append(str(obj))

    link = [
        "hipcc",
        "-shared",
        f"--offload-arch={ARCH}",
        *objs,
        f"-L{libdir}",
        "-ltorch",
        "-ltorch_hip",
        "-ltorch_cpu",
        "-lc10",
        "-lc10_hip",
        "-ltorch_python",
        "-L/opt/rocm/lib",
        "-lamdhip64",
        f"-Wl,-rpath,{libdir}",
        "-o",
        str(SO_PATH),
    ]
    if verbose:
        print(" ".join(link), file=sys.stderr)
    _run(link)
    return SO_PATH


if __name__ == "__main__":
    build(verbose=True, force="--force" in sys.argv)
    print(f"built {SO_PATH}")
