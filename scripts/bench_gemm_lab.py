#!/usr/bin/env python3
"""A/B the gemm8 schedule-lab variants against hipBLASLt on the serving
GEMM shapes (r03 roofline table). Run on an MI355X box:

    python scripts/bench_gemm_lab.py [--iters 50] [--modes 0,1,2,3]

Numerics: every variant is checked against F.linear (bf16 via fp32 ref
tolerance) before timing; a mismatching variant reports ERR and is skipped.
"""
from __future__ import annotations

import argparse
import json
import time

import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import torch
import torch.nn.functional as F

from gpustack_amd import ops

SHAPES = [
    ("qkv_dec", 512, 6144, 4096),
    ("o_dec", 512, 4096, 4096),
    ("gate_up_dec", 512, 28672, 4096),
    ("down_dec", 512, 4096, 14336),
    ("lm_head_dec", 512, 128256, 4096),
    ("square4k", 4096, 4096, 4096),
    ("qkv_pre", 8192, 6144, 4096),
    ("gate_up_pre", 8192, 28672, 4096),
]


def time_fn(fn, iters):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=50)
    ap.add_argument("--modes", default="0,1,2,3")
    ap.add_argument("--shapes", default=None, help="comma list of shape names")
    args = ap.parse_args()
    modes = [int(m) for m in args.modes.split(",")]
    torch.manual_seed(0)
    hip = ops._load_hip()
    dev = "cuda:0"
    results = []
    for name, M, N, K in SHAPES:
        if args.shapes and name not in args.shapes.split(","):
            continue
        x = torch.randn(M, K, dtype=torch.bfloat16, device=dev) / 8
        w = torch.randn(N, K, dtype=torch.bfloat16, device=dev) / 8
        ref = F.linear(x, w)
        flops = 2.0 * M * N * K
        t_blt = time_fn(lambda: F.linear(x, w), args.iters)
        row = {"shape": name, "M": M, "N": N, "K": K,
               "blaslt_us": round(t_blt * 1e6, 1),
               "blaslt_tf": round(flops / t_blt / 1e12, 0)}
        for mode in modes:
            out = torch.empty(M, N, dtype=torch.bfloat16, device=dev)
            try:
                hip.gemm_lab(out, x, w, mode)
                torch.cuda.synchronize()
            except Exception as e:  # noqa: BLE001
                row[f"m{mode}"] = f"LAUNCH_ERR {e}"[:60]
                continue
            rel = (out.float() - ref.float()).abs().max().item()
            scale = ref.float().abs().max().item() + 1e-6
            if rel / scale > 2e-2:
                row[f"m{mode}"] = f"NUMERICS_ERR {rel/scale:.3e}"
                continue
            t = time_fn(lambda: hip.gemm_lab(out, x, w, mode), args.iters)
            row[f"m{mode}_us"] = round(t * 1e6, 1)
            row[f"m{mode}_tf"] = round(flops / t / 1e12, 0)
            row[f"m{mode}_vs_blt"] = round(t_blt / t, 3)
        results.append(row)
        print(json.dumps(row), flush=True)
    return results


if __name__ == "__main__" and "--skinny" not in __import__("sys").argv:
    main()


def skinny_sweep(iters=50):
    """skinny_gemm v1/v2 split-K sweep on the small-grid decode shapes."""
    dev = "cuda:0"
    for name, M, N, K in SHAPES:
        if M > 1024:
            continue
        x = torch.randn(M, K, dtype=torch.bfloat16, device=dev) / 8
        w = torch.randn(N, K, dtype=torch.bfloat16, device=dev) / 8
        ref = F.linear(x, w)
        flops = 2.0 * M * N * K
        t_blt = time_fn(lambda: F.linear(x, w), iters)
        row = {"shape": name, "blaslt_us": round(t_blt * 1e6, 1),
               "blaslt_tf": round(flops / t_blt / 1e12, 0)}
        for ver in (1, 2):
            for sk in (1, 2, 4, 8, 16):
                if K % (64 * sk) or (ver == 2 and N % 256):
                    continue
                try:
                    out = ops.skinny_gemm(x, w, splitk=sk, version=ver)
                    torch.cuda.synchronize()
                except Exception:  # noqa: BLE001
                    continue
                rel = (out.float() - ref.float()).abs().max().item()
                scale = ref.float().abs().max().item() + 1e-6
                if rel / scale > 2e-2:
                    row[f"v{ver}k{sk}"] = "ERR"
                    continue
                t = time_fn(lambda: ops.skinny_gemm(x, w, splitk=sk, version=ver), iters)
                row[f"v{ver}k{sk}_tf"] = round(flops / t / 1e12, 0)
        print(json.dumps(row), flush=True)


import sys as _sys  # noqa: E402
if "--skinny" in _sys.argv:
    skinny_sweep()
