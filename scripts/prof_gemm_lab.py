#!/usr/bin/env python3
"""Minimal driver for rocprofv3 PMC collection on the GEMM lab kernels:
runs one variant on one shape in a tight loop so counter rows are clean.

    rocprofv3 --pmc SQ_LDS_BANK_CONFLICT SQ_INSTS_MFMA SQ_WAIT_ANY SQ_BUSY_CYCLES \
        -d gpurun_out/pmc -- python scripts/prof_gemm_lab.py --mode 0 --shape square4k
"""
from __future__ import annotations

import argparse
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import torch
import torch.nn.functional as F

from gpustack_amd import ops

SHAPES = {
    "qkv_dec": (512, 6144, 4096),
    "gate_up_dec": (512, 28672, 4096),
    "lm_head_dec": (512, 128256, 4096),
    "square4k": (4096, 4096, 4096),
}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--mode", type=int, default=0, help="-1 = hipBLASLt")
    ap.add_argument("--shape", default="square4k")
    ap.add_argument("--iters", type=int, default=20)
    args = ap.parse_args()
    M, N, K = SHAPES[args.shape]
    dev = "cuda:0"
    torch.manual_seed(0)
    x = torch.randn(M, K, dtype=torch.bfloat16, device=dev) / 8
    w = torch.randn(N, K, dtype=torch.bfloat16, device=dev) / 8
    out = torch.empty(M, N, dtype=torch.bfloat16, device=dev)
    hip = ops._load_hip()
    for _ in range(args.iters):
        if args.mode < 0:
            F.linear(x, w)
        else:
            hip.gemm_lab(out, x, w, args.mode)
    torch.cuda.synchronize()
    print("done", args.shape, "mode", args.mode)


if __name__ == "__main__":
    main()
