"""Microbench: skinny_gemm vs hipBLASLt (F.linear), splitk sweep."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch
import torch.nn.functional as F

from gpustack_amd import ops

torch.backends.cuda.preferred_blas_library("cublaslt")

SHAPES = [
    (512, 4096, 4096), (512, 6144, 4096), (512, 28672, 4096), (512, 4096, 14336),
    (4096, 4096, 4096),
]


def time_fn(fn, iters=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(True); e = torch.cuda.Event(True)
    s.record()
    for _ in range(iters):
        fn()
    e.record(); torch.cuda.synchronize()
    return s.elapsed_time(e) / iters * 1000


for M, N, K in SHAPES:
    x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda") * 0.1
    w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.1
    t_ref = time_fn(lambda: F.linear(x, w))
    line = f"({M},{N},{K}) blaslt={t_ref:7.1f}us"
    for sk in (1, 2, 4, 8):
        if K % (64 * sk):
            continue
        t = time_fn(lambda: ops.skinny_gemm(x, w, splitk=sk))
        tf = 2 * M * N * K / (t * 1e-6) / 1e12
        line += f" sk{sk}={t:6.1f}({tf:4.0f}TF)"
        if N % 256 == 0:
            t2 = time_fn(lambda: ops.skinny_gemm(x, w, splitk=sk, version=2))
            tf2 = 2 * M * N * K / (t2 * 1e-6) / 1e12
            line += f" v2:{t2:6.1f}({tf2:4.0f}TF)"
    print(line)
