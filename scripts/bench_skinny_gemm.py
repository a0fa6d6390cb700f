"""Microbench: skinny_gemm vs hipBLASLt (F.linear) on decode shapes."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch
import torch.nn.functional as F

from gpustack_amd import ops

torch.backends.cuda.preferred_blas_library("cublaslt")

SHAPES = [  # (M, N, K) = decode projections of llama-3-8b
    (256, 6144, 4096), (256, 4096, 4096), (256, 28672, 4096), (256, 4096, 14336),
    (512, 6144, 4096), (512, 4096, 4096), (512, 28672, 4096), (512, 4096, 14336),
]


def time_fn(fn, iters=50):
    for _ in range(10):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(True)
    e = torch.cuda.Event(True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters * 1000  # us


print(f"{'shape':>22} {'blaslt us':>10} {'skinny us':>10} {'speedup':>8} {'GB/s(W)':>8}")
for M, N, K in SHAPES:
    x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda") * 0.1
    w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.1
    t_ref = time_fn(lambda: F.linear(x, w))
    t_sg = time_fn(lambda: ops.skinny_gemm(x, w))
    bw = N * K * 2 / (t_sg * 1e-6) / 1e9
    print(f"{(M,N,K)!s:>22} {t_ref:10.1f} {t_sg:10.1f} {t_ref/t_sg:8.2f} {bw:8.0f}")
