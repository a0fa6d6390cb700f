"""A/B the 8-phase GEMM vs hipBLASLt on its target shapes."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from gpustack_amd import ops

torch.backends.cuda.preferred_blas_library("cublaslt")
SHAPES = [("4096^3", 4096, 4096, 4096),
          ("8k.4k.4k", 8192, 4096, 4096),
          ("lmhd dec", 512, 128256, 4096),
          ("gate dec", 512, 28672, 4096),
          ("qkv dec", 512, 6144, 4096)]


def t(fn, n=30):
    for _ in range(8):
        fn()
    torch.cuda.synchronize()
    a = torch.cuda.Event(True); b = torch.cuda.Event(True)
    a.record()
    for _ in range(n):
        fn()
    b.record(); torch.cuda.synchronize()
    return a.elapsed_time(b) * 1000 / n


def main():
    for name, M, N, K in SHAPES:
        x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
        w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda")
        o1 = torch.empty(M, N, dtype=torch.bfloat16, device="cuda")
        u_blas = t(lambda: torch.mm(x, w.t(), out=o1))
        u_g8 = t(lambda: ops.gemm8(x, w))
        fl = 2 * M * N * K
        print(f"{name}: blaslt {u_blas:8.1f}us ({fl/u_blas/1e6:6.0f} TF)"
              f"  gemm8 {u_g8:8.1f}us ({fl/u_g8/1e6:6.0f} TF)"
              f"  ratio {u_blas/u_g8:4.2f}x")


if __name__ == "__main__":
    main()
