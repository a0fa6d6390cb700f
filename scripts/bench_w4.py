import sys, time
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parents[1]))
import gpustack_amd.engine  # noqa: F401  (import-order: break package cycle)
import torch
import torch.nn.functional as F
from gpustack_amd import ops
from gpustack_amd.models.quantized import pack_w4_runtime

def t(fn, n=40):
    for _ in range(5): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e6

hip = ops._load_hip()
dev = "cuda"
for name, M, N, K in [("qkv", 512, 6144, 4096), ("o", 512, 4096, 4096),
                      ("gate_up", 512, 28672, 4096), ("down", 512, 4096, 14336),
                      ("lm_head", 512, 128256, 4096)]:
    torch.manual_seed(0)
    x = torch.randn(M, K, dtype=torch.bfloat16, device=dev) / 8
    wf = torch.randn(N, K, dtype=torch.bfloat16, device=dev) * 0.02
    # quantize
    wv = wf.float().view(N, K // 128, 128)
    mn, mx = wv.amin(-1), wv.amax(-1)
    s = ((mx - mn) / 15).clamp_min(1e-8)
    z = (-mn / s).round().clamp(0, 15)
    q = ((wv / s.unsqueeze(-1)) + z.unsqueeze(-1)).round().clamp(0, 15).view(N, K).to(torch.uint8)
    qw, sc, zs = pack_w4_runtime(q, s, z, 128)
    out = torch.empty(M, N, dtype=torch.bfloat16, device=dev)
    t_blt = t(lambda: F.linear(x, wf))
    t_w4 = t(lambda: hip.w4_gemm(out, x, qw, sc, zs))
    wt = torch.empty(N, K, dtype=torch.bfloat16, device=dev)
    t_dq = t(lambda: hip.w4_dequant(wt, qw, sc, zs))
    print(f"{name}: blaslt_bf16 {t_blt:.1f}us  w4_gemm {t_w4:.1f}us ({t_blt/t_w4:.2f}x)  dequant {t_dq:.1f}us", flush=True)
