"""MLA absorbed-decode kernel bench + numerics harness (r3, GPU box).

Usage (needs GPUSTACK_AMD_MLA_KERNEL=1 for the engine path; this script
drives the kernel directly so the env gate does not apply):

    gpurun -- 'python scripts/bench_mla.py > gpurun_out/mla.txt 2>&1'

Measures ops.mla_decode against the torch absorbed reference at
DeepSeek-V3 decode shapes (H q-heads sharing one 576-wide latent
stream), sweeping batch and context. Reports us/call and effective
latent-read bandwidth (the kernel's roofline: each (seq, 16-head block)
reads L*576*2 bytes, so traffic = N * ceil(H/16) * L * 1152 B).
"""
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch  # noqa: E402

from gpustack_amd import ops  # noqa: E402


def ref(q, lat, bt, lens, scale, R):
    BS = lat.shape[2]
    flat = lat.view(-1, lat.shape[-1]).float()
    outs = []
    for i, L in enumerate(lens):
        idx = (bt[i][torch.arange(L, device=q.device) // BS].long() * BS
               + torch.arange(L, device=q.device) % BS)
        C = flat[idx]
        probs = torch.softmax((q[i].float() @ C.T) * scale, dim=-1)
        outs.append(probs @ C[:, :R])
    return torch.stack(outs)


def main():
    assert torch.cuda.is_available(), "GPU box required"
    R, DR, BS, H = 512, 64, 16, 128
    LD = R + DR
    scale = 1.0 / (192 ** 0.5)
    torch.manual_seed(0)
    for N, L in [(1, 512), (16, 512), (64, 1024), (256, 1024), (256, 4096)]:
        lens = [L] * N
        nblocks = N * ((L + BS - 1) // BS) + 1
        lat = torch.randn(nblocks, 1, BS, LD, dtype=torch.bfloat16,
                          device="cuda") / 4
        bt = torch.arange(nblocks - 1, dtype=torch.int32,
                          device="cuda").view(N, -1)
        q = torch.randn(N, H, LD, dtype=torch.bfloat16, device="cuda") / 4
        sl = torch.tensor(lens, dtype=torch.int32, device="cuda")
        ctx = torch.empty(N, H, R, dtype=torch.float32, device="cuda")
        ops.mla_decode(ctx, q, lat, bt, sl, scale)
        want = ref(q, lat, bt, lens, scale, R)
        err = (ctx - want).abs().max().item()
        # timing
        for _ in range(3):
            ops.mla_decode(ctx, q, lat, bt, sl, scale)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        iters = 20
        for _ in range(iters):
            ops.mla_decode(ctx, q, lat, bt, sl, scale)
        torch.cuda.synchronize()
        us = (time.perf_counter() - t0) / iters * 1e6
        bytes_read = N * (H // 16) * L * LD * 2
        print(f"N={N:4d} L={L:5d}: {us:9.1f} us  "
              f"latent-traffic {bytes_read / us / 1e3:8.1f} GB/s  "
              f"max_err {err:.4f}")


if __name__ == "__main__":
    main()
