"""gpustack_amd.ops — dispatch layer for the CDNA4 HIP kernels.

On a GPU box the in-tree `_hip_ops.so` (hand-written gfx950 kernels) is
REQUIRED: there is no silent eager fallback — a missing extension raises at
first use so a broken build can't masquerade as the native path. On CPU-only
hosts the fp32 torch reference implementations run instead, which keeps the
engine/control-plane logic testable without hardware.
"""
from __future__ import annotations

import os

import torch

from . import torch_ref

_HIP = None
_HIP_ERR: str | None = None


def _load_hip():
    global _HIP, _HIP_ERR
    if _HIP is not None or _HIP_ERR is not None:
        return _HIP
    try:
        import importlib.util
        from pathlib import Path

        so = Path(__file__).resolve().parent / "_hip_ops.so"
        if not so.exists():
            raise ImportError(
                f"{so} not found — build it with `python -m gpustack_amd.ops.build` "
                "(hipcc --offload-arch=gfx950)"
            )
        spec = importlib.util.spec_from_file_location("gpustack_amd.ops._hip_ops", so)
        mod = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(mod)
        _HIP = mod
    except Exception as e:  # noqa: BLE001
        _HIP_ERR = str(e)
        raise
    return _HIP


def hip_available() -> bool:
    if not torch.cuda.is_available():
        return False
    try:
        return _load_hip() is not None
    except Exception:  # noqa: BLE001
        return False


def _backend(t: torch.Tensor):
    if t.is_cuda:
        return _load_hip()  # raises loudly if the native build is missing
    return None


def _oss_kernels_enabled() -> bool:
    """GPT-OSS CDNA4 kernel variants (head_dim-64 + attention sinks +
    sliding window in attn_decode/attn_prefill.hip) are written and
    compile-checked but not yet GPU-validated — opt in explicitly until
    the r3 numerics pass (tests/test_ops_gpu.py oss tests)."""
    return os.environ.get("GPUSTACK_AMD_OSS_KERNELS", "0") == "1"


def _mla_kernel_enabled() -> bool:
    return os.environ.get("GPUSTACK_AMD_MLA_KERNEL", "0") == "1"


def _sinks_f32(sinks):
    if sinks is not None and (sinks.dtype != torch.float32
                              or not sinks.is_contiguous()):
        sinks = sinks.float().contiguous()
    return sinks


# --- op surface -----------------------------------------------------------

def rms_norm(out, x, weight, eps: float) -> None:
    hip = _backend(x)
    if hip is not None:
        hip.rms_norm(out, x, weight, eps)
    else:
        torch_ref.rms_norm(out, x, weight, eps)


def fused_add_rms_norm(x, residual, weight, eps: float) -> None:
    hip = _backend(x)
    if hip is not None:
        hip.fused_add_rms_norm(x, residual, weight, eps)
    else:
        torch_ref.fused_add_rms_norm(x, residual, weight, eps)


def rotary_embedding(positions, q, k, cos_sin, head_dim: int, rot_dim: int,
                     mode: str = "neox") -> None:
    hip = _backend(q)
    if hip is not None:
        if mode != "neox":
            raise NotImplementedError(
                "pairwise (GPT-J style) rotary is not in the CDNA4 kernel "
                "yet (Ernie/Hunyuan/MiniMax GPU serving lands with it in "
                "r3); CPU serving is available")
        hip.rotary_embedding(positions, q, k, cos_sin, head_dim, rot_dim)
    else:
        torch_ref.rotary_embedding(positions, q, k, cos_sin, head_dim,
                                   rot_dim, mode=mode)


def silu_and_mul(out, x) -> None:
    hip = _backend(x)
    if hip is not None:
        hip.silu_and_mul(out, x)
    else:
        torch_ref.silu_and_mul(out, x)


def reshape_and_cache(k, v, k_cache, v_cache, slots) -> None:
    hip = _backend(k)
    if hip is not None:
        hip.reshape_and_cache(k, v, k_cache, v_cache, slots)
    else:
        torch_ref.reshape_and_cache(k, v, k_cache, v_cache, slots)


def greedy_sample_into(out, logits) -> None:
    hip = _backend(logits)
    if hip is not None:
        hip.greedy_sample(out, logits)
    else:
        torch_ref.greedy_sample(out, logits)


def greedy_sample(logits) -> torch.Tensor:
    out = torch.empty(logits.shape[0], dtype=torch.long, device=logits.device)
    hip = _backend(logits)
    if hip is not None:
        hip.greedy_sample(out, logits)
    else:
        torch_ref.greedy_sample(out, logits)
    return out


def paged_attn_decode(out, q, k_cache, v_cache, block_tables, seq_lens,
                      scale: float, sinks=None, window: int = 0,
                      softcap: float = 0.0) -> None:
    hip = _backend(q)
    if hip is not None:
        if (sinks is not None or window or softcap or q.shape[-1] != 128) \
                and not _oss_kernels_enabled():
            # the sinks/window/softcap and D!=128 kernel variants are
            # written but not yet GPU-validated — fail loudly rather than
            # silently mis-attend (opt in with GPUSTACK_AMD_OSS_KERNELS=1;
            # r3 flips the default after the validation pass)
            raise NotImplementedError(
                "sinks/window/softcap/D!=128 CDNA4 decode kernel variants "
                "are unvalidated — set GPUSTACK_AMD_OSS_KERNELS=1 to opt in")
        hip.paged_attn_decode(out, q, k_cache, v_cache, block_tables, seq_lens,
                              scale, sinks=_sinks_f32(sinks), window=window,
                              softcap=softcap)
    else:
        torch_ref.paged_attn_decode(out, q, k_cache, v_cache, block_tables,
                                    seq_lens, scale, sinks=sinks,
                                    window=window, softcap=softcap)


def mla_decode(ctx_out, q_cat, lat_cache, block_tables, seq_lens,
               scale: float) -> None:
    """MLA absorbed decode (DeepSeek): ctx_out[N,H,512] f32 = softmax
    (q_cat · lat) · lat[:, :512] over the paged latent pool. GPU-only —
    the CPU oracle lives in models/llama.py MLAAttention."""
    hip = _backend(q_cat)
    if hip is None:
        raise NotImplementedError("mla_decode is the CDNA4 kernel entry; "
                                  "CPU serving uses the torch MLA path")
    hip.mla_decode(ctx_out, q_cat, lat_cache, block_tables, seq_lens, scale)


_PREFILL_BQ = 64


def build_prefill_tiles(seq_lens: list[int], device) -> tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Map varlen sequences to fixed 64-row q tiles for the MFMA kernel."""
    starts, q0s, lens = [], [], []
    tok = 0
    for L in seq_lens:
        for q0 in range(0, L, _PREFILL_BQ):
            starts.append(tok)
            q0s.append(q0)
            lens.append(L)
        tok += L
    mk = lambda a: torch.tensor(a, dtype=torch.int32, device=device)
    return mk(starts), mk(q0s), mk(lens)


def varlen_prefill_attn(out, q, k, v, seq_lens: list[int], scale: float,
                        tiles=None, sinks=None, window: int = 0,
                        softcap: float = 0.0) -> None:
    hip = _backend(q)
    if hip is not None:
        d_ok = (q.shape[-1] == 128
                or (q.shape[-1] == 192 and _mla_kernel_enabled()))
        if (sinks is not None or window or softcap or not d_ok) \
                and not _oss_kernels_enabled():
            raise NotImplementedError(
                "sinks/window/softcap/D!=128 CDNA4 prefill kernel variants "
                "are unvalidated — set GPUSTACK_AMD_OSS_KERNELS=1 (or "
                "GPUSTACK_AMD_MLA_KERNEL=1 for the 192/128 MLA expand "
                "shape) to opt in")
        if tiles is None or tiles[0] is None:
            tiles = build_prefill_tiles(seq_lens, q.device)
        hip.flash_prefill(out, q, k, v, tiles[0], tiles[1], tiles[2], scale,
                          sinks=_sinks_f32(sinks), window=window,
                          softcap=softcap)
    else:
        torch_ref.varlen_prefill_attn(out, q, k, v, seq_lens, scale,
                                      sinks=sinks, window=window,
                                      softcap=softcap)


def build_paged_prefill_tiles(seq_starts: list[int], seq_hists: list[int],
                              seq_news: list[int], device):
    """Map per-seq suffix rows to fixed 64-row q tiles for the paged
    prefill-with-history kernel (one entry set per (seq, q-tile))."""
    qstart, q0s, hists, news, seqi = [], [], [], [], []
    for i, (st, h, n) in enumerate(zip(seq_starts, seq_hists, seq_news)):
        for q0 in range(0, n, _PREFILL_BQ):
            qstart.append(st)
            q0s.append(q0)
            hists.append(h)
            news.append(n)
            seqi.append(i)
    mk = lambda a: torch.tensor(a, dtype=torch.int32, device=device)
    return mk(qstart), mk(q0s), mk(hists), mk(news), mk(seqi)


def paged_prefill_attn(out, q, k_cache, v_cache, block_tables,
                       seq_starts: list[int], seq_hists: list[int],
                       seq_news: list[int], scale: float, tiles=None,
                       sinks=None, window: int = 0,
                       softcap: float = 0.0) -> None:
    """Prefill-with-history: suffix/chunk rows attend to cached paged KV +
    their own freshly written positions through the MFMA flash kernel
    (removes the r1-measured ~3.5x paged-decode-row penalty)."""
    hip = _backend(q)
    if hip is not None:
        if (sinks is not None or window or softcap or q.shape[-1] != 128) \
                and not _oss_kernels_enabled():
            raise NotImplementedError(
                "sinks/window/softcap/D!=128 CDNA4 paged-prefill kernel "
                "variants are unvalidated — set GPUSTACK_AMD_OSS_KERNELS=1 "
                "to opt in")
        if tiles is None:
            tiles = build_paged_prefill_tiles(seq_starts, seq_hists,
                                              seq_news, q.device)
        hip.flash_prefill_paged(out, q, k_cache, v_cache, block_tables,
                                tiles[0], tiles[1], tiles[2], tiles[3],
                                tiles[4], scale, sinks=_sinks_f32(sinks),
                                window=window, softcap=softcap)
    else:
        torch_ref.paged_prefill_attn(out, q, k_cache, v_cache, block_tables,
                                     seq_starts, seq_hists, seq_news, scale,
                                     sinks=sinks, window=window,
                                     softcap=softcap)


_SG_WORKSPACES: dict = {}


def gemm8(x, w, out=None, safe: bool = False):
    """8-phase pipelined 256x256 MFMA GEMM: out[M,N] = x[M,K] @ w[N,K]^T
    (bf16, f32 accumulate). GPU-only; requires N%256==0, K%128==0 —
    falls back to F.linear otherwise. `safe=True` runs the fully-drained
    variant (race isolation for tests)."""
    import torch.nn.functional as F

    M, K = x.shape
    N = w.shape[0]
    if not x.is_cuda or N % 256 != 0 or K % 128 != 0:
        return F.linear(x, w, None)
    if out is None:
        out = torch.empty(M, N, dtype=x.dtype, device=x.device)
    hip = _backend(x)
    if hip is None:
        return F.linear(x, w, None)
    hip.gemm8(out, x.contiguous(), w, 1 if safe else 0)
    return out


def skinny_gemm(x, w, out=None, splitk: int | None = None, version: int = 1):
    """Split-K decode GEMM: out[M,N] = x[M,K] @ w[N,K]^T (bf16, f32 acc).
    GPU-only (falls back to F.linear elsewhere or for unsupported shapes)."""
    import torch.nn.functional as F

    M, K = x.shape
    N = w.shape[0]
    if not x.is_cuda or N % 128 != 0 or K % 64 != 0:
        return F.linear(x, w, None)
    if splitk is None:
        mt = (M + 255) // 256
        nt = N // 128
        splitk = 1
        while (mt * nt * splitk < 400 and splitk < 8
               and K % (64 * splitk * 2) == 0):
            splitk *= 2
    if out is None:
        out = torch.empty(M, N, dtype=x.dtype, device=x.device)
    ws = None
    if splitk > 1:
        key = (splitk, M, N, x.device.index)
        ws = _SG_WORKSPACES.get(key)
        if ws is None:
            ws = torch.empty(splitk * M * N, dtype=torch.float32, device=x.device)
            _SG_WORKSPACES[key] = ws
    _load_hip().skinny_gemm(out, x, w, ws, splitk, version)
    return out


def linear_auto(x, w):
    """Shape-gated GEMM dispatch: the split-K skinny kernel where it
    measured faster than hipBLASLt (scripts/bench_skinny_gemm.py on
    MI355X: down_proj-like N=4096 K>=8192 M>=384 at splitk=4, 824 TF vs
    hipBLASLt 680), F.linear everywhere else."""
    import os

    import torch.nn.functional as F

    if (x.is_cuda and x.dim() == 2
            and os.environ.get("GPUSTACK_AMD_SKINNY_GEMM", "1") == "1"):
        M, K = x.shape
        N = w.shape[0]
        if (N == 4096 and K >= 8192 and 384 <= M <= 1024
                and K % 256 == 0 and x.stride(1) == 1):
            return skinny_gemm(x, w, splitk=4)
    return F.linear(x, w)


def mfma_probe(a, b):
    return _load_hip().mfma_probe(a, b)


build_cos_sin_cache = torch_ref.build_cos_sin_cache
