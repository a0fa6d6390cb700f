"""Quantized-checkpoint loading: GPTQ / AWQ W4(8)A16 (reference: vLLM
--quantization gptq/awq selected through gpustack backend_parameters).

Design (same rationale as GGUF execution, utils/gguf.py): dequantize to
bf16 at LOAD time and serve on the stock MFMA bf16 path. On a 288 GB
HBM3E part the capacity argument for keeping weights packed at runtime is
weak for the model sizes a single GPU serves, while a fused
dequant-GEMM kernel costs a separate hot-path implementation — so W4
checkpoints pay a one-time load cost instead of a per-step one.

Formats (packing is little-endian within each int32):
  * GPTQ (AutoGPTQ v2): qweight int32 [in/pack, out] packed along IN,
    qzeros int32 [groups, out/pack] (+1 offset quirk), scales fp16
    [groups, out], optional g_idx int32 [in] (desc_act group order).
  * AWQ (AutoAWQ gemm): qweight int32 [in, out/pack] packed along OUT in
    the interleaved order (0,2,4,6,1,3,5,7), qzeros int32
    [groups, out/pack] same packing, scales fp16 [groups, out].

Both dequantize to w[out, in] (HF Linear layout).
"""
from __future__ import annotations

import torch

AWQ_ORDER = (0, 2, 4, 6, 1, 3, 5, 7)


def _unpack_int32(t: torch.Tensor, bits: int, dim: int) -> torch.Tensor:
    """Unpack int32 along `dim` into 32/bits unsigned values each
    (little-endian within the word)."""
    per = 32 // bits
    mask = (1 << bits) - 1
    t64 = t.to(torch.int64).movedim(dim, -1)           # [..., n]
    shifts = torch.arange(per, device=t.device, dtype=torch.int64) * bits
    u = (t64.unsqueeze(-1) >> shifts) & mask           # [..., n, per]
    u = u.reshape(*t64.shape[:-1], t64.shape[-1] * per)
    return u.movedim(-1, dim)


def dequant_gptq(qweight: torch.Tensor, qzeros: torch.Tensor,
                 scales: torch.Tensor, g_idx: torch.Tensor | None,
                 bits: int = 4) -> torch.Tensor:
    """-> w [out, in] float32."""
    per = 32 // bits
    q = _unpack_int32(qweight, bits, dim=0).float()        # [in, out]
    zeros = _unpack_int32(qzeros, bits, dim=1).float() + 1  # [groups, out]
    scales = scales.float()                                # [groups, out]
    n_in = q.shape[0]
    if g_idx is not None and g_idx.numel() == n_in:
        gi = g_idx.long()
    else:
        group = n_in // scales.shape[0]
        gi = torch.arange(n_in) // group
    w = (q - zeros[gi]) * scales[gi]                       # [in, out]
    return w.t().contiguous()


def dequant_awq(qweight: torch.Tensor, qzeros: torch.Tensor,
                scales: torch.Tensor, bits: int = 4) -> torch.Tensor:
    """-> w [out, in] float32."""
    per = 32 // bits
    order = torch.tensor(AWQ_ORDER[:per])
    inv = torch.argsort(order)

    def unpack_out(t):  # packed along the last (out) dim, interleaved
        u = _unpack_int32(t, bits, dim=t.dim() - 1)        # [..., out] seq
        u = u.reshape(*t.shape[:-1], t.shape[-1], per)[..., inv]
        return u.reshape(*t.shape[:-1], t.shape[-1] * per)

    q = unpack_out(qweight).float()                        # [in, out]
    zeros = unpack_out(qzeros).float()                     # [groups, out]
    scales = scales.float()
    group = q.shape[0] // scales.shape[0]
    gi = torch.arange(q.shape[0]) // group
    w = (q - zeros[gi]) * scales[gi]
    return w.t().contiguous()


def quant_config(model_dir) -> dict | None:
    """quantization_config from config.json (None = fp checkpoint)."""
    import json
    from pathlib import Path

    cfg_path = Path(model_dir) / "config.json"
    if not cfg_path.exists():
        return None
    with open(cfg_path) as f:
        qc = json.load(f).get("quantization_config")
    if not qc:
        return None
    method = (qc.get("quant_method") or "").lower()
    if method not in ("gptq", "awq"):
        raise NotImplementedError(f"quant_method {method!r} not supported "
                                  "(gptq/awq are)")
    return {"method": method, "bits": int(qc.get("bits", 4))}


def maybe_dequant(tensors: dict, name: str, qc: dict | None):
    """Resolve `<base>.weight`: plain tensor, or dequantized from the
    checkpoint's packed qweight/qzeros/scales[/g_idx] group."""
    if name in tensors:
        return tensors[name]
    if qc is None or not name.endswith(".weight"):
        raise KeyError(name)
    base = name[: -len(".weight")]
    if base + ".qweight" not in tensors:
        raise KeyError(name)
    qw = tensors[base + ".qweight"]
    qz = tensors[base + ".qzeros"]
    sc = tensors[base + ".scales"]
    if qc["method"] == "gptq":
        return dequant_gptq(qw, qz, sc, tensors.get(base + ".g_idx"),
                            qc["bits"])
    return dequant_awq(qw, qz, sc, qc["bits"])


# ---------------------------------------------------------------------------
# W4 RUNTIME format (round 2): weights stay packed int4 in HBM and the
# ops/csrc/w4_gemm.hip kernel dequantizes in-register. Canonical layout:
#   qw u8 [N, K/2] in FRAGMENT ORDER (see w4_gemm.hip header),
#   sc/zs bf16 [N, K/128] (one group per 128-k block; zs = zero * scale).
# ---------------------------------------------------------------------------

W4_BLOCK = 128


def _frag_perm(K: int, device=None) -> torch.Tensor:
    """k-index permutation: position p in packed order -> source k.

    Within each 128-k block, packed order is [lg(4)][chunk(4)][j(8)] while
    the source order is [chunk][lg][j] — this puts the 32 values of lane
    lg's four MFMA B-fragments into one contiguous 16-byte load."""
    arr = torch.arange(K, device=device).view(-1, 4, 4, 8)  # [blk, c, lg, j]
    return arr.permute(0, 2, 1, 3).reshape(-1)


def pack_w4_runtime(q: torch.Tensor, scales: torch.Tensor,
                    zeros: torch.Tensor, group_size: int):
    """q: [N, K] uint4 values (0..15); scales/zeros: [N, K/group_size].

    -> (qw u8 [N, K/2], sc bf16 [N, K/128], zs bf16 [N, K/128]) or None
    if the shape/grouping cannot run packed (caller falls back to
    dequant-at-load)."""
    N, K = q.shape
    if K % W4_BLOCK != 0:
        return None
    if group_size % W4_BLOCK != 0:
        return None  # sub-block groups: scale changes inside a k-block
    rep = group_size // W4_BLOCK
    sc_f = scales.float().repeat_interleave(rep, dim=1)   # [N, K/128]
    zr_f = zeros.float().repeat_interleave(rep, dim=1)
    perm = _frag_perm(K, q.device)
    qp = q.index_select(1, perm).to(torch.uint8)
    qw = (qp[:, 0::2] | (qp[:, 1::2] << 4)).contiguous()
    sc = sc_f.to(torch.bfloat16).contiguous()
    zs = (zr_f * sc_f).to(torch.bfloat16).contiguous()
    return qw, sc, zs


def dequant_w4_runtime(qw: torch.Tensor, sc: torch.Tensor,
                       zs: torch.Tensor) -> torch.Tensor:
    """Packed runtime format -> bf16 [N, K] (prefill transient / CPU ref)."""
    N = qw.shape[0]
    K = qw.shape[1] * 2
    lo = (qw & 0xF)
    hi = (qw >> 4)
    qp = torch.stack([lo, hi], dim=2).reshape(N, K)  # packed order
    perm = _frag_perm(K, qw.device)
    inv = torch.empty_like(perm)
    inv[perm] = torch.arange(K, device=qw.device)
    q = qp.index_select(1, inv).float()
    s = sc.float().repeat_interleave(W4_BLOCK, dim=1)
    z = zs.float().repeat_interleave(W4_BLOCK, dim=1)
    return (q * s - z).to(torch.bfloat16)
