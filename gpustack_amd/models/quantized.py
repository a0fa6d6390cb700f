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
