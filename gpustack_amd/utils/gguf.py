"""First-party GGUF metadata reader (pure Python, no Go binary).

The reference shells out to the gguf-parser Go binary to read GGUF
metadata and estimate VRAM (reference: scheduler/calculator.py:553,1134,
worker/tools_manager.py:158-187). MI355X-native design: parse the GGUF
header directly in-process — the scheduler only needs architecture
hyper-parameters and exact per-tensor byte sizes, both of which live in
the GGUF v2/v3 header — and feed the same analytic memory model used for
safetensors models (scheduler/policies.py).

Execution of GGUF-quantized weights is NOT supported by the engine yet
(round 2); this module covers scheduling/placement parity: metadata,
weight-size, and ModelSpec derivation.
"""
from __future__ import annotations

import struct
from dataclasses import dataclass, field
from pathlib import Path

GGUF_MAGIC = b"GGUF"

# metadata value types (gguf spec)
_T_U8, _T_I8, _T_U16, _T_I16, _T_U32, _T_I32, _T_F32, _T_BOOL = range(8)
_T_STR, _T_ARR, _T_U64, _T_I64, _T_F64 = 8, 9, 10, 11, 12

_SCALAR_FMT = {
    _T_U8: "<B", _T_I8: "<b", _T_U16: "<H", _T_I16: "<h",
    _T_U32: "<I", _T_I32: "<i", _T_F32: "<f", _T_U64: "<Q",
    _T_I64: "<q", _T_F64: "<d",
}

# ggml tensor type -> (bytes per block, elements per block)
# Covers every type the llama/qwen GGUF zoo ships; unknown ids fall back
# to 1 byte/element (a safe over-estimate for sub-byte IQ variants is the
# 256-element K-block default below).
GGML_TYPE_SIZES: dict[int, tuple[int, int]] = {
    0: (4, 1),      # F32
    1: (2, 1),      # F16
    2: (18, 32),    # Q4_0
    3: (20, 32),    # Q4_1
    6: (22, 32),    # Q5_0
    7: (24, 32),    # Q5_1
    8: (34, 32),    # Q8_0
    9: (36, 32),    # Q8_1
    10: (84, 256),  # Q2_K
    11: (110, 256),  # Q3_K
    12: (144, 256),  # Q4_K
    13: (176, 256),  # Q5_K
    14: (210, 256),  # Q6_K
    15: (292, 256),  # Q8_K
    16: (66, 256),   # IQ2_XXS
    17: (74, 256),   # IQ2_XS
    18: (98, 256),   # IQ3_XXS
    19: (50, 256),   # IQ1_S
    20: (136, 256),  # IQ4_NL is 32-block (18,32) in ggml; conservative
    23: (110, 256),  # IQ3_S
    24: (1, 1),      # I8
    25: (2, 1),      # I16
    26: (4, 1),      # I32
    27: (8, 1),      # I64
    28: (8, 1),      # F64
    30: (2, 1),      # BF16
}


@dataclass
class GGUFTensorInfo:
    name: str
    shape: tuple[int, ...]
    ggml_type: int
    offset: int

    @property
    def nbytes(self) -> int:
        n = 1
        for d in self.shape:
            n *= d
        bs, epb = GGML_TYPE_SIZES.get(self.ggml_type, (1, 1))
        return (n + epb - 1) // epb * bs


@dataclass
class GGUFInfo:
    version: int
    metadata: dict = field(default_factory=dict)
    tensors: list[GGUFTensorInfo] = field(default_factory=list)

    @property
    def architecture(self) -> str:
        return self.metadata.get("general.architecture", "llama")

    @property
    def weight_bytes(self) -> int:
        return sum(t.nbytes for t in self.tensors)

    @property
    def n_params(self) -> int:
        total = 0
        for t in self.tensors:
            n = 1
            for d in t.shape:
                n *= d
            total += n
        return total

    def arch_key(self, key: str, default=None):
        return self.metadata.get(f"{self.architecture}.{key}", default)


class _Reader:
    def __init__(self, f):
        self.f = f

    def scalar(self, t: int):
        fmt = _SCALAR_FMT[t]
        return struct.unpack(fmt, self.f.read(struct.calcsize(fmt)))[0]

    def string(self) -> str:
        n = self.scalar(_T_U64)
        return self.f.read(n).decode("utf-8", errors="replace")

    def value(self, t: int):
        if t == _T_BOOL:
            return bool(self.f.read(1)[0])
        if t == _T_STR:
            return self.string()
        if t == _T_ARR:
            et = self.scalar(_T_U32)
            n = self.scalar(_T_U64)
            # token/merge lists can be 100k+ entries; keep them (cheap) but
            # a caller only after sizes may ignore them
            return [self.value(et) for _ in range(n)]
        return self.scalar(t)


def read_gguf(path: str | Path) -> GGUFInfo:
    """Parse a GGUF v2/v3 header: metadata KVs + tensor infos (no data)."""
    with open(path, "rb") as f:
        if f.read(4) != GGUF_MAGIC:
            raise ValueError(f"{path}: not a GGUF file")
        r = _Reader(f)
        version = r.scalar(_T_U32)
        if version not in (2, 3):
            raise ValueError(f"{path}: unsupported GGUF version {version}")
        n_tensors = r.scalar(_T_U64)
        n_kv = r.scalar(_T_U64)
        meta: dict = {}
        for _ in range(n_kv):
            key = r.string()
            vt = r.scalar(_T_U32)
            meta[key] = r.value(vt)
        tensors = []
        for _ in range(n_tensors):
            name = r.string()
            nd = r.scalar(_T_U32)
            dims = tuple(r.scalar(_T_U64) for _ in range(nd))
            ttype = r.scalar(_T_U32)
            off = r.scalar(_T_U64)
            tensors.append(GGUFTensorInfo(name, dims, ttype, off))
        return GGUFInfo(version=version, metadata=meta, tensors=tensors)


def spec_from_gguf(path: str | Path):
    """Derive a ModelSpec from GGUF metadata (scheduler estimation only)."""
    from ..engine.config import ModelSpec

    info = read_gguf(path)
    arch = info.architecture
    nh = int(info.arch_key("attention.head_count", 32))
    hidden = int(info.arch_key("embedding_length", 4096))
    vocab = info.metadata.get(f"{arch}.vocab_size")
    if vocab is None:
        toks = info.metadata.get("tokenizer.ggml.tokens")
        vocab = len(toks) if isinstance(toks, list) else 32000
    hf_arch = {"llama": "LlamaForCausalLM", "qwen2": "Qwen2ForCausalLM",
               "qwen3": "Qwen3ForCausalLM"}.get(arch, "LlamaForCausalLM")
    return ModelSpec(
        architecture=hf_arch,
        vocab_size=int(vocab),
        hidden_size=hidden,
        intermediate_size=int(info.arch_key("feed_forward_length", 11008)),
        num_layers=int(info.arch_key("block_count", 32)),
        num_heads=nh,
        num_kv_heads=int(info.arch_key("attention.head_count_kv", nh)),
        head_dim=int(info.arch_key("attention.key_length", hidden // nh)),
        rope_theta=float(info.arch_key("rope.freq_base", 10000.0)),
        rms_norm_eps=float(
            info.arch_key("attention.layer_norm_rms_epsilon", 1e-6)),
        max_position_embeddings=int(info.arch_key("context_length", 4096)),
        qk_norm=arch == "qwen3",
        attention_bias=arch == "qwen2",
    )


# ---- writer (tests + tooling) ---------------------------------------------

def write_gguf(path: str | Path, metadata: dict,
               tensors: list[tuple[str, tuple[int, ...], int]],
               version: int = 3) -> None:
    """Write a header-only GGUF (tensor data zero-filled). Test fixture
    generator — mirrors the reference's recorded gguf-parser fixtures
     without needing real model files."""
    def p_str(s: str) -> bytes:
        b = s.encode()
        return struct.pack("<Q", len(b)) + b

    def p_val(v) -> bytes:
        if isinstance(v, bool):
            return struct.pack("<I", _T_BOOL) + struct.pack("<B", int(v))
        if isinstance(v, int):
            return struct.pack("<I", _T_U64) + struct.pack("<Q", v)
        if isinstance(v, float):
            return struct.pack("<I", _T_F32) + struct.pack("<f", v)
        if isinstance(v, str):
            return struct.pack("<I", _T_STR) + p_str(v)
        if isinstance(v, list):  # string arrays only (token lists)
            out = struct.pack("<I", _T_ARR) + struct.pack("<I", _T_STR)
            out += struct.pack("<Q", len(v))
            for s in v:
                out += p_str(s)
            return out
        raise TypeError(type(v))

    buf = bytearray()
    buf += GGUF_MAGIC
    buf += struct.pack("<I", version)
    buf += struct.pack("<Q", len(tensors))
    buf += struct.pack("<Q", len(metadata))
    for k, v in metadata.items():
        buf += p_str(k) + p_val(v)
    offset = 0
    datasz = 0
    for name, shape, ttype in tensors:
        buf += p_str(name)
        buf += struct.pack("<I", len(shape))
        for d in shape:
            buf += struct.pack("<Q", d)
        buf += struct.pack("<I", ttype)
        buf += struct.pack("<Q", offset)
        ti = GGUFTensorInfo(name, shape, ttype, offset)
        sz = (ti.nbytes + 31) // 32 * 32  # default alignment
        offset += sz
        datasz += sz
    align = 32
    pad = (-len(buf)) % align
    buf += b"\x00" * pad
    with open(path, "wb") as f:
        f.write(buf)
        f.write(b"\x00" * datasz)
