"""First-party GGUF metadata reader (pure Python, no Go binary).

The reference shells out to the gguf-parser Go binary to read GGUF
metadata and estimate VRAM (reference: scheduler/calculator.py:553,1134,
worker/tools_manager.py:158-187). MI355X-native design: parse the GGUF
header directly in-process — the scheduler only needs architecture
hyper-parameters and exact per-tensor byte sizes, both of which live in
the GGUF v2/v3 header — and feed the same analytic memory model used for
safetensors models (scheduler/policies.py).

Execution: `read_tensor` dequantizes tensor data (F32/F16/BF16, Q8_0,
Q4_0, Q4_1, Q4_K, Q6_K — the types Q4_K_M/Q8_0 checkpoints ship) with
vectorized numpy at load time; models/weights.py:load_gguf maps the
llama.cpp tensor names onto the fused bf16 serving layout. Dequant is a
one-time load cost, so serving runs the same MFMA bf16 path as
safetensors checkpoints (fused dequant-GEMM is only worth it for
VRAM-constrained deployments, not this 288 GB part).
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
    data_start: int = 0  # file offset of the (aligned) tensor-data section

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
        align = int(meta.get("general.alignment", 32))
        data_start = (f.tell() + align - 1) // align * align
        return GGUFInfo(version=version, metadata=meta, tensors=tensors,
                        data_start=data_start)


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


# ---- tensor data: dequantization (execution path) ---------------------------
# Block layouts follow ggml's on-disk structs exactly; every dequant below is
# vectorized numpy over all blocks of a tensor at once (load-time cost only).

GGML_F32, GGML_F16, GGML_Q4_0, GGML_Q4_1 = 0, 1, 2, 3
GGML_Q8_0, GGML_Q4_K, GGML_Q6_K, GGML_BF16 = 8, 12, 14, 30


def _blocks(raw: "np.ndarray", n: int, bs: int, epb: int):
    import numpy as np

    nb = (n + epb - 1) // epb
    return np.frombuffer(raw, dtype=np.uint8, count=nb * bs).reshape(nb, bs)


def _deq_q8_0(raw, n):
    import numpy as np

    b = _blocks(raw, n, 34, 32)
    d = b[:, :2].copy().view(np.float16).astype(np.float32)      # [nb,1]
    qs = b[:, 2:].view(np.int8).astype(np.float32)               # [nb,32]
    return (d * qs).reshape(-1)[:n]


def _deq_q4_0(raw, n):
    import numpy as np

    b = _blocks(raw, n, 18, 32)
    d = b[:, :2].copy().view(np.float16).astype(np.float32)
    qs = b[:, 2:]
    lo = (qs & 0xF).astype(np.int8) - 8                          # elems 0..15
    hi = (qs >> 4).astype(np.int8) - 8                           # elems 16..31
    out = np.concatenate([lo, hi], axis=1).astype(np.float32) * d
    return out.reshape(-1)[:n]


def _deq_q4_1(raw, n):
    import numpy as np

    b = _blocks(raw, n, 20, 32)
    d = b[:, :2].copy().view(np.float16).astype(np.float32)
    m = b[:, 2:4].copy().view(np.float16).astype(np.float32)
    qs = b[:, 4:]
    lo = (qs & 0xF).astype(np.float32)
    hi = (qs >> 4).astype(np.float32)
    out = np.concatenate([lo, hi], axis=1) * d + m
    return out.reshape(-1)[:n]


def _q4k_scales(scales):
    """ggml get_scale_min_k4, vectorized: scales[nb,12] -> (sc, mn) [nb,8]."""
    import numpy as np

    q = scales.astype(np.uint16)
    sc = np.empty((q.shape[0], 8), dtype=np.float32)
    mn = np.empty((q.shape[0], 8), dtype=np.float32)
    for j in range(4):
        sc[:, j] = (q[:, j] & 63).astype(np.float32)
        mn[:, j] = (q[:, j + 4] & 63).astype(np.float32)
    for j in range(4, 8):
        sc[:, j] = ((q[:, j + 4] & 0xF) | ((q[:, j - 4] >> 6) << 4)).astype(np.float32)
        mn[:, j] = ((q[:, j + 4] >> 4) | ((q[:, j] >> 6) << 4)).astype(np.float32)
    return sc, mn


def _deq_q4_k(raw, n):
    import numpy as np

    b = _blocks(raw, n, 144, 256)
    nb = b.shape[0]
    d = b[:, 0:2].copy().view(np.float16).astype(np.float32).reshape(nb)
    dmin = b[:, 2:4].copy().view(np.float16).astype(np.float32).reshape(nb)
    sc, mn = _q4k_scales(b[:, 4:16])                      # [nb,8]
    qs = b[:, 16:144]                                     # [nb,128]
    out = np.empty((nb, 256), dtype=np.float32)
    # chunk j (0..3) of 64 elements: low nibbles of qs[32j:32j+32] are
    # sub-block 2j, high nibbles sub-block 2j+1
    for j in range(4):
        q = qs[:, 32 * j:32 * j + 32]
        lo = (q & 0xF).astype(np.float32)
        hi = (q >> 4).astype(np.float32)
        base = 64 * j
        out[:, base:base + 32] = (d * sc[:, 2 * j])[:, None] * lo \
            - (dmin * mn[:, 2 * j])[:, None]
        out[:, base + 32:base + 64] = (d * sc[:, 2 * j + 1])[:, None] * hi \
            - (dmin * mn[:, 2 * j + 1])[:, None]
    return out.reshape(-1)[:n]


def _deq_q6_k(raw, n):
    import numpy as np

    b = _blocks(raw, n, 210, 256)
    nb = b.shape[0]
    ql = b[:, 0:128]
    qh = b[:, 128:192]
    sc = b[:, 192:208].view(np.int8).astype(np.float32)   # [nb,16]
    d = b[:, 208:210].copy().view(np.float16).astype(np.float32).reshape(nb)
    out = np.empty((nb, 256), dtype=np.float32)
    for half in range(2):  # two 128-element chunks per superblock
        qlh = ql[:, 64 * half:64 * half + 64]
        qhh = qh[:, 32 * half:32 * half + 32]
        sch = sc[:, 8 * half:8 * half + 8]
        l = np.arange(32)
        is_ = l // 16                                     # [32] in {0,1}
        q1 = ((qlh[:, :32] & 0xF) | (((qhh >> 0) & 3) << 4)).astype(np.int16) - 32
        q2 = ((qlh[:, 32:] & 0xF) | (((qhh >> 2) & 3) << 4)).astype(np.int16) - 32
        q3 = ((qlh[:, :32] >> 4) | (((qhh >> 4) & 3) << 4)).astype(np.int16) - 32
        q4 = ((qlh[:, 32:] >> 4) | (((qhh >> 6) & 3) << 4)).astype(np.int16) - 32
        base = 128 * half
        out[:, base + 0:base + 32] = d[:, None] * sch[:, is_ + 0] * q1
        out[:, base + 32:base + 64] = d[:, None] * sch[:, is_ + 2] * q2
        out[:, base + 64:base + 96] = d[:, None] * sch[:, is_ + 4] * q3
        out[:, base + 96:base + 128] = d[:, None] * sch[:, is_ + 6] * q4
    return out.reshape(-1)[:n]


_DEQUANT = {
    GGML_Q8_0: _deq_q8_0,
    GGML_Q4_0: _deq_q4_0,
    GGML_Q4_1: _deq_q4_1,
    GGML_Q4_K: _deq_q4_k,
    GGML_Q6_K: _deq_q6_k,
}


def dequantize(raw: bytes, ggml_type: int, n: int):
    """Raw block data -> float32 numpy array of n elements."""
    import numpy as np

    if ggml_type == GGML_F32:
        return np.frombuffer(raw, dtype=np.float32, count=n).copy()
    if ggml_type == GGML_F16:
        return np.frombuffer(raw, dtype=np.float16, count=n).astype(np.float32)
    if ggml_type == GGML_BF16:
        u = np.frombuffer(raw, dtype=np.uint16, count=n).astype(np.uint32) << 16
        return u.view(np.float32).copy()
    fn = _DEQUANT.get(ggml_type)
    if fn is None:
        raise NotImplementedError(f"GGUF execution: ggml type {ggml_type} "
                                  "not supported (F32/F16/BF16/Q8_0/Q4_0/"
                                  "Q4_1/Q4_K/Q6_K are)")
    return fn(raw, n)


def read_tensor(path: str | Path, info: GGUFInfo, t: GGUFTensorInfo):
    """Dequantize one tensor to a float32 torch tensor in ROW-MAJOR [out, in]
    order (ggml dims are stored innermost-first, so the logical shape is
    reversed(t.shape))."""
    import torch

    with open(path, "rb") as f:
        f.seek(info.data_start + t.offset)
        raw = f.read(t.nbytes)
    n = 1
    for d in t.shape:
        n *= d
    arr = dequantize(raw, t.ggml_type, n)
    return torch.from_numpy(arr).reshape(tuple(reversed(t.shape)))


# ---- quantizers (tests + tooling; match ggml's reference quantize) ---------

def quantize_q8_0(x) -> bytes:
    """float array (multiple of 32) -> Q8_0 blocks."""
    import numpy as np

    x = np.asarray(x, dtype=np.float32).reshape(-1, 32)
    amax = np.abs(x).max(axis=1)
    d = (amax / 127.0).astype(np.float32)
    inv = np.where(d > 0, 1.0 / np.where(d == 0, 1, d), 0.0)
    qs = np.round(x * inv[:, None]).clip(-127, 127).astype(np.int8)
    out = bytearray()
    for i in range(x.shape[0]):
        out += np.float16(d[i]).tobytes() + qs[i].tobytes()
    return bytes(out)


def quantize_q4_0(x) -> bytes:
    import numpy as np

    x = np.asarray(x, dtype=np.float32).reshape(-1, 32)
    imax = np.abs(x).argmax(axis=1)
    maxv = x[np.arange(x.shape[0]), imax]       # signed max (ggml semantics)
    d = (maxv / -8.0).astype(np.float32)
    inv = np.where(d != 0, 1.0 / np.where(d == 0, 1, d), 0.0)
    q = (x * inv[:, None] + 8.5).clip(0, 15).astype(np.uint8)
    packed = (q[:, :16] | (q[:, 16:] << 4)).astype(np.uint8)
    out = bytearray()
    for i in range(x.shape[0]):
        out += np.float16(d[i]).tobytes() + packed[i].tobytes()
    return bytes(out)


def write_gguf_with_data(path: str | Path, metadata: dict,
                         tensors: "list[tuple[str, tuple[int, ...], int, bytes]]",
                         version: int = 3) -> None:
    """Write a complete GGUF file (header + aligned tensor data).

    `tensors` entries are (name, ggml_shape, ggml_type, block_data) with
    ggml dim order (innermost first). Used by tests and export tooling —
    pairs with read_tensor for round-trip verification."""
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
        if isinstance(v, list):
            out = struct.pack("<I", _T_ARR) + struct.pack("<I", _T_STR)
            out += struct.pack("<Q", len(v))
            for s in v:
                out += p_str(s)
            return out
        raise TypeError(type(v))

    align = 32
    buf = bytearray()
    buf += GGUF_MAGIC
    buf += struct.pack("<I", version)
    buf += struct.pack("<Q", len(tensors))
    buf += struct.pack("<Q", len(metadata))
    for k, v in metadata.items():
        buf += p_str(k) + p_val(v)
    offset = 0
    offsets = []
    for name, shape, ttype, data in tensors:
        buf += p_str(name)
        buf += struct.pack("<I", len(shape))
        for d in shape:
            buf += struct.pack("<Q", d)
        buf += struct.pack("<I", ttype)
        buf += struct.pack("<Q", offset)
        offsets.append(offset)
        offset += (len(data) + align - 1) // align * align
    buf += b"\x00" * ((-len(buf)) % align)
    with open(path, "wb") as f:
        f.write(buf)
        for (name, shape, ttype, data), off in zip(tensors, offsets):
            f.write(data)
            f.write(b"\x00" * ((-len(data)) % align))
