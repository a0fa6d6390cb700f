"""Plain-PyTorch fp32 reference implementations of every HIP op.

These are the numerics oracle for the GPU kernel tests (tests compare the
HIP kernels against these at fp32), and the CPU execution path that keeps
the whole serving stack testable without a GPU (the reference
tests everything host-only; we do the same for the control plane + engine
logic and add real kernel tests on-device).
"""
from __future__ import annotations

import torch


def rms_norm(out: torch.Tensor, x: torch.Tensor, weight: torch.Tensor, eps: float) -> None:
    xf = x.float()
    inv = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    out.copy_((xf * inv * weight.float()).to(out.dtype))


def fused_add_rms_norm(x: torch.Tensor, residual: torch.Tensor, weight: torch.Tensor, eps: float) -> None:
    summed = (x.float() + residual.float()).to(residual.dtype)
    residual.copy_(summed)
    sf = summed.float()
    inv = torch.rsqrt(sf.pow(2).mean(-1, keepdim=True) + eps)
    x.copy_((sf * inv * weight.float()).to(x.dtype))


def build_cos_sin_cache(
    head_dim: int,
    rot_dim: int,
    max_pos: int,
    base: float = 10000.0,
    scaling: dict | None = None,
    device: torch.device | str = "cpu",
) -> torch.Tensor:
    """[max_pos, rot_dim] f32: first rot_dim/2 cos then rot_dim/2 sin.

    `scaling` supports llama3-style rope scaling (keys: factor,
    low_freq_factor, high_freq_factor, original_max_position_embeddings).
    """
    half = rot_dim // 2
    inv_freq = 1.0 / (base ** (torch.arange(0, half, dtype=torch.float64) * 2 / rot_dim))
    attention_factor = 1.0
    if scaling and scaling.get("rope_type", scaling.get("type")) == "yarn":
        # YaRN (arXiv 2309.00071; HF _compute_yarn_parameters): blend
        # interpolated and extrapolated inverse frequencies over a linear
        # ramp between the beta_fast/beta_slow correction dims, and scale
        # cos/sin by the attention factor (mscale)
        import math

        factor = scaling.get("factor")
        orig = scaling["original_max_position_embeddings"]
        if factor is None:
            factor = max_pos / orig
        beta_fast = scaling.get("beta_fast") or 32
        beta_slow = scaling.get("beta_slow") or 1
        truncate = scaling.get("truncate", True)

        def get_mscale(scale, m=1):
            return 1.0 if scale <= 1 else 0.1 * m * math.log(scale) + 1.0

        attention_factor = scaling.get("attention_factor")
        if attention_factor is None:
            ms, msd = scaling.get("mscale"), scaling.get("mscale_all_dim")
            if ms and msd:
                attention_factor = float(get_mscale(factor, ms)
                                         / get_mscale(factor, msd))
            else:
                attention_factor = get_mscale(factor)

        def corr_dim(nrot):
            return (rot_dim * math.log(orig / (nrot * 2 * math.pi))
                    / (2 * math.log(base)))

        low, high = corr_dim(beta_fast), corr_dim(beta_slow)
        if truncate:
            low, high = math.floor(low), math.ceil(high)
        low, high = max(low, 0), min(high, rot_dim - 1)
        if low == high:
            high += 0.001
        ramp = ((torch.arange(half, dtype=torch.float64) - low)
                / (high - low)).clamp(0, 1)
        extrap_f = 1 - ramp
        inv_freq = (inv_freq / factor) * (1 - extrap_f) + inv_freq * extrap_f
    elif scaling and scaling.get("rope_type", scaling.get("type")) == "longrope":
        # LongRoPE (Phi-3/4; HF _compute_longrope_parameters): per-dim
        # frequency rescale factors, long vs short chosen by the cache's
        # target length vs the pretraining length — a single static cache
        # per engine (vLLM-style), not HF's per-forward dynamic switch
        import math

        orig = scaling["original_max_position_embeddings"]
        factors = (scaling["long_factor"] if max_pos > orig
                   else scaling["short_factor"])
        inv_freq = inv_freq / torch.tensor(factors, dtype=torch.float64)[:half]
        factor = scaling.get("factor") or (max_pos / orig)
        attention_factor = scaling.get("attention_factor")
        if attention_factor is None:
            attention_factor = (1.0 if factor <= 1.0 else
                                math.sqrt(1 + math.log(factor)
                                          / math.log(orig)))
    elif scaling and scaling.get("rope_type", scaling.get("type")) == "linear":
        # linear position interpolation (Gemma-3 global layers: factor 8)
        inv_freq = inv_freq / scaling["factor"]
    elif scaling and scaling.get("rope_type", scaling.get("type")) == "llama3":
        factor = scaling["factor"]
        lo = scaling["low_freq_factor"]
        hi = scaling["high_freq_factor"]
        orig = scaling["original_max_position_embeddings"]
        wavelen = 2 * torch.pi / inv_freq
        lo_wl = orig / lo
        hi_wl = orig / hi
        new = torch.where(wavelen > lo_wl, inv_freq / factor, inv_freq)
        smooth = (orig / wavelen - lo) / (hi - lo)
        mid = (1 - smooth) * inv_freq / factor + smooth * inv_freq
        is_mid = (wavelen <= lo_wl) & (wavelen >= hi_wl)
        inv_freq = torch.where(is_mid, mid, new)
    t = torch.arange(max_pos, dtype=torch.float64)
    freqs = torch.outer(t, inv_freq)
    cache = torch.cat([freqs.cos(), freqs.sin()], dim=-1).float()
    if attention_factor != 1.0:
        cache = cache * attention_factor
    return cache.to(device)


def rotary_embedding(
    positions: torch.Tensor,
    q: torch.Tensor,
    k: torch.Tensor,
    cos_sin: torch.Tensor,
    head_dim: int,
    rot_dim: int,
    mode: str = "neox",
) -> None:
    """In-place rotation. q: [T, Hq*D] or [T, Hq, D]; same for k.
    mode "neox" rotates the two halves (llama lineage); "pairwise"
    rotates adjacent (even, odd) pairs by one frequency each (GPT-J
    layout; Ernie-4.5 / Hunyuan / MiniMax checkpoints)."""
    half = rot_dim // 2
    cs = cos_sin[positions]  # [T, rot]
    cos = cs[:, :half].unsqueeze(1)  # [T, 1, half]
    sin = cs[:, half:].unsqueeze(1)
    for x in (q, k):
        T = x.shape[0]
        xs = x if x.dim() == 3 else x.view(T, -1, head_dim)
        if mode == "pairwise":
            x1 = xs[..., 0:rot_dim:2].float()
            x2 = xs[..., 1:rot_dim:2].float()
            o1 = x1 * cos - x2 * sin
            o2 = x2 * cos + x1 * sin
            xs[..., 0:rot_dim:2] = o1.to(x.dtype)
            xs[..., 1:rot_dim:2] = o2.to(x.dtype)
            continue
        x1 = xs[..., :half].float()
        x2 = xs[..., half : 2 * half].float()
        o1 = x1 * cos - x2 * sin
        o2 = x2 * cos + x1 * sin
        xs[..., :half] = o1.to(x.dtype)
        xs[..., half : 2 * half] = o2.to(x.dtype)


def silu_and_mul(out: torch.Tensor, x: torch.Tensor) -> None:
    d = out.shape[-1]
    xf = x.float()
    out.copy_((torch.nn.functional.silu(xf[..., :d]) * xf[..., d:]).to(out.dtype))


def reshape_and_cache(
    k: torch.Tensor,
    v: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    slots: torch.Tensor,
) -> None:
    """k/v: [T, Hkv, D]; caches: [B, Hkv, BS, D]; slots: [T] int64 (-1 skip)."""
    bs = k_cache.shape[2]
    mask = slots >= 0
    idx = slots[mask]
    blk = torch.div(idx, bs, rounding_mode="floor")
    off = idx % bs
    k_cache[blk, :, off] = k[mask].to(k_cache.dtype)
    v_cache[blk, :, off] = v[mask].to(v_cache.dtype)


def greedy_sample(out: torch.Tensor, logits: torch.Tensor) -> None:
    out.copy_(logits.float().argmax(dim=-1))


def _sink_softmax(att: torch.Tensor, sinks: torch.Tensor | None):
    """softmax over the kv axis with optional per-head SINK logits joining
    the denominator only (GPT-OSS: probability mass the sinks absorb is
    dropped — no value contribution). att: [Hq, Tq, L]."""
    if sinks is None:
        return torch.softmax(att, dim=-1)
    s = sinks.float().view(-1, 1, 1).expand(att.shape[0], att.shape[1], 1)
    combined = torch.cat([att, s], dim=-1)
    probs = torch.softmax(combined, dim=-1)
    return probs[..., :-1]


def _softcap(att: torch.Tensor, cap: float):
    """Gemma-2 attention logit softcapping: tanh(att/cap)*cap applied to
    the SCALED scores before the mask (HF eager semantics)."""
    if cap:
        return torch.tanh(att / cap) * cap
    return att


def _window_mask(qpos: torch.Tensor, kpos: torch.Tensor, window: int):
    """Causal (+ optional sliding-window) additive mask [Tq, L]."""
    m = torch.where(kpos.unsqueeze(0) <= qpos.unsqueeze(1), 0.0, float("-inf"))
    if window:
        m = torch.where(kpos.unsqueeze(0) > qpos.unsqueeze(1) - window,
                        m, torch.tensor(float("-inf")))
    return m


def paged_attn_decode(
    out: torch.Tensor,
    q: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,
    seq_lens: torch.Tensor,
    scale: float,
    sinks: torch.Tensor | None = None,
    window: int = 0,
    softcap: float = 0.0,
) -> None:
    """q/out: [N, Hq, D]; caches [B, Hkv, BS, D]."""
    N, Hq, D = q.shape
    Hkv = k_cache.shape[1]
    BS = k_cache.shape[2]
    GQ = Hq // Hkv
    for i in range(N):
        L = int(seq_lens[i])
        nblk = (L + BS - 1) // BS
        blocks = block_tables[i, :nblk].long()
        keys = k_cache[blocks].float().permute(1, 0, 2, 3).reshape(Hkv, -1, D)[:, :L]
        vals = v_cache[blocks].float().permute(1, 0, 2, 3).reshape(Hkv, -1, D)[:, :L]
        keys = keys.repeat_interleave(GQ, dim=0)  # [Hq, L, D]
        vals = vals.repeat_interleave(GQ, dim=0)
        qi = q[i].float().unsqueeze(1)  # [Hq, 1, D]
        att = _softcap((qi @ keys.transpose(1, 2)) * scale, softcap)
        att = att + _window_mask(torch.tensor([L - 1], device=q.device),
                                 torch.arange(L, device=q.device), window)
        att = _sink_softmax(att, sinks)
        out[i] = (att @ vals).squeeze(1).to(out.dtype)


def varlen_prefill_attn(
    out: torch.Tensor,
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    seq_lens: list[int],
    scale: float,
    sinks: torch.Tensor | None = None,
    window: int = 0,
    softcap: float = 0.0,
) -> None:
    """q: [T, Hq, D]; k/v: [T, Hkv, D]; causal within each sequence."""
    Hq = q.shape[1]
    Hkv = k.shape[1]
    GQ = Hq // Hkv
    start = 0
    for L in seq_lens:
        qs = q[start : start + L].float().permute(1, 0, 2)  # [Hq, L, D]
        ks = k[start : start + L].float().permute(1, 0, 2).repeat_interleave(GQ, dim=0)
        vs = v[start : start + L].float().permute(1, 0, 2).repeat_interleave(GQ, dim=0)
        att = _softcap((qs @ ks.transpose(1, 2)) * scale, softcap)
        pos = torch.arange(L, device=q.device)
        att = att + _window_mask(pos, pos, window)
        att = _sink_softmax(att, sinks)
        out[start : start + L] = (att @ vs).permute(1, 0, 2).to(out.dtype)
        start += L


def paged_prefill_attn(
    out: torch.Tensor,
    q: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,
    seq_starts: list[int],
    seq_hists: list[int],
    seq_news: list[int],
    scale: float,
    sinks: torch.Tensor | None = None,
    window: int = 0,
    softcap: float = 0.0,
) -> None:
    """Prefill-with-history: q/out hold only the NEW (suffix) rows of each
    sequence; K/V for positions [0, hist+new) are gathered from the paged
    pool. Row r of sequence i attends causally to positions [0, hist+r].

    q/out: [T, Hq, D]; caches [B, Hkv, BS, D]; block_tables [nseq, maxb].
    """
    Hq = q.shape[1]
    D = q.shape[2]
    Hkv = k_cache.shape[1]
    BS = k_cache.shape[2]
    GQ = Hq // Hkv
    row = 0
    for i, (hist, new) in enumerate(zip(seq_hists, seq_news)):
        L = hist + new
        nblk = (L + BS - 1) // BS
        blocks = block_tables[i, :nblk].long()
        keys = k_cache[blocks].float().permute(1, 0, 2, 3).reshape(Hkv, -1, D)[:, :L]
        vals = v_cache[blocks].float().permute(1, 0, 2, 3).reshape(Hkv, -1, D)[:, :L]
        keys = keys.repeat_interleave(GQ, dim=0)  # [Hq, L, D]
        vals = vals.repeat_interleave(GQ, dim=0)
        start = seq_starts[i]
        qs = q[start:start + new].float().permute(1, 0, 2)  # [Hq, new, D]
        att = _softcap((qs @ keys.transpose(1, 2)) * scale,
                       softcap)                             # [Hq, new, L]
        pos = torch.arange(L, device=q.device)
        qpos = hist + torch.arange(new, device=q.device)
        att = att + _window_mask(qpos, pos, window)
        att = _sink_softmax(att, sinks)
        out[start:start + new] = (att @ vals).permute(1, 0, 2).to(out.dtype)
        row += new
