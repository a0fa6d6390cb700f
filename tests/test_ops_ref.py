"""CPU sanity tests for the torch reference ops (the GPU-kernel oracle).

Mirrors the reference's host-only test strategy for the parts
that can run here; the same semantics are re-checked against the HIP kernels
on-device in test_ops_gpu.py.
"""
import math

import pytest
import torch

from gpustack_amd.ops import torch_ref as R

torch.manual_seed(0)


def test_rms_norm_matches_formula():
    x = torch.randn(4, 64, dtype=torch.bfloat16)
    w = torch.randn(64, dtype=torch.bfloat16)
    out = torch.empty_like(x)
    R.rms_norm(out, x, w, 1e-6)
    xf = x.float()
    expect = xf / (xf.pow(2).mean(-1, keepdim=True) + 1e-6).sqrt() * w.float()
    assert torch.allclose(out.float(), expect, atol=2e-2, rtol=2e-2)


def test_fused_add_rms_norm_updates_both():
    x = torch.randn(4, 64, dtype=torch.bfloat16)
    res = torch.randn(4, 64, dtype=torch.bfloat16)
    w = torch.randn(64, dtype=torch.bfloat16)
    x0, r0 = x.clone(), res.clone()
    R.fused_add_rms_norm(x, res, w, 1e-6)
    assert torch.allclose(res.float(), (x0.float() + r0.float()), atol=2e-2)
    out = torch.empty_like(x)
    R.rms_norm(out, res, w, 1e-6)
    assert torch.allclose(x.float(), out.float(), atol=2e-2)


def test_rope_matches_hf_rotate_half():
    # neox-style: rotate_half as in HF transformers Llama
    T, H, D = 5, 3, 32
    q = torch.randn(T, H, D, dtype=torch.bfloat16)
    k = torch.randn(T, 2, D, dtype=torch.bfloat16)
    pos = torch.tensor([0, 1, 5, 9, 2])
    cache = R.build_cos_sin_cache(D, D, 16)
    q2, k2 = q.clone(), k.clone()
    R.rotary_embedding(pos, q2, k2, cache, D, D)

    inv = 1.0 / (10000.0 ** (torch.arange(0, D, 2).float() / D))
    freqs = torch.outer(pos.float(), inv)
    emb = torch.cat([freqs, freqs], dim=-1)
    cos, sin = emb.cos()[:, None, :], emb.sin()[:, None, :]

    def rotate_half(x):
        x1, x2 = x[..., : D // 2], x[..., D // 2 :]
        return torch.cat([-x2, x1], dim=-1)

    qe = q.float() * cos + rotate_half(q.float()) * sin
    assert torch.allclose(q2.float(), qe, atol=2e-2, rtol=2e-2)


def test_silu_and_mul():
    x = torch.randn(3, 32, dtype=torch.bfloat16)
    out = torch.empty(3, 16, dtype=torch.bfloat16)
    R.silu_and_mul(out, x)
    e = torch.nn.functional.silu(x.float()[:, :16]) * x.float()[:, 16:]
    assert torch.allclose(out.float(), e, atol=2e-2, rtol=2e-2)


def test_reshape_and_cache_roundtrip():
    T, Hkv, D, BS, B = 7, 2, 16, 16, 4
    k = torch.randn(T, Hkv, D, dtype=torch.bfloat16)
    v = torch.randn(T, Hkv, D, dtype=torch.bfloat16)
    kc = torch.zeros(B, Hkv, BS, D, dtype=torch.bfloat16)
    vc = torch.zeros_like(kc)
    slots = torch.tensor([0, 1, 17, 18, 35, -1, 63])
    R.reshape_and_cache(k, v, kc, vc, slots)
    assert torch.equal(kc[0, :, 0], k[0])
    assert torch.equal(kc[1, :, 1], k[2])
    assert torch.equal(vc[2, :, 3], v[4])
    assert torch.equal(kc[3, :, 15], k[6])
    assert kc[0, :, 5].abs().sum() == 0  # skipped slot wrote nothing


def test_paged_decode_matches_dense_attention():
    torch.manual_seed(1)
    N, Hq, Hkv, D, BS = 3, 8, 2, 64, 16
    lens = [5, 16, 33]
    nblocks = 12
    kc = torch.randn(nblocks, Hkv, BS, D, dtype=torch.bfloat16)
    vc = torch.randn(nblocks, Hkv, BS, D, dtype=torch.bfloat16)
    bt = torch.tensor([[0, 1, 2], [3, 4, 5], [6, 7, 8]], dtype=torch.int32)
    q = torch.randn(N, Hq, D, dtype=torch.bfloat16)
    out = torch.empty_like(q)
    scale = 1 / math.sqrt(D)
    R.paged_attn_decode(out, q, kc, vc, bt, torch.tensor(lens, dtype=torch.int32), scale)

    # dense check for seq 2 (crosses block boundary)
    i = 2
    L = lens[i]
    blocks = bt[i, : (L + BS - 1) // BS].long()
    keys = kc[blocks].permute(1, 0, 2, 3).reshape(Hkv, -1, D)[:, :L].float()
    vals = vc[blocks].permute(1, 0, 2, 3).reshape(Hkv, -1, D)[:, :L].float()
    for h in range(Hq):
        kh = keys[h // (Hq // Hkv)]
        vh = vals[h // (Hq // Hkv)]
        att = torch.softmax((q[i, h].float() @ kh.T) * scale, dim=-1)
        expect = att @ vh
        assert torch.allclose(out[i, h].float(), expect, atol=3e-2, rtol=3e-2)


def test_varlen_prefill_matches_sdpa():
    torch.manual_seed(2)
    Hq, Hkv, D = 4, 2, 32
    lens = [3, 9]
    T = sum(lens)
    q = torch.randn(T, Hq, D, dtype=torch.bfloat16)
    k = torch.randn(T, Hkv, D, dtype=torch.bfloat16)
    v = torch.randn(T, Hkv, D, dtype=torch.bfloat16)
    out = torch.empty_like(q)
    scale = 1 / math.sqrt(D)
    R.varlen_prefill_attn(out, q, k, v, lens, scale)
    start = 0
    for L in lens:
        qs = q[start : start + L].float().permute(1, 0, 2)
        ks = k[start : start + L].float().permute(1, 0, 2).repeat_interleave(Hq // Hkv, 0)
        vs = v[start : start + L].float().permute(1, 0, 2).repeat_interleave(Hq // Hkv, 0)
        e = torch.nn.functional.scaled_dot_product_attention(
            qs, ks, vs, is_causal=True, scale=scale
        )
        assert torch.allclose(out[start : start + L].float(), e.permute(1, 0, 2), atol=3e-2, rtol=3e-2)
        start += L


def test_greedy_sample():
    logits = torch.randn(4, 100, dtype=torch.bfloat16)
    out = torch.empty(4, dtype=torch.long)
    R.greedy_sample(out, logits)
    assert torch.equal(out, logits.float().argmax(-1))


def test_build_prefill_tiles():
    from gpustack_amd.ops import build_prefill_tiles

    ts, tq, tl = build_prefill_tiles([70, 64, 10], "cpu")
    assert ts.tolist() == [0, 0, 70, 134]
    assert tq.tolist() == [0, 64, 0, 0]
    assert tl.tolist() == [70, 70, 64, 10]


def test_yarn_rope_matches_hf():
    """YaRN rope scaling (arXiv 2309.00071) cache == HF transformers'
    _compute_yarn_parameters for the same config."""
    import torch
    from transformers import Qwen2Config
    from transformers.modeling_rope_utils import ROPE_INIT_FUNCTIONS

    from gpustack_amd.ops.torch_ref import build_cos_sin_cache

    scaling = {"rope_type": "yarn", "factor": 4.0,
               "original_max_position_embeddings": 32768,
               "beta_fast": 32, "beta_slow": 1}
    cfg = Qwen2Config(hidden_size=1024, num_attention_heads=8,
                      max_position_embeddings=131072,
                      rope_parameters=dict(scaling, rope_theta=1000000.0))
    inv, att = ROPE_INIT_FUNCTIONS["yarn"](cfg)
    cache = build_cos_sin_cache(128, 128, 512, base=1000000.0,
                                scaling=scaling)
    t = torch.arange(512, dtype=torch.float32)
    freqs = torch.outer(t, inv.float())
    ref = torch.cat([freqs.cos(), freqs.sin()], -1) * att
    assert (cache - ref).abs().max().item() < 1e-4
