"""On-device numerics: HIP/CDNA4 kernels vs the plain-torch fp32 reference.

Every op the engine uses on the hot path is checked here against
gpustack_amd.ops.torch_ref with random (asymmetric) inputs — including the
MFMA fragment-layout probe (guide G9: transpose-detecting checks).
"""
import math

import pytest
import torch

pytestmark = pytest.mark.gpu

from gpustack_amd import ops
from gpustack_amd.ops import torch_ref as R


def _close(a, b, atol=2e-2, rtol=2e-2, frac=1.0):
    a, b = a.float().cpu(), b.float().cpu()
    ok = torch.isclose(a, b, atol=atol, rtol=rtol)
    if frac >= 1.0:
        assert bool(ok.all()), f"max abs err {(a-b).abs().max().item()}"
    else:
        assert ok.float().mean().item() >= frac


@pytest.fixture(scope="module", autouse=True)
def _seed():
    torch.manual_seed(1234)


def test_hip_ext_loads():
    assert ops.hip_available(), "native _hip_ops.so must load on a GPU box"


def test_mfma_probe_layout():
    # Random asymmetric A, B: catches any transposed fragment mapping.
    a = torch.randn(16, 32, dtype=torch.bfloat16, device="cuda")
    b = torch.randn(32, 16, dtype=torch.bfloat16, device="cuda")
    d = ops.mfma_probe(a, b)
    e = a.float() @ b.float()
    _close(d, e, atol=5e-2, rtol=5e-2)


@pytest.mark.parametrize("T,H", [(1, 4096), (17, 4096), (256, 8192), (3, 16384)])
def test_rms_norm(T, H):
    x = torch.randn(T, H, dtype=torch.bfloat16, device="cuda")
    w = torch.rand(H, dtype=torch.bfloat16, device="cuda") + 0.5
    out = torch.empty_like(x)
    ops.rms_norm(out, x, w, 1e-5)
    ref = torch.empty_like(x)
    R.rms_norm(ref, x, w, 1e-5)
    _close(out, ref)


@pytest.mark.parametrize("T,H", [(5, 4096), (128, 8192)])
def test_fused_add_rms_norm(T, H):
    x = torch.randn(T, H, dtype=torch.bfloat16, device="cuda")
    res = torch.randn(T, H, dtype=torch.bfloat16, device="cuda")
    w = torch.rand(H, dtype=torch.bfloat16, device="cuda") + 0.5
    x2, res2 = x.clone(), res.clone()
    ops.fused_add_rms_norm(x, res, w, 1e-5)
    R.fused_add_rms_norm(x2, res2, w, 1e-5)
    _close(res, res2)
    _close(x, x2)


def test_rotary_embedding():
    T, Hq, Hk, D = 33, 32, 8, 128
    pos = torch.randint(0, 4096, (T,), device="cuda")
    cache = ops.build_cos_sin_cache(D, D, 8192, device="cuda")
    q = torch.randn(T, Hq, D, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(T, Hk, D, dtype=torch.bfloat16, device="cuda")
    q2, k2 = q.clone(), k.clone()
    ops.rotary_embedding(pos, q, k, cache, D, D)
    R.rotary_embedding(pos, q2, k2, cache, D, D)
    _close(q, q2)
    _close(k, k2)


def test_silu_and_mul():
    x = torch.randn(77, 2 * 14336, dtype=torch.bfloat16, device="cuda")
    out = torch.empty(77, 14336, dtype=torch.bfloat16, device="cuda")
    ref = torch.empty_like(out)
    ops.silu_and_mul(out, x)
    R.silu_and_mul(ref, x)
    _close(out, ref)


def test_reshape_and_cache():
    T, Hkv, D, BS, B = 65, 8, 128, 16, 32
    k = torch.randn(T, Hkv, D, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(T, Hkv, D, dtype=torch.bfloat16, device="cuda")
    kc = torch.zeros(B, Hkv, BS, D, dtype=torch.bfloat16, device="cuda")
    vc = torch.zeros_like(kc)
    kc2, vc2 = kc.clone(), vc.clone()
    slots = torch.randperm(B * BS, device="cuda")[:T]
    slots[5] = -1
    ops.reshape_and_cache(k, v, kc, vc, slots)
    R.reshape_and_cache(k, v, kc2, vc2, slots)
    assert torch.equal(kc, kc2)
    assert torch.equal(vc, vc2)


@pytest.mark.parametrize("Hq,Hkv", [(32, 8), (8, 8), (64, 8), (40, 8)])
@pytest.mark.parametrize("lens", [[1], [16], [17, 5, 160, 33], [2048]])
def test_paged_attn_decode(Hq, Hkv, lens):
    D, BS = 128, 16
    N = len(lens)
    maxb = (max(lens) + BS - 1) // BS
    nblocks = sum((l + BS - 1) // BS for l in lens) + 2
    kc = torch.randn(nblocks, Hkv, BS, D, dtype=torch.bfloat16, device="cuda")
    vc = torch.randn(nblocks, Hkv, BS, D, dtype=torch.bfloat16, device="cuda")
    bt = torch.zeros(N, maxb, dtype=torch.int32, device="cuda")
    nxt = 0
    for i, l in enumerate(lens):
        nb = (l + BS - 1) // BS
        bt[i, :nb] = torch.arange(nxt, nxt + nb, dtype=torch.int32)
        nxt += nb
    q = torch.randn(N, Hq, D, dtype=torch.bfloat16, device="cuda")
    sl = torch.tensor(lens, dtype=torch.int32, device="cuda")
    out = torch.empty_like(q)
    ref = torch.empty_like(q)
    scale = 1 / math.sqrt(D)
    ops.paged_attn_decode(out, q, kc, vc, bt, sl, scale)
    R.paged_attn_decode(ref, q, kc, vc, bt, sl, scale)
    _close(out, ref, atol=3e-2, rtol=3e-2)


@pytest.mark.parametrize("Hq,Hkv", [(32, 8), (8, 8)])
@pytest.mark.parametrize("lens", [[1], [64], [63, 70, 5], [300], [1024]])
def test_flash_prefill(Hq, Hkv, lens):
    D = 128
    T = sum(lens)
    q = torch.randn(T, Hq, D, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(T, Hkv, D, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(T, Hkv, D, dtype=torch.bfloat16, device="cuda")
    out = torch.empty_like(q)
    ref = torch.empty_like(q)
    scale = 1 / math.sqrt(D)
    ops.varlen_prefill_attn(out, q, k, v, lens, scale)
    R.varlen_prefill_attn(ref, q, k, v, lens, scale)
    _close(out, ref, atol=4e-2, rtol=4e-2)


def test_greedy_sample():
    logits = torch.randn(64, 128256, dtype=torch.bfloat16, device="cuda")
    got = ops.greedy_sample(logits)
    assert torch.equal(got.cpu(), logits.float().argmax(-1).cpu())


# ---- strided (fused qkv view) paths: exactly what the model runs ----

def _fused_qkv(T, Hq, Hkv, D):
    qkv = torch.randn(T, (Hq + 2 * Hkv) * D, dtype=torch.bfloat16, device="cuda")
    nq, nk = Hq * D, Hkv * D
    q = qkv[:, :nq].unflatten(1, (Hq, D))
    k = qkv[:, nq:nq + nk].unflatten(1, (Hkv, D))
    v = qkv[:, nq + nk:].unflatten(1, (Hkv, D))
    return qkv, q, k, v


def test_rope_strided_matches_contiguous():
    T, Hq, Hkv, D = 33, 32, 8, 128
    qkv, q, k, v = _fused_qkv(T, Hq, Hkv, D)
    pos = torch.randint(0, 1024, (T,), device="cuda")
    cache = ops.build_cos_sin_cache(D, D, 2048, device="cuda")
    qc, kc = q.contiguous(), k.contiguous()
    ops.rotary_embedding(pos, q, k, cache, D, D)
    ops.rotary_embedding(pos, qc, kc, cache, D, D)
    _close(q, qc, atol=0, rtol=0)
    _close(k, kc, atol=0, rtol=0)


def test_reshape_and_cache_strided():
    T, Hq, Hkv, D, BS, B = 48, 32, 8, 128, 16, 8
    _, _, k, v = _fused_qkv(T, Hq, Hkv, D)
    kc = torch.zeros(B, Hkv, BS, D, dtype=torch.bfloat16, device="cuda")
    vc = torch.zeros_like(kc)
    kc2, vc2 = kc.clone(), vc.clone()
    slots = torch.randperm(B * BS, device="cuda")[:T]
    ops.reshape_and_cache(k, v, kc, vc, slots)
    ops.reshape_and_cache(k.contiguous(), v.contiguous(), kc2, vc2, slots)
    assert torch.equal(kc, kc2) and torch.equal(vc, vc2)


def test_paged_decode_strided_q():
    import math
    Hq, Hkv, D, BS = 32, 8, 128, 16
    lens = [7, 40, 180]
    N = len(lens)
    nb = sum((l + BS - 1) // BS for l in lens)
    kcache = torch.randn(nb, Hkv, BS, D, dtype=torch.bfloat16, device="cuda")
    vcache = torch.randn(nb, Hkv, BS, D, dtype=torch.bfloat16, device="cuda")
    maxb = (max(lens) + BS - 1) // BS
    bt = torch.zeros(N, maxb, dtype=torch.int32, device="cuda")
    nxt = 0
    for i, l in enumerate(lens):
        n = (l + BS - 1) // BS
        bt[i, :n] = torch.arange(nxt, nxt + n, dtype=torch.int32)
        nxt += n
    _, q, _, _ = _fused_qkv(N, Hq, Hkv, D)
    sl = torch.tensor(lens, dtype=torch.int32, device="cuda")
    scale = 1 / math.sqrt(D)
    out = torch.empty(N, Hq, D, dtype=torch.bfloat16, device="cuda")
    out2 = torch.empty_like(out)
    ops.paged_attn_decode(out, q, kcache, vcache, bt, sl, scale)
    ops.paged_attn_decode(out2, q.contiguous(), kcache, vcache, bt, sl, scale)
    assert torch.equal(out, out2)


def test_flash_prefill_strided():
    import math
    Hq, Hkv, D = 32, 8, 128
    lens = [70, 130, 5]
    T = sum(lens)
    _, q, k, v = _fused_qkv(T, Hq, Hkv, D)
    out = torch.empty(T, Hq, D, dtype=torch.bfloat16, device="cuda")
    out2 = torch.empty_like(out)
    scale = 1 / math.sqrt(D)
    ops.varlen_prefill_attn(out, q, k, v, lens, scale)
    ops.varlen_prefill_attn(out2, q.contiguous(), k.contiguous(), v.contiguous(), lens, scale)
    assert torch.equal(out, out2)


@pytest.mark.parametrize("M,N,K,splitk", [
    (512, 6144, 4096, None),    # qkv
    (512, 4096, 4096, 4),       # o_proj, forced split
    (256, 28672, 4096, 1),      # gate_up, no split
    (512, 4096, 14336, 4),      # down_proj
    (320, 4096, 4096, 2),       # M not a tile multiple (clamped rows)
    (16, 4096, 4096, 8),        # tiny M
])
@pytest.mark.parametrize("version", [1, 2])
def test_skinny_gemm(M, N, K, splitk, version):
    if version == 2 and N % 256:
        pytest.skip("v2 needs N%256==0")
    x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda") * 0.1
    w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.1
    got = ops.skinny_gemm(x, w, splitk=splitk, version=version)
    ref = (x.float() @ w.float().T)
    assert torch.allclose(got.float(), ref, atol=0.5, rtol=3e-2), (
        (got.float() - ref).abs().max().item()
    )


@pytest.mark.parametrize("lens", [[17, 5, 160], [2048]])
def test_paged_attn_decode_fp8(lens):
    Hq, Hkv, D, BS = 32, 8, 128, 16
    N = len(lens)
    maxb = (max(lens) + BS - 1) // BS
    nb = sum((l + BS - 1) // BS for l in lens) + 1
    kc = (torch.randn(nb, Hkv, BS, D, device="cuda") * 2).to(torch.float8_e4m3fn)
    vc = (torch.randn(nb, Hkv, BS, D, device="cuda") * 2).to(torch.float8_e4m3fn)
    bt = torch.zeros(N, maxb, dtype=torch.int32, device="cuda")
    nxt = 0
    for i, l in enumerate(lens):
        n = (l + BS - 1) // BS
        bt[i, :n] = torch.arange(nxt, nxt + n, dtype=torch.int32)
        nxt += n
    q = torch.randn(N, Hq, D, dtype=torch.bfloat16, device="cuda")
    sl = torch.tensor(lens, dtype=torch.int32, device="cuda")
    out = torch.empty_like(q)
    ref = torch.empty_like(q)
    scale = 1 / math.sqrt(D)
    ops.paged_attn_decode(out, q, kc, vc, bt, sl, scale)
    R.paged_attn_decode(ref, q, kc, vc, bt, sl, scale)
    _close(out, ref, atol=3e-2, rtol=3e-2)


def test_reshape_and_cache_fp8():
    T, Hkv, D, BS, B = 33, 8, 128, 16, 8
    k = torch.randn(T, Hkv, D, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(T, Hkv, D, dtype=torch.bfloat16, device="cuda")
    kc = torch.zeros(B, Hkv, BS, D, device="cuda").to(torch.float8_e4m3fn)
    vc = torch.zeros(B, Hkv, BS, D, device="cuda").to(torch.float8_e4m3fn)
    kc2, vc2 = kc.clone(), vc.clone()
    slots = torch.randperm(B * BS, device="cuda")[:T]
    ops.reshape_and_cache(k, v, kc, vc, slots)
    R.reshape_and_cache(k, v, kc2, vc2, slots)
    # HW cvt and torch cvt are both OCP e4m3 RNE: bit-identical expected
    assert torch.equal(kc.view(torch.uint8), kc2.view(torch.uint8))
    assert torch.equal(vc.view(torch.uint8), vc2.view(torch.uint8))


@pytest.mark.parametrize("safe", [True, False])
@pytest.mark.parametrize("shape", [(256, 256, 128), (320, 512, 512),
                                   (512, 1024, 896), (100, 256, 256)])
def test_gemm8(shape, safe):
    M, N, K = shape
    x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda")
    out = ops.gemm8(x, w, safe=safe)
    ref = (x.float() @ w.float().t())
    _close(out, ref.to(torch.bfloat16), atol=8e-2, rtol=8e-2)


def test_gemm8_matches_safe_variant():
    # the pipelined schedule must agree bit-for-bit with the drained one
    torch.manual_seed(7)
    x = torch.randn(512, 4096, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(4096, 4096, dtype=torch.bfloat16, device="cuda")
    a = ops.gemm8(x, w, safe=False)
    b = ops.gemm8(x, w, safe=True)
    assert torch.equal(a, b)


@pytest.mark.parametrize("Hq,Hkv", [(32, 8), (8, 8)])
@pytest.mark.parametrize("cases", [
    [(0, 64)],                 # no history (pure paged prefill)
    [(96, 64)],                # block-aligned history
    [(100, 29)],               # unaligned history + odd suffix
    [(1000, 513), (16, 1)],    # long history, multi-tile suffix, 1-row seq
])
def test_flash_prefill_paged(Hq, Hkv, cases):
    """Prefill-with-history kernel vs torch_ref: K/V gathered from the
    paged pool through shuffled block tables; history + suffix causality."""
    D, BS = 128, 16
    dev = "cuda"
    nseq = len(cases)
    tot_new = sum(n for _, n in cases)
    maxb = max((h + n + BS - 1) // BS for h, n in cases)
    nblocks = sum((h + n + BS - 1) // BS for h, n in cases) + 3
    k_cache = torch.randn(nblocks, Hkv, BS, D, dtype=torch.bfloat16, device=dev) / 4
    v_cache = torch.randn(nblocks, Hkv, BS, D, dtype=torch.bfloat16, device=dev) / 4
    # shuffled block assignment (realistic non-contiguous tables)
    perm = torch.randperm(nblocks - 1)[: sum((h + n + BS - 1) // BS for h, n in cases)] + 1
    bt = torch.zeros(nseq, maxb, dtype=torch.int32, device=dev)
    off = 0
    for i, (h, n) in enumerate(cases):
        nb = (h + n + BS - 1) // BS
        bt[i, :nb] = perm[off:off + nb].to(torch.int32)
        off += nb
    q = torch.randn(tot_new, Hq, D, dtype=torch.bfloat16, device=dev) / 4
    starts, hists, news = [], [], []
    r = 0
    for h, n in cases:
        starts.append(r)
        hists.append(h)
        news.append(n)
        r += n
    out = torch.empty_like(q)
    ops.paged_prefill_attn(out, q, k_cache, v_cache, bt, starts, hists, news,
                           1.0 / math.sqrt(D))
    ref = torch.empty_like(q)
    R.paged_prefill_attn(ref, q, k_cache, v_cache, bt, starts, hists, news,
                         1.0 / math.sqrt(D))
    _close(out, ref)


def test_flash_prefill_paged_matches_contiguous():
    """hist=0 paged prefill == the contiguous varlen prefill kernel on the
    same data (cross-kernel consistency)."""
    D, BS, Hq, Hkv = 128, 16, 32, 8
    lens = [64, 129, 7]
    T = sum(lens)
    dev = "cuda"
    q = torch.randn(T, Hq, D, dtype=torch.bfloat16, device=dev) / 4
    k = torch.randn(T, Hkv, D, dtype=torch.bfloat16, device=dev) / 4
    v = torch.randn(T, Hkv, D, dtype=torch.bfloat16, device=dev) / 4
    out_c = torch.empty_like(q)
    ops.varlen_prefill_attn(out_c, q, k, v, lens, 1.0 / math.sqrt(D))
    # scatter k/v into a pool
    nb_per = [(L + BS - 1) // BS for L in lens]
    nblocks = sum(nb_per) + 1
    k_cache = torch.zeros(nblocks, Hkv, BS, D, dtype=torch.bfloat16, device=dev)
    v_cache = torch.zeros(nblocks, Hkv, BS, D, dtype=torch.bfloat16, device=dev)
    maxb = max(nb_per)
    bt = torch.zeros(len(lens), maxb, dtype=torch.int32, device=dev)
    blk = 1
    row = 0
    slots = []
    for i, L in enumerate(lens):
        for j in range(nb_per[i]):
            bt[i, j] = blk + j
        for p in range(L):
            slots.append((blk + p // BS) * BS + p % BS)
        blk += nb_per[i]
        row += L
    slots_t = torch.tensor(slots, dtype=torch.long, device=dev)
    ops.reshape_and_cache(k, v, k_cache, v_cache, slots_t)
    starts = [sum(lens[:i]) for i in range(len(lens))]
    out_p = torch.empty_like(q)
    ops.paged_prefill_attn(out_p, q, k_cache, v_cache, bt, starts,
                           [0] * len(lens), lens, 1.0 / math.sqrt(D))
    _close(out_p, out_c)


@pytest.mark.parametrize("T,E,topk,H,I", [
    (7, 16, 2, 256, 128),     # tiny, many empty experts
    (128, 128, 8, 2048, 768), # qwen3-30b-a3b decode shape
    (64, 8, 2, 4096, 1792),   # mixtral-like (i//tp 8 -> 1792)
])
def test_fused_moe_kernels(T, E, topk, H, I):
    """Fused grouped MoE kernels vs a plain per-expert torch loop."""
    import torch.nn.functional as F

    dev = "cuda"
    x = torch.randn(T, H, dtype=torch.bfloat16, device=dev) / 8
    w_gu = torch.randn(E, 2 * I, H, dtype=torch.bfloat16, device=dev) / 16
    w_d = torch.randn(E, H, I, dtype=torch.bfloat16, device=dev) / 16
    logits = torch.randn(T, E, device=dev)
    weights, experts = torch.topk(torch.softmax(logits, -1), topk, dim=-1)
    flat_exp = experts.reshape(-1)
    flat_tok = torch.arange(T, device=dev).repeat_interleave(topk)
    flat_w = weights.reshape(-1).float()

    order = torch.argsort(flat_exp, stable=True)
    s_tok = flat_tok[order].to(torch.int32)
    counts = torch.zeros(E, dtype=torch.int32, device=dev)
    counts.scatter_add_(0, flat_exp, torch.ones_like(flat_exp, dtype=torch.int32))
    offs = (counts.cumsum(0, dtype=torch.int32) - counts).to(torch.int32)
    hip = ops._load_hip()
    act = x.new_empty(T * topk, I)
    hip.moe_gate_up_silu(act, x, w_gu, s_tok, offs, counts)
    contrib = x.new_empty(T * topk, H)
    hip.moe_down_scale(contrib, act, w_d, offs, counts,
                       order.to(torch.int32), flat_w)
    got = contrib.view(T, topk, H).sum(1)

    # reference: per-expert loop in fp32-ish (bf16 GEMM via F.linear)
    ref = torch.zeros(T, H, dtype=torch.float32, device=dev)
    for e in range(E):
        rows = torch.nonzero(flat_exp == e).flatten()
        if not rows.numel():
            continue
        xe = x[flat_tok[rows]]
        gu = F.linear(xe, w_gu[e]).float()
        a = (F.silu(gu[:, :I]) * gu[:, I:]).to(torch.bfloat16)
        he = F.linear(a, w_d[e]).float() * flat_w[rows].unsqueeze(1)
        ref.index_add_(0, flat_tok[rows], he)
    _close(got, ref, atol=5e-2, rtol=5e-2)


def test_moe_model_fused_matches_fallback():
    """End-to-end MoEMLP: fused HIP dispatch == torch fallback dispatch."""
    import os

    from gpustack_amd.engine.config import ModelSpec
    from gpustack_amd.models.llama import MoEMLP
    from gpustack_amd.parallel import Communicator

    spec = ModelSpec(hidden_size=512, intermediate_size=1024, num_layers=1,
                     num_heads=8, num_kv_heads=8, head_dim=64,
                     num_experts=32, num_experts_per_tok=4,
                     moe_intermediate_size=256, vocab_size=1000)
    torch.manual_seed(7)
    m = MoEMLP(spec, 1, Communicator(), torch.bfloat16)
    for p in m.parameters():
        p.data.normal_(0, 0.05)
    m = m.to("cuda")
    x = torch.randn(33, 512, dtype=torch.bfloat16, device="cuda") / 8
    out_fused = m(x.clone())
    os.environ["GPUSTACK_AMD_FUSED_MOE"] = "0"
    try:
        out_ref = m(x.clone())
    finally:
        os.environ["GPUSTACK_AMD_FUSED_MOE"] = "1"
    _close(out_fused, out_ref, atol=3e-2, rtol=3e-2)


@pytest.mark.parametrize("M,N,K", [(1, 128, 256), (17, 256, 512),
                                   (512, 1024, 4096), (300, 128, 1024)])
def test_w4_gemm(M, N, K):
    """W4A16 kernel vs dequantized-bf16 F.linear."""
    from gpustack_amd.models.quantized import (dequant_w4_runtime,
                                               pack_w4_runtime)

    torch.manual_seed(3)
    dev = "cuda"
    x = torch.randn(M, K, dtype=torch.bfloat16, device=dev) / 8
    q = torch.randint(0, 16, (N, K), device=dev)
    sc = (torch.rand(N, K // 128, device=dev) * 0.05 + 0.01)
    zr = torch.randint(0, 16, (N, K // 128), device=dev).float()
    qw, s, zs = pack_w4_runtime(q, sc, zr, 128)
    wt = dequant_w4_runtime(qw, s, zs)
    ref = torch.nn.functional.linear(x, wt)
    out = torch.empty(M, N, dtype=torch.bfloat16, device=dev)
    ops._load_hip().w4_gemm(out, x, qw, s, zs)
    _close(out, ref, atol=3e-2, rtol=3e-2)


@pytest.mark.parametrize("model", ["llama-3-8b"])
def test_engine_w4_runtime_gpu(model):
    """End-to-end W4 serving on GPU: packs active, decode runs the w4
    kernel, output deterministic."""
    from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams

    cfg = EngineConfig(model=model, device="cuda:0", max_model_len=512,
                       max_num_seqs=8, gpu_memory_utilization=0.2,
                       quantize_runtime="w4")
    cfg.spec.num_layers = 4
    eng = LLMEngine(cfg)
    assert eng.runner.model.layers[0].attn.qkv_pack is not None
    p = SamplingParams(max_tokens=8, ignore_eos=True)
    out = eng.generate([[1, 2, 3, 4, 5] * 10], p)[0]
    assert len(out) == 8


def test_w4_dequant_kernel():
    from gpustack_amd.models.quantized import (dequant_w4_runtime,
                                               pack_w4_runtime)

    torch.manual_seed(5)
    dev = "cuda"
    N, K = 192, 640
    q = torch.randint(0, 16, (N, K), device=dev)
    sc = torch.rand(N, K // 128, device=dev) * 0.05 + 0.01
    zr = torch.randint(0, 16, (N, K // 128), device=dev).float()
    qw, s, zs = pack_w4_runtime(q, sc, zr, 128)
    ref = dequant_w4_runtime(qw, s, zs)
    out = torch.empty(N, K, dtype=torch.bfloat16, device=dev)
    ops._load_hip().w4_dequant(out, qw, s, zs)
    assert torch.equal(out, ref) or (out.float() - ref.float()).abs().max() < 1e-2


# ---- GPT-OSS kernel variants: head_dim 64 + attention sinks + sliding ----
# window. Written+compile-checked in r2, GPU-validated in r3 — opt in with
# GPUSTACK_AMD_OSS_KERNELS=1 so the default round-end suite stays on the
# validated configs only.

import os as _os

oss = pytest.mark.skipif(
    _os.environ.get("GPUSTACK_AMD_OSS_KERNELS") != "1",
    reason="set GPUSTACK_AMD_OSS_KERNELS=1 to test the unvalidated "
           "sinks/window/D64 kernel variants")


def _paged_pool(lens, Hkv, D, BS=16, dev="cuda"):
    N = len(lens)
    maxb = (max(lens) + BS - 1) // BS
    nblocks = sum((l + BS - 1) // BS for l in lens) + 2
    kc = torch.randn(nblocks, Hkv, BS, D, dtype=torch.bfloat16, device=dev) / 4
    vc = torch.randn(nblocks, Hkv, BS, D, dtype=torch.bfloat16, device=dev) / 4
    bt = torch.zeros(N, maxb, dtype=torch.int32, device=dev)
    nxt = 0
    for i, l in enumerate(lens):
        nb = (l + BS - 1) // BS
        bt[i, :nb] = torch.arange(nxt, nxt + nb, dtype=torch.int32)
        nxt += nb
    return kc, vc, bt


@oss
@pytest.mark.parametrize("D,Hq,Hkv", [(64, 64, 8), (64, 8, 8), (128, 32, 8)])
@pytest.mark.parametrize("lens", [[1], [16], [17, 5, 160, 33], [700]])
@pytest.mark.parametrize("sink,window", [
    (True, 0), (False, 128), (True, 128), (False, 0),
])
def test_oss_paged_attn_decode(D, Hq, Hkv, lens, sink, window):
    kc, vc, bt = _paged_pool(lens, Hkv, D)
    q = torch.randn(len(lens), Hq, D, dtype=torch.bfloat16, device="cuda")
    sl = torch.tensor(lens, dtype=torch.int32, device="cuda")
    sinks = torch.randn(Hq, device="cuda") if sink else None
    out = torch.empty_like(q)
    ref = torch.empty_like(q)
    scale = 1 / math.sqrt(D)
    ops.paged_attn_decode(out, q, kc, vc, bt, sl, scale, sinks=sinks,
                          window=window)
    R.paged_attn_decode(ref, q, kc, vc, bt, sl, scale, sinks=sinks,
                        window=window)
    _close(out, ref, atol=3e-2, rtol=3e-2)


@oss
@pytest.mark.parametrize("D,Hq,Hkv", [(64, 64, 8), (128, 32, 8)])
@pytest.mark.parametrize("lens", [[64], [63, 70, 5], [300]])
@pytest.mark.parametrize("sink,window", [
    (True, 0), (False, 128), (True, 128), (False, 0),
])
def test_oss_varlen_prefill(D, Hq, Hkv, lens, sink, window):
    T = sum(lens)
    q = torch.randn(T, Hq, D, dtype=torch.bfloat16, device="cuda") / 4
    k = torch.randn(T, Hkv, D, dtype=torch.bfloat16, device="cuda") / 4
    v = torch.randn(T, Hkv, D, dtype=torch.bfloat16, device="cuda") / 4
    sinks = torch.randn(Hq, device="cuda") if sink else None
    out = torch.empty_like(q)
    ref = torch.empty_like(q)
    scale = 1 / math.sqrt(D)
    ops.varlen_prefill_attn(out, q, k, v, lens, scale, sinks=sinks,
                            window=window)
    R.varlen_prefill_attn(ref, q, k, v, lens, scale, sinks=sinks,
                          window=window)
    _close(out, ref, atol=4e-2, rtol=4e-2)


@oss
@pytest.mark.parametrize("D", [64, 128])
@pytest.mark.parametrize("cases", [
    [(0, 64)], [(100, 29)], [(1000, 200), (16, 1)],
])
@pytest.mark.parametrize("sink,window", [(True, 128), (True, 0), (False, 96)])
def test_oss_paged_prefill(D, cases, sink, window):
    """Chunked-prefill continuation rows with sinks/window: the sliding
    window must count from the ABSOLUTE position (history + row)."""
    Hq, Hkv, BS = 16, 8, 16
    dev = "cuda"
    tot_new = sum(n for _, n in cases)
    lens = [h + n for h, n in cases]
    kc, vc, bt = _paged_pool(lens, Hkv, D)
    q = torch.randn(tot_new, Hq, D, dtype=torch.bfloat16, device=dev) / 4
    starts, hists, news = [], [], []
    r = 0
    for h, n in cases:
        starts.append(r)
        hists.append(h)
        news.append(n)
        r += n
    sinks = torch.randn(Hq, device=dev) if sink else None
    out = torch.empty_like(q)
    ref = torch.empty_like(q)
    scale = 1 / math.sqrt(D)
    ops.paged_prefill_attn(out, q, kc, vc, bt, starts, hists, news, scale,
                           sinks=sinks, window=window)
    R.paged_prefill_attn(ref, q, kc, vc, bt, starts, hists, news, scale,
                         sinks=sinks, window=window)
    _close(out, ref)


@oss
@pytest.mark.parametrize("bias", [True, False])
def test_oss_fused_moe_clamped_swiglu(bias):
    """moe_gate_up_silu act_mode=1 (+expert biases) + moe_down_scale bias
    vs plain torch: clamped swiglu per models/llama.py MoEMLP._act_mul."""
    torch.manual_seed(0)
    E, H, I, T, K = 8, 256, 128, 33, 2
    dev = "cuda"
    x = torch.randn(T, H, dtype=torch.bfloat16, device=dev) / 4
    w_gu = torch.randn(E, 2 * I, H, dtype=torch.bfloat16, device=dev) / 8
    w_d = torch.randn(E, H, I, dtype=torch.bfloat16, device=dev) / 8
    b_gu = (torch.randn(E, 2 * I, dtype=torch.bfloat16, device=dev) / 4
            if bias else None)
    b_d = (torch.randn(E, H, dtype=torch.bfloat16, device=dev) / 4
           if bias else None)
    flat_exp = torch.randint(0, E, (T * K,), device=dev)
    flat_tok = torch.arange(T, device=dev).repeat_interleave(K)
    flat_w32 = torch.rand(T * K, dtype=torch.float32, device=dev)

    order = torch.argsort(flat_exp, stable=True)
    s_tok = flat_tok[order].to(torch.int32)
    counts = torch.zeros(E, dtype=torch.int32, device=dev)
    counts.scatter_add_(0, flat_exp, torch.ones_like(flat_exp, dtype=torch.int32))
    offs = (counts.cumsum(0, dtype=torch.int32) - counts).to(torch.int32)

    hip = ops._load_hip()
    act = x.new_empty(T * K, I)
    hip.moe_gate_up_silu(act, x, w_gu, s_tok, offs, counts, bias=b_gu,
                         act_mode=1)
    contrib = x.new_empty(T * K, H)
    hip.moe_down_scale(contrib, act, w_d, offs, counts,
                       order.to(torch.int32), flat_w32, bias=b_d)

    # torch reference (fp32)
    ref = torch.zeros(T * K, H, dtype=torch.float32, device=dev)
    for j in range(T * K):
        e = int(flat_exp[j])
        t = int(flat_tok[j])
        gu = x[t].float() @ w_gu[e].float().T
        if b_gu is not None:
            gu = gu + b_gu[e].float()
        g = gu[:I].clamp(max=7.0)
        u = gu[I:].clamp(-7.0, 7.0)
        a = (u + 1.0) * (g * torch.sigmoid(g * 1.702))
        d = a @ w_d[e].float().T
        if b_d is not None:
            d = d + b_d[e].float()
        ref[j] = d * flat_w32[j]
    _close(contrib, ref.to(torch.bfloat16), atol=6e-2, rtol=6e-2)


mla = pytest.mark.skipif(
    _os.environ.get("GPUSTACK_AMD_MLA_KERNEL") != "1",
    reason="set GPUSTACK_AMD_MLA_KERNEL=1 to test the unvalidated MLA "
           "absorbed-decode kernel")


@mla
@pytest.mark.parametrize("H", [16, 128])
@pytest.mark.parametrize("lens", [[1], [16], [17, 5, 160, 33], [1200]])
def test_mla_decode_kernel(H, lens):
    """mla_decode vs a plain fp32 torch reference of the absorbed
    formulation over the paged latent pool (R=512, DR=64)."""
    torch.manual_seed(0)
    R, DR, BS = 512, 64, 16
    LD = R + DR
    N = len(lens)
    maxb = (max(lens) + BS - 1) // BS
    nblocks = sum((l + BS - 1) // BS for l in lens) + 2
    lat = torch.randn(nblocks, 1, BS, LD, dtype=torch.bfloat16,
                      device="cuda") / 4
    bt = torch.zeros(N, maxb, dtype=torch.int32, device="cuda")
    nxt = 0
    for i, l in enumerate(lens):
        nb = (l + BS - 1) // BS
        bt[i, :nb] = torch.arange(nxt, nxt + nb, dtype=torch.int32)
        nxt += nb
    q = torch.randn(N, H, LD, dtype=torch.bfloat16, device="cuda") / 4
    sl = torch.tensor(lens, dtype=torch.int32, device="cuda")
    scale = 1.0 / (192 ** 0.5)
    ctx = torch.empty(N, H, R, dtype=torch.float32, device="cuda")
    ops.mla_decode(ctx, q, lat, bt, sl, scale)

    flat = lat.view(-1, LD).float()
    for i, L in enumerate(lens):
        idx = (bt[i][torch.arange(L, device="cuda") // BS].long() * BS
               + torch.arange(L, device="cuda") % BS)
        C = flat[idx]                               # [L, LD]
        scores = (q[i].float() @ C.T) * scale       # [H, L]
        probs = torch.softmax(scores, dim=-1)
        want = probs @ C[:, :R]                     # [H, R]
        assert torch.allclose(ctx[i], want, atol=2e-2, rtol=2e-2), \
            (ctx[i] - want).abs().max()


@oss
@pytest.mark.parametrize("D,Hq,Hkv", [(256, 8, 4), (128, 32, 8)])
@pytest.mark.parametrize("lens", [[1], [17, 5, 160, 33], [700]])
@pytest.mark.parametrize("window,softcap", [
    (0, 50.0), (8, 50.0), (0, 0.0),
])
def test_oss_decode_softcap_d256(D, Hq, Hkv, lens, window, softcap):
    """Gemma-2/3 kernel variants: D=256 decode template + tanh logit
    softcapping (optionally with a sliding window) vs torch_ref."""
    kc, vc, bt = _paged_pool(lens, Hkv, D)
    q = torch.randn(len(lens), Hq, D, dtype=torch.bfloat16, device="cuda")
    sl = torch.tensor(lens, dtype=torch.int32, device="cuda")
    out = torch.empty_like(q)
    ref = torch.empty_like(q)
    scale = 1 / math.sqrt(D)
    ops.paged_attn_decode(out, q, kc, vc, bt, sl, scale, window=window,
                          softcap=softcap)
    R.paged_attn_decode(ref, q, kc, vc, bt, sl, scale, window=window,
                        softcap=softcap)
    _close(out, ref, atol=3e-2, rtol=3e-2)


@oss
@pytest.mark.parametrize("D", [256, 128])
@pytest.mark.parametrize("lens", [[64], [63, 70, 5], [300]])
@pytest.mark.parametrize("window,softcap", [(0, 50.0), (8, 50.0)])
def test_oss_prefill_softcap_d256(D, lens, window, softcap):
    Hq, Hkv = 8, 4
    T = sum(lens)
    q = torch.randn(T, Hq, D, dtype=torch.bfloat16, device="cuda") / 4
    k = torch.randn(T, Hkv, D, dtype=torch.bfloat16, device="cuda") / 4
    v = torch.randn(T, Hkv, D, dtype=torch.bfloat16, device="cuda") / 4
    out = torch.empty_like(q)
    ref = torch.empty_like(q)
    scale = 1 / math.sqrt(D)
    ops.varlen_prefill_attn(out, q, k, v, lens, scale, window=window,
                            softcap=softcap)
    R.varlen_prefill_attn(ref, q, k, v, lens, scale, window=window,
                          softcap=softcap)
    _close(out, ref, atol=4e-2, rtol=4e-2)


@mla
@pytest.mark.parametrize("lens", [[64], [63, 70, 5], [300]])
def test_mla_expand_prefill_shape(lens):
    """flash_prefill at the MLA expand shape (D_qk=192 over D_v=128) vs
    torch_ref (which handles asymmetric v dims natively)."""
    Hq, Hkv, D, DV = 16, 16, 192, 128
    T = sum(lens)
    q = torch.randn(T, Hq, D, dtype=torch.bfloat16, device="cuda") / 4
    k = torch.randn(T, Hkv, D, dtype=torch.bfloat16, device="cuda") / 4
    v = torch.randn(T, Hkv, DV, dtype=torch.bfloat16, device="cuda") / 4
    out = torch.empty(T, Hq, DV, dtype=torch.bfloat16, device="cuda")
    scale = 1 / math.sqrt(D)
    ops.varlen_prefill_attn(out, q, k, v, lens, scale)
    # fp32 reference
    off = 0
    for L in lens:
        qs, ks, vs = (t[off:off + L].float() for t in (q, k, v))
        for h in range(Hq):
            att = (qs[:, h] @ ks[:, h].T) * scale
            mask = torch.triu(torch.ones(L, L, device="cuda"), 1).bool()
            att = att.masked_fill(mask, float("-inf"))
            want = torch.softmax(att, -1) @ vs[:, h]
            got = out[off:off + L, h].float()
            assert torch.allclose(got, want, atol=4e-2, rtol=4e-2), \
                (got - want).abs().max()
        off += L
