"""W4 runtime quantization: packed int4 weights + in-register dequant GEMM.

CPU tier: pack/dequant roundtrip + engine e2e on the torch fallback.
GPU tier (test_ops_gpu.py): kernel vs dequant reference.
"""
import pytest
import torch

from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams
from gpustack_amd.models.quantized import (dequant_w4_runtime,
                                           pack_w4_runtime)


def test_pack_roundtrip():
    torch.manual_seed(0)
    N, K, g = 64, 256, 128
    q = torch.randint(0, 16, (N, K))
    sc = torch.rand(N, K // g) * 0.1 + 0.01
    zr = torch.randint(0, 16, (N, K // g)).float()
    qw, s, zs = pack_w4_runtime(q, sc, zr, g)
    assert qw.shape == (N, K // 2) and qw.dtype == torch.uint8
    w = dequant_w4_runtime(qw, s, zs).float()
    ref = (q.float() - zr.repeat_interleave(g, 1)) * sc.repeat_interleave(g, 1)
    assert (w - ref).abs().max().item() < 2e-2


def test_pack_group_multiple_and_reject():
    q = torch.randint(0, 16, (64, 256))
    assert pack_w4_runtime(q, torch.rand(64, 1), torch.zeros(64, 1), 256) is not None
    assert pack_w4_runtime(q, torch.rand(64, 4), torch.zeros(64, 4), 64) is None
    assert pack_w4_runtime(torch.randint(0, 16, (64, 100)),
                           torch.rand(64, 1), torch.zeros(64, 1), 100) is None


def test_engine_w4_runtime_cpu():
    """Engine serves with packed weights on the CPU fallback path; packs
    exist and the bf16 parameters are freed."""
    cfg = EngineConfig(model="tiny", device="cpu", kv_cache_blocks=64,
                       quantize_runtime="w4")
    eng = LLMEngine(cfg)
    model = eng.runner.model
    packed = sum(1 for layer in model.layers
                 if layer.attn.qkv_pack is not None)
    assert packed > 0
    for layer in model.layers:
        if layer.attn.qkv_pack is not None:
            assert layer.attn.qkv_w.numel() == 0
    out = eng.generate([[1, 2, 3, 4]],
                       SamplingParams(max_tokens=6, ignore_eos=True))[0]
    assert len(out) == 6
    # determinism
    eng2 = LLMEngine(EngineConfig(model="tiny", device="cpu",
                                  kv_cache_blocks=64, quantize_runtime="w4"))
    assert eng2.generate([[1, 2, 3, 4]],
                         SamplingParams(max_tokens=6, ignore_eos=True))[0] == out


def test_w4_runtime_on_mla_model():
    """W4 runtime on a DeepSeek-shaped model: the MLA attention has no
    fused qkv tensor — conversion must pack what it can (o_proj/lm_head)
    and serving must stay exact vs the bf16-free fp32 reference for
    shapes that stay unpacked (tiny dims are W4-ineligible: graceful
    no-op, not a crash)."""
    import dataclasses

    import gpustack_amd.engine.config as C
    from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams
    from gpustack_amd.engine.config import PRESETS

    base = PRESETS["tiny-mla"]
    C.PRESETS["tiny-mla-w4"] = dataclasses.replace(base)
    try:
        plain = LLMEngine(EngineConfig(
            model="tiny-mla-w4", device="cpu", kv_cache_blocks=64,
            max_model_len=128, seed=0, dtype="float32"))
        w4 = LLMEngine(EngineConfig(
            model="tiny-mla-w4", device="cpu", kv_cache_blocks=64,
            max_model_len=128, seed=0, dtype="float32",
            quantize_runtime="w4"))
        p = SamplingParams(max_tokens=5, ignore_eos=True)
        prompts = [[3, 1, 4, 1, 5]]
        out = w4.generate(prompts, p)
        assert len(out[0]) == 5  # serves (no fused-qkv crash)
        # MLA attention stays bf16; the (lossy) packing hit lm_head only
        m = w4.runner.model
        assert not hasattr(m.layers[0].attn, "qkv_pack") \
            or m.layers[0].attn.qkv_pack is None
        assert m.layers[0].attn.o_w.numel() > 0  # unpacked
        assert m.lm_head_pack is not None        # eligible + packed
        # determinism holds under the packed head
        assert w4.generate(prompts, p) == out
    finally:
        C.PRESETS.pop("tiny-mla-w4", None)


def test_engine_w4_runtime_moe_expert_banks():
    """MoE W4: the expert banks pack PER EXPERT (the capacity bulk of
    MoE checkpoints) and dispatch dequants transiently. Output must
    exactly match a reference engine whose bf16 experts are REPLACED by
    the dequantized packed values — proving the serving math reads the
    packed weights and nothing else."""
    import dataclasses

    import gpustack_amd.engine.config as C
    from gpustack_amd.engine.config import PRESETS
    from gpustack_amd.models.quantized import dequant_w4_runtime

    # widen the expert intermediate so BOTH banks are W4-eligible
    # (gate_up [2i, h]: N%64, K%128; down [h, i]: K=i needs %128)
    base = PRESETS["tiny-moe"]
    C.PRESETS["tiny-moe-w4e"] = dataclasses.replace(
        base, moe_intermediate_size=128)
    try:
        p = SamplingParams(max_tokens=6, ignore_eos=True)
        prompts = [[3, 1, 4, 1, 5, 9]]
        w4 = LLMEngine(EngineConfig(model="tiny-moe-w4e", device="cpu",
                                    kv_cache_blocks=64, max_model_len=128,
                                    seed=0, dtype="float32",
                                    quantize_runtime="w4"))
        mlp = w4.runner.model.layers[0].mlp
        assert mlp.gate_up_packs is not None and mlp.down_packs is not None
        assert mlp.gate_up_w.numel() == 0 and mlp.down_w.numel() == 0

        ref = LLMEngine(EngineConfig(model="tiny-moe-w4e", device="cpu",
                                     kv_cache_blocks=64, max_model_len=128,
                                     seed=0, dtype="float32"))
        # graft the dequantized packed values into the reference engine
        for lw4, lref in zip(w4.runner.model.layers,
                             ref.runner.model.layers):
            m4, mr = lw4.mlp, lref.mlp
            if getattr(m4, "gate_up_packs", None) is None:
                continue
            for e, pk in enumerate(m4.gate_up_packs):
                mr.gate_up_w.data[e].copy_(
                    dequant_w4_runtime(pk.qw, pk.sc, pk.zs))
            for e, pk in enumerate(m4.down_packs):
                mr.down_w.data[e].copy_(
                    dequant_w4_runtime(pk.qw, pk.sc, pk.zs))
            # attention/lm_head also packed: graft those too
            if lw4.attn.qkv_pack is not None:
                from gpustack_amd.models.quantized import \
                    dequant_w4_runtime as dq

                lref.attn.qkv_w.data.copy_(
                    dq(lw4.attn.qkv_pack.qw, lw4.attn.qkv_pack.sc,
                       lw4.attn.qkv_pack.zs))
            if lw4.attn.o_pack is not None:
                lref.attn.o_w.data.copy_(
                    dequant_w4_runtime(lw4.attn.o_pack.qw,
                                       lw4.attn.o_pack.sc,
                                       lw4.attn.o_pack.zs))
        if w4.runner.model.lm_head_pack is not None:
            ref.runner.model.lm_head.data.copy_(dequant_w4_runtime(
                w4.runner.model.lm_head_pack.qw,
                w4.runner.model.lm_head_pack.sc,
                w4.runner.model.lm_head_pack.zs))
        assert w4.generate(prompts, p) == ref.generate(prompts, p)
    finally:
        C.PRESETS.pop("tiny-moe-w4e", None)
