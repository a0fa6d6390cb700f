"""GPT-OSS family (GptOssForCausalLM): attention sinks, alternating
sliding-window/full layers, clamped-swiglu MoE with expert/router biases,
YaRN rope — logits-exact vs HF transformers at fp32 on CPU.

GPU serving is an r3 item (head_dim-64 + sinks/window CDNA4 kernels); the
ops layer fails loudly on those configs — these tests are the numerics
oracle those kernels will be verified against.
"""
import dataclasses

import pytest
import torch

import gpustack_amd.engine.config as C
from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams
from gpustack_amd.engine.config import ModelSpec

TINY_OSS = ModelSpec(
    architecture="GptOssForCausalLM", vocab_size=512, hidden_size=128,
    intermediate_size=128, num_layers=4, num_heads=4, num_kv_heads=2,
    head_dim=32, max_position_embeddings=512, rope_theta=150000.0,
    rope_scaling={"rope_type": "yarn", "factor": 8.0, "beta_fast": 32.0,
                  "beta_slow": 1.0, "truncate": False,
                  "original_max_position_embeddings": 64},
    eos_token_id=1, num_experts=8, num_experts_per_tok=2,
    moe_intermediate_size=128, moe_act="clamped_swiglu", moe_bias=True,
    router_logit_bias=True, attention_bias=True, o_proj_bias=True,
    attention_sinks=True, sliding_window=8,
)


@pytest.fixture(autouse=True)
def _tiny_oss_preset():
    C.PRESETS["tiny-oss"] = dataclasses.replace(TINY_OSS)
    yield
    C.PRESETS.pop("tiny-oss", None)


def _engine(**kw):
    return LLMEngine(EngineConfig(model="tiny-oss", device="cpu",
                                  dtype=kw.pop("dtype", "float32"),
                                  kv_cache_blocks=64, **kw))


def test_gpt_oss_matches_hf_transformers_logits():
    from transformers import GptOssConfig, GptOssForCausalLM

    eng = _engine()
    spec = eng.cfg.spec
    hf_cfg = GptOssConfig(
        vocab_size=spec.vocab_size, hidden_size=spec.hidden_size,
        intermediate_size=spec.moe_intermediate_size,
        num_hidden_layers=spec.num_layers,
        num_attention_heads=spec.num_heads,
        num_key_value_heads=spec.num_kv_heads,
        head_dim=spec.head_dim, rms_norm_eps=spec.rms_norm_eps,
        max_position_embeddings=spec.max_position_embeddings,
        tie_word_embeddings=False, attention_bias=True,
        num_local_experts=spec.num_experts,
        num_experts_per_tok=spec.num_experts_per_tok,
        sliding_window=spec.sliding_window,
        rope_parameters={"rope_type": "yarn", "factor": 8.0,
                         "beta_fast": 32.0, "beta_slow": 1.0,
                         "truncate": False,
                         "original_max_position_embeddings": 64,
                         "rope_theta": spec.rope_theta},
        attn_implementation="eager",
    )
    hf = GptOssForCausalLM(hf_cfg).eval().float()
    assert hf_cfg.layer_types[0] == "sliding_attention"  # alternation matches
    m = eng.runner.model
    d = spec.head_dim
    hq, hkv = spec.num_heads, spec.num_kv_heads
    mi = spec.moe_intermediate_size
    sd = {
        "model.embed_tokens.weight": m.embed.data,
        "model.norm.weight": m.final_norm.data,
        "lm_head.weight": m.lm_head.data,
    }
    for li, layer in enumerate(m.layers):
        p = f"model.layers.{li}."
        qkv = layer.attn.qkv_w.data
        nq, nk = hq * d, hkv * d
        sd[p + "self_attn.q_proj.weight"] = qkv[:nq]
        sd[p + "self_attn.k_proj.weight"] = qkv[nq:nq + nk]
        sd[p + "self_attn.v_proj.weight"] = qkv[nq + nk:]
        qkvb = layer.attn.qkv_b.data
        sd[p + "self_attn.q_proj.bias"] = qkvb[:nq]
        sd[p + "self_attn.k_proj.bias"] = qkvb[nq:nq + nk]
        sd[p + "self_attn.v_proj.bias"] = qkvb[nq + nk:]
        sd[p + "self_attn.o_proj.weight"] = layer.attn.o_w.data
        sd[p + "self_attn.o_proj.bias"] = layer.attn.o_b.data
        sd[p + "self_attn.sinks"] = layer.attn.sinks.data
        sd[p + "input_layernorm.weight"] = layer.input_norm.data
        sd[p + "post_attention_layernorm.weight"] = layer.post_attn_norm.data
        sd[p + "mlp.router.weight"] = layer.mlp.router_w.data
        sd[p + "mlp.router.bias"] = layer.mlp.router_bias.data
        # ours [E, 2i, h] fused [gate; up]; HF [E, h, 2i] interleaved
        gu = layer.mlp.gate_up_w.data
        gate, up = gu[:, :mi], gu[:, mi:]
        inter = torch.empty(spec.num_experts, spec.hidden_size, 2 * mi)
        inter[:, :, 0::2] = gate.transpose(1, 2)
        inter[:, :, 1::2] = up.transpose(1, 2)
        sd[p + "mlp.experts.gate_up_proj"] = inter
        gub = layer.mlp.gate_up_b.data
        interb = torch.empty(spec.num_experts, 2 * mi)
        interb[:, 0::2] = gub[:, :mi]
        interb[:, 1::2] = gub[:, mi:]
        sd[p + "mlp.experts.gate_up_proj_bias"] = interb
        sd[p + "mlp.experts.down_proj"] = \
            layer.mlp.down_w.data.transpose(1, 2)
        sd[p + "mlp.experts.down_proj_bias"] = layer.mlp.down_b.data
    missing, unexpected = hf.load_state_dict(sd, strict=False)
    assert not unexpected, unexpected
    assert all("rotary" in k for k in missing), missing

    prompt = [3, 1, 4, 1, 5, 9, 2, 6, 5, 3, 5, 8, 9, 7]
    with torch.inference_mode():
        hf_logits = hf(torch.tensor([prompt])).logits[0, -1]

    from gpustack_amd.engine.scheduler import ScheduledBatch
    from gpustack_amd.engine.sequence import Sequence

    seq = Sequence("t", prompt)
    seq.block_table = eng.scheduler.kv.allocator.allocate(2)
    batch = ScheduledBatch(
        is_prefill=True, seqs=[seq], token_ids=prompt,
        positions=list(range(len(prompt))),
        slot_mapping=eng.scheduler.kv.slots_for(seq.block_table, 0,
                                                len(prompt)),
        seq_lens=[len(prompt)],
    )
    tokens, meta = eng.runner._meta(batch)
    logits = eng.runner.model(tokens, meta, eng.runner.kv)[0]
    assert torch.allclose(logits, hf_logits, atol=3e-4, rtol=1e-3), (
        (logits - hf_logits).abs().max()
    )


def test_gpt_oss_decode_deterministic():
    """Decode path (sinks + sliding window through paged attention) is
    deterministic and consistent run-to-run."""
    p = SamplingParams(max_tokens=10, ignore_eos=True)
    prompt = [2, 7, 1, 8, 2, 8, 1, 8] * 3  # crosses the window=8 boundary
    a = _engine().generate([prompt], p)[0]
    b = _engine().generate([prompt], p)[0]
    assert a == b and len(a) == 10


def test_gpt_oss_gpu_ops_fail_loudly():
    """The CDNA4 kernels don't support sinks/window yet — the ops layer
    must raise, never silently mis-attend (r3 lands the kernels)."""
    import gpustack_amd.ops as O

    class FakeCuda(torch.Tensor):
        pass

    q = torch.randn(1, 4, 32)
    sinks = torch.zeros(4)
    # emulate the GPU branch by calling with a fake hip module
    orig = O._backend
    O._backend = lambda t: object()
    try:
        with pytest.raises(NotImplementedError):
            O.paged_attn_decode(q, q, q, q, None, None, 1.0, sinks=sinks)
        with pytest.raises(NotImplementedError):
            O.varlen_prefill_attn(q, q, q, q, [1], 1.0, window=8)
    finally:
        O._backend = orig
