"""GLM-4.5 family (Glm4MoeForCausalLM): sigmoid+bias grouped routing,
shared expert, dense-first layers, partial rotary — logits-exact vs HF
transformers at fp32 on CPU (the same oracle discipline as the Llama and
Qwen tests)."""
import dataclasses

import pytest
import torch

import gpustack_amd.engine.config as C
from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams
from gpustack_amd.engine.config import ModelSpec

TINY_GLM = ModelSpec(
    architecture="Glm4MoeForCausalLM", vocab_size=512, hidden_size=128,
    intermediate_size=256, num_layers=3, num_heads=4, num_kv_heads=2,
    head_dim=32, max_position_embeddings=512, rope_theta=10000.0,
    eos_token_id=1, num_experts=8, num_experts_per_tok=2,
    moe_intermediate_size=64, router_mode="sigmoid_bias",
    n_shared_experts=1, first_k_dense_replace=1, partial_rotary_factor=0.5,
    qk_norm=True, n_group=2, topk_group=1, routed_scaling_factor=1.5,
)


@pytest.fixture(autouse=True)
def _tiny_glm_preset():
    C.PRESETS["tiny-glm"] = dataclasses.replace(TINY_GLM)
    yield
    C.PRESETS.pop("tiny-glm", None)


def _engine(**kw):
    return LLMEngine(EngineConfig(model="tiny-glm", device="cpu",
                                  dtype=kw.pop("dtype", "float32"),
                                  kv_cache_blocks=64, **kw))


def test_glm_matches_hf_transformers_logits():
    from transformers import Glm4MoeConfig, Glm4MoeForCausalLM

    eng = _engine()
    spec = eng.cfg.spec
    hf_cfg = Glm4MoeConfig(
        vocab_size=spec.vocab_size, hidden_size=spec.hidden_size,
        intermediate_size=spec.intermediate_size,
        num_hidden_layers=spec.num_layers,
        num_attention_heads=spec.num_heads,
        num_key_value_heads=spec.num_kv_heads,
        head_dim=spec.head_dim, rms_norm_eps=spec.rms_norm_eps,
        rope_theta=spec.rope_theta,
        max_position_embeddings=spec.max_position_embeddings,
        tie_word_embeddings=False, attention_bias=False,
        use_qk_norm=True, partial_rotary_factor=spec.partial_rotary_factor,
        n_routed_experts=spec.num_experts,
        num_experts_per_tok=spec.num_experts_per_tok,
        moe_intermediate_size=spec.moe_intermediate_size,
        n_shared_experts=spec.n_shared_experts,
        first_k_dense_replace=spec.first_k_dense_replace,
        n_group=spec.n_group, topk_group=spec.topk_group,
        routed_scaling_factor=spec.routed_scaling_factor,
        norm_topk_prob=spec.norm_topk_prob,
    )
    hf = Glm4MoeForCausalLM(hf_cfg).eval().float()
    m = eng.runner.model
    d = spec.head_dim
    sd = {
        "model.embed_tokens.weight": m.embed.data,
        "model.norm.weight": m.final_norm.data,
        "lm_head.weight": m.lm_head.data,
    }
    for li, layer in enumerate(m.layers):
        p = f"model.layers.{li}."
        qkv = layer.attn.qkv_w.data
        nq, nk = spec.num_heads * d, spec.num_kv_heads * d
        sd[p + "self_attn.q_proj.weight"] = qkv[:nq]
        sd[p + "self_attn.k_proj.weight"] = qkv[nq:nq + nk]
        sd[p + "self_attn.v_proj.weight"] = qkv[nq + nk:]
        sd[p + "self_attn.o_proj.weight"] = layer.attn.o_w.data
        sd[p + "self_attn.q_norm.weight"] = layer.attn.q_norm.data
        sd[p + "self_attn.k_norm.weight"] = layer.attn.k_norm.data
        sd[p + "input_layernorm.weight"] = layer.input_norm.data
        sd[p + "post_attention_layernorm.weight"] = layer.post_attn_norm.data
        if hasattr(layer.mlp, "router_w"):
            sd[p + "mlp.gate.weight"] = layer.mlp.router_w.data
            sd[p + "mlp.gate.e_score_correction_bias"] = \
                layer.mlp.router_bias.data
            sd[p + "mlp.experts.gate_up_proj"] = layer.mlp.gate_up_w.data
            sd[p + "mlp.experts.down_proj"] = layer.mlp.down_w.data
            sgu = layer.mlp.shared_gate_up_w.data
            si = layer.mlp.shared_i
            sd[p + "mlp.shared_experts.gate_proj.weight"] = sgu[:si]
            sd[p + "mlp.shared_experts.up_proj.weight"] = sgu[si:]
            sd[p + "mlp.shared_experts.down_proj.weight"] = \
                layer.mlp.shared_down_w.data
        else:
            gu = layer.mlp.gate_up_w.data
            ii = spec.intermediate_size
            sd[p + "mlp.gate_proj.weight"] = gu[:ii]
            sd[p + "mlp.up_proj.weight"] = gu[ii:]
            sd[p + "mlp.down_proj.weight"] = layer.mlp.down_w.data
    missing, unexpected = hf.load_state_dict(sd, strict=False)
    assert not unexpected, unexpected
    assert all("rotary" in k or "bias" in k for k in missing), missing

    prompt = [3, 1, 4, 1, 5, 9, 2, 6, 5, 3, 5, 8]
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


def test_glm_decode_matches_prefill():
    eng = _engine()
    p = SamplingParams(max_tokens=8, ignore_eos=True)
    out = eng.generate([[2, 7, 1, 8, 2, 8]], p)[0]
    assert len(out) == 8
    # determinism
    eng2 = _engine()
    assert eng2.generate([[2, 7, 1, 8, 2, 8]], p)[0] == out


def test_glm_spec_from_hf_config():
    from gpustack_amd.engine.config import ModelSpec as MS

    spec = MS.from_hf_config({
        "architectures": ["Glm4MoeForCausalLM"], "vocab_size": 151552,
        "hidden_size": 4096, "intermediate_size": 10944,
        "num_hidden_layers": 46, "num_attention_heads": 96,
        "num_key_value_heads": 8, "head_dim": 128,
        "n_routed_experts": 128, "num_experts_per_tok": 8,
        "moe_intermediate_size": 1408, "n_shared_experts": 1,
        "first_k_dense_replace": 1, "partial_rotary_factor": 0.5,
        "use_qk_norm": True, "routed_scaling_factor": 1.0,
        "n_group": 1, "topk_group": 1, "rope_theta": 1000000.0,
    })
    assert spec.router_mode == "sigmoid_bias"
    assert spec.n_shared_experts == 1 and spec.first_k_dense_replace == 1
    assert spec.partial_rotary_factor == 0.5 and spec.qk_norm
    assert spec.num_experts == 128 and spec.moe_intermediate_size == 1408
