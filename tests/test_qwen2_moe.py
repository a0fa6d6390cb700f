"""Qwen2-MoE family (Qwen2MoeForCausalLM, Qwen1.5-MoE-A2.7B /
Qwen2-57B-A14B shape): softmax router with UN-renormalized full-softmax
weights + a shared expert whose output is scaled by a token-wise
sigmoid gate (Linear(h->1)) + qkv bias — logits-exact vs HF
transformers on CPU."""
import dataclasses

import pytest
import torch

import gpustack_amd.engine.config as C
from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams
from gpustack_amd.engine.config import ModelSpec

TINY_Q2MOE = ModelSpec(
    architecture="Qwen2MoeForCausalLM", vocab_size=512, hidden_size=128,
    intermediate_size=256, num_layers=3, num_heads=4, num_kv_heads=2,
    head_dim=32, max_position_embeddings=512, rope_theta=10000.0,
    rms_norm_eps=1e-6, eos_token_id=1, attention_bias=True,
    num_experts=8, num_experts_per_tok=2, moe_intermediate_size=64,
    norm_topk_prob=False, n_shared_experts=2, shared_expert_gated=True,
)


@pytest.fixture(autouse=True)
def _preset():
    C.PRESETS["tiny-q2moe"] = dataclasses.replace(TINY_Q2MOE)
    yield
    C.PRESETS.pop("tiny-q2moe", None)


def _engine(**kw):
    return LLMEngine(EngineConfig(model="tiny-q2moe", device="cpu",
                                  dtype="float32", kv_cache_blocks=64, **kw))


def _hf_from(eng):
    from transformers import Qwen2MoeConfig, Qwen2MoeForCausalLM

    spec = eng.cfg.spec
    hf_cfg = Qwen2MoeConfig(
        vocab_size=spec.vocab_size, hidden_size=spec.hidden_size,
        intermediate_size=spec.intermediate_size,
        num_hidden_layers=spec.num_layers,
        num_attention_heads=spec.num_heads,
        num_key_value_heads=spec.num_kv_heads,
        rms_norm_eps=spec.rms_norm_eps,
        max_position_embeddings=spec.max_position_embeddings,
        rope_theta=spec.rope_theta, tie_word_embeddings=False,
        num_experts=spec.num_experts,
        num_experts_per_tok=spec.num_experts_per_tok,
        moe_intermediate_size=spec.moe_intermediate_size,
        shared_expert_intermediate_size=(spec.moe_intermediate_size
                                         * spec.n_shared_experts),
        norm_topk_prob=False, decoder_sparse_step=1, mlp_only_layers=[],
        eos_token_id=1, pad_token_id=0, attn_implementation="eager",
    )
    hf = Qwen2MoeForCausalLM(hf_cfg).eval().float()
    m = eng.runner.model
    d = spec.head_dim
    nq, nk = spec.num_heads * d, spec.num_kv_heads * d
    sd = {
        "model.embed_tokens.weight": m.embed.data,
        "model.norm.weight": m.final_norm.data,
        "lm_head.weight": m.lm_head.data,
    }
    si = spec.moe_intermediate_size * spec.n_shared_experts
    for li, layer in enumerate(m.layers):
        p = f"model.layers.{li}."
        qkv = layer.attn.qkv_w.data
        sd[p + "self_attn.q_proj.weight"] = qkv[:nq]
        sd[p + "self_attn.k_proj.weight"] = qkv[nq:nq + nk]
        sd[p + "self_attn.v_proj.weight"] = qkv[nq + nk:]
        qkv_b = layer.attn.qkv_b.data
        sd[p + "self_attn.q_proj.bias"] = qkv_b[:nq]
        sd[p + "self_attn.k_proj.bias"] = qkv_b[nq:nq + nk]
        sd[p + "self_attn.v_proj.bias"] = qkv_b[nq + nk:]
        sd[p + "self_attn.o_proj.weight"] = layer.attn.o_w.data
        sd[p + "input_layernorm.weight"] = layer.input_norm.data
        sd[p + "post_attention_layernorm.weight"] = layer.post_attn_norm.data
        sd[p + "mlp.experts.gate_up_proj"] = layer.mlp.gate_up_w.data
        sd[p + "mlp.experts.down_proj"] = layer.mlp.down_w.data
        sd[p + "mlp.gate.weight"] = layer.mlp.router_w.data
        sgu = layer.mlp.shared_gate_up_w.data
        sd[p + "mlp.shared_expert.gate_proj.weight"] = sgu[:si]
        sd[p + "mlp.shared_expert.up_proj.weight"] = sgu[si:]
        sd[p + "mlp.shared_expert.down_proj.weight"] = \
            layer.mlp.shared_down_w.data
        sd[p + "mlp.shared_expert_gate.weight"] = \
            layer.mlp.shared_gate_w.data
    missing, unexpected = hf.load_state_dict(sd, strict=False)
    assert not unexpected, unexpected
    assert all("rotary" in k for k in missing), missing
    return hf


def _prefill_logits(eng, prompt):
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
    return eng.runner.model(tokens, meta, eng.runner.kv)[0]


def test_qwen2_moe_matches_hf_transformers_logits():
    eng = _engine()
    hf = _hf_from(eng)
    prompt = [3, 1, 4, 1, 5, 9, 2, 6, 5, 3]
    with torch.inference_mode():
        want = hf(torch.tensor([prompt])).logits[0, -1]
    got = _prefill_logits(eng, prompt)
    assert torch.allclose(got, want, atol=3e-4, rtol=1e-3), \
        (got - want).abs().max()


def test_qwen2_moe_decode_matches_hf_generation():
    eng = _engine()
    hf = _hf_from(eng)
    prompt = [2, 7, 1, 8, 2, 8]
    with torch.inference_mode():
        out = hf.generate(torch.tensor([prompt]), max_new_tokens=8,
                          do_sample=False, eos_token_id=None)
    want = out[0, len(prompt):].tolist()
    got = eng.generate([prompt], SamplingParams(max_tokens=8,
                                                ignore_eos=True))[0]
    assert got == want


def test_qwen2_moe_spec_from_hf_config():
    spec = ModelSpec.from_hf_config({
        "architectures": ["Qwen2MoeForCausalLM"], "vocab_size": 151936,
        "hidden_size": 2048, "intermediate_size": 5632,
        "num_hidden_layers": 24, "num_attention_heads": 16,
        "num_key_value_heads": 16, "num_experts": 60,
        "num_experts_per_tok": 4, "moe_intermediate_size": 1408,
        "shared_expert_intermediate_size": 5632,
        "norm_topk_prob": False, "rope_theta": 1000000.0,
        "max_position_embeddings": 32768,
    })  # qwen1.5-moe-a2.7b config shape
    assert spec.shared_expert_gated
    assert spec.n_shared_experts == 4  # 5632 / 1408
    assert spec.attention_bias and not spec.norm_topk_prob
