"""dots.llm1 family (Dots1ForCausalLM): DeepSeek-style grouped
sigmoid+bias routing (e_score_correction_bias on the CHOICE only,
renormalized gathered scores x routed_scaling_factor) + shared experts
+ dense first layers + Qwen3-style per-head qk-norm — logits-exact vs
HF transformers on CPU. Everything composes from existing spec fields;
only the arch mapping is new."""
import dataclasses

import pytest
import torch

import gpustack_amd.engine.config as C
from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams
from gpustack_amd.engine.config import ModelSpec

TINY_DOTS = ModelSpec(
    architecture="Dots1ForCausalLM", vocab_size=512, hidden_size=128,
    intermediate_size=256, num_layers=3, num_heads=4, num_kv_heads=2,
    head_dim=32, max_position_embeddings=512, rope_theta=10000.0,
    rms_norm_eps=1e-6, eos_token_id=1, qk_norm=True,
    num_experts=8, num_experts_per_tok=2, moe_intermediate_size=64,
    router_mode="sigmoid_bias", norm_topk_prob=True, n_group=2,
    topk_group=1, routed_scaling_factor=2.5, n_shared_experts=1,
    first_k_dense_replace=1,
)


@pytest.fixture(autouse=True)
def _preset():
    C.PRESETS["tiny-dots"] = dataclasses.replace(TINY_DOTS)
    yield
    C.PRESETS.pop("tiny-dots", None)


def _engine(**kw):
    return LLMEngine(EngineConfig(model="tiny-dots", device="cpu",
                                  dtype="float32", kv_cache_blocks=64, **kw))


def _hf_from(eng):
    from transformers import Dots1Config, Dots1ForCausalLM

    spec = eng.cfg.spec
    hf_cfg = Dots1Config(
        vocab_size=spec.vocab_size, hidden_size=spec.hidden_size,
        intermediate_size=spec.intermediate_size,
        num_hidden_layers=spec.num_layers,
        num_attention_heads=spec.num_heads,
        num_key_value_heads=spec.num_kv_heads,
        head_dim=spec.head_dim, rms_norm_eps=spec.rms_norm_eps,
        max_position_embeddings=spec.max_position_embeddings,
        rope_theta=spec.rope_theta, tie_word_embeddings=False,
        attention_bias=False, num_local_experts=spec.num_experts,
        num_experts_per_tok=spec.num_experts_per_tok,
        moe_intermediate_size=spec.moe_intermediate_size,
        n_shared_experts=spec.n_shared_experts,
        n_group=spec.n_group, topk_group=spec.topk_group,
        routed_scaling_factor=spec.routed_scaling_factor,
        norm_topk_prob=True,
        first_k_dense_replace=spec.first_k_dense_replace,
        eos_token_id=1, pad_token_id=0, attn_implementation="eager",
    )
    hf = Dots1ForCausalLM(hf_cfg).eval().float()
    m = eng.runner.model
    spec_ = spec
    d = spec.head_dim
    nq, nk = spec.num_heads * d, spec.num_kv_heads * d
    sd = {
        "model.embed_tokens.weight": m.embed.data,
        "model.norm.weight": m.final_norm.data,
        "lm_head.weight": m.lm_head.data,
    }
    for li, layer in enumerate(m.layers):
        p = f"model.layers.{li}."
        qkv = layer.attn.qkv_w.data
        sd[p + "self_attn.q_proj.weight"] = qkv[:nq]
        sd[p + "self_attn.k_proj.weight"] = qkv[nq:nq + nk]
        sd[p + "self_attn.v_proj.weight"] = qkv[nq + nk:]
        sd[p + "self_attn.o_proj.weight"] = layer.attn.o_w.data
        sd[p + "self_attn.q_norm.weight"] = layer.attn.q_norm.data
        sd[p + "self_attn.k_norm.weight"] = layer.attn.k_norm.data
        sd[p + "input_layernorm.weight"] = layer.input_norm.data
        sd[p + "post_attention_layernorm.weight"] = layer.post_attn_norm.data
        if hasattr(layer.mlp, "router_w"):  # MoE layers (li >= 1)
            sd[p + "mlp.experts.gate_up_proj"] = layer.mlp.gate_up_w.data
            sd[p + "mlp.experts.down_proj"] = layer.mlp.down_w.data
            sd[p + "mlp.gate.weight"] = layer.mlp.router_w.data
            sd[p + "mlp.gate.e_score_correction_bias"] = \
                layer.mlp.router_bias.data
            si = spec_.moe_intermediate_size * spec_.n_shared_experts
            sgu = layer.mlp.shared_gate_up_w.data
            sd[p + "mlp.shared_experts.gate_proj.weight"] = sgu[:si]
            sd[p + "mlp.shared_experts.up_proj.weight"] = sgu[si:]
            sd[p + "mlp.shared_experts.down_proj.weight"] = \
                layer.mlp.shared_down_w.data
        else:  # dense first layer
            gu = layer.mlp.gate_up_w.data
            ii = spec_.intermediate_size
            sd[p + "mlp.gate_proj.weight"] = gu[:ii]
            sd[p + "mlp.up_proj.weight"] = gu[ii:]
            sd[p + "mlp.down_proj.weight"] = layer.mlp.down_w.data
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


def test_dots1_matches_hf_transformers_logits():
    eng = _engine()
    assert not hasattr(eng.runner.model.layers[0].mlp, "router_w")
    assert hasattr(eng.runner.model.layers[1].mlp, "router_w")
    hf = _hf_from(eng)
    prompt = [3, 1, 4, 1, 5, 9, 2, 6, 5, 3]
    with torch.inference_mode():
        want = hf(torch.tensor([prompt])).logits[0, -1]
    got = _prefill_logits(eng, prompt)
    assert torch.allclose(got, want, atol=3e-4, rtol=1e-3), \
        (got - want).abs().max()


def test_dots1_decode_matches_hf_generation():
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


def test_dots1_spec_from_hf_config():
    spec = ModelSpec.from_hf_config({
        "architectures": ["Dots1ForCausalLM"], "vocab_size": 152064,
        "hidden_size": 4608, "intermediate_size": 10944,
        "num_hidden_layers": 62, "num_attention_heads": 32,
        "num_key_value_heads": 32, "head_dim": 128,
        "num_local_experts": 128, "num_experts_per_tok": 6,
        "moe_intermediate_size": 1408, "n_shared_experts": 2,
        "n_group": 1, "topk_group": 1, "routed_scaling_factor": 2.5,
        "norm_topk_prob": True, "first_k_dense_replace": 1,
        "rope_theta": 10000000.0, "max_position_embeddings": 32768,
    })  # dots.llm1 142B config shape
    assert spec.qk_norm and not spec.qk_norm_full
    assert spec.router_mode == "sigmoid_bias"
    assert spec.num_experts == 128 and spec.n_shared_experts == 2
    assert spec.routed_scaling_factor == 2.5
    assert spec.first_k_dense_replace == 1
