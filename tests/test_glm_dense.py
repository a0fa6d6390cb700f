"""GLM dense families: GlmForCausalLM (glm-4-9b lineage: partial 0.5
PAIRWISE rotary + qkv bias, standard two-norm flow) and
Glm4ForCausalLM (GLM-4-0414: adds post_self_attn / post_mlp norms —
the four-norm sandwich flow with plain RMSNorm weights) — logits-exact
vs HF transformers on CPU. The MoE sibling (Glm4Moe) is covered in
test_glm_moe.py."""
import dataclasses

import pytest
import torch

import gpustack_amd.engine.config as C
from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams
from gpustack_amd.engine.config import ModelSpec

TINY_GLM = ModelSpec(
    architecture="GlmForCausalLM", vocab_size=512, hidden_size=128,
    intermediate_size=256, num_layers=3, num_heads=4, num_kv_heads=2,
    head_dim=32, max_position_embeddings=512, rope_theta=10000.0,
    eos_token_id=1, attention_bias=True, partial_rotary_factor=0.5,
    rope_mode="pairwise",
)

TINY_GLM4 = dataclasses.replace(
    TINY_GLM, architecture="Glm4ForCausalLM", sandwich_norms=True)


@pytest.fixture(autouse=True)
def _presets():
    C.PRESETS["tiny-glm"] = dataclasses.replace(TINY_GLM)
    C.PRESETS["tiny-glm4"] = dataclasses.replace(TINY_GLM4)
    yield
    C.PRESETS.pop("tiny-glm", None)
    C.PRESETS.pop("tiny-glm4", None)


def _engine(model, **kw):
    return LLMEngine(EngineConfig(model=model, device="cpu",
                                  dtype="float32", kv_cache_blocks=64, **kw))


def _base_sd(eng):
    spec = eng.cfg.spec
    m = eng.runner.model
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
        qkv_b = layer.attn.qkv_b.data
        sd[p + "self_attn.q_proj.bias"] = qkv_b[:nq]
        sd[p + "self_attn.k_proj.bias"] = qkv_b[nq:nq + nk]
        sd[p + "self_attn.v_proj.bias"] = qkv_b[nq + nk:]
        sd[p + "self_attn.o_proj.weight"] = layer.attn.o_w.data
        # HF Glm/Glm4 fuse the gate+up projection ([gate; up] rows —
        # same layout as our fused tensor)
        sd[p + "mlp.gate_up_proj.weight"] = layer.mlp.gate_up_w.data
        sd[p + "mlp.down_proj.weight"] = layer.mlp.down_w.data
    return sd


def _hf_glm(eng):
    from transformers import GlmConfig, GlmForCausalLM

    spec = eng.cfg.spec
    hf_cfg = GlmConfig(
        vocab_size=spec.vocab_size, hidden_size=spec.hidden_size,
        intermediate_size=spec.intermediate_size,
        num_hidden_layers=spec.num_layers,
        num_attention_heads=spec.num_heads,
        num_key_value_heads=spec.num_kv_heads,
        head_dim=spec.head_dim, rms_norm_eps=spec.rms_norm_eps,
        max_position_embeddings=spec.max_position_embeddings,
        rope_theta=spec.rope_theta, tie_word_embeddings=False,
        attention_bias=True, partial_rotary_factor=0.5,
        eos_token_id=1, pad_token_id=0, attn_implementation="eager",
    )
    hf = GlmForCausalLM(hf_cfg).eval().float()
    m = eng.runner.model
    sd = _base_sd(eng)
    for li, layer in enumerate(m.layers):
        p = f"model.layers.{li}."
        sd[p + "input_layernorm.weight"] = layer.input_norm.data
        sd[p + "post_attention_layernorm.weight"] = layer.post_attn_norm.data
    missing, unexpected = hf.load_state_dict(sd, strict=False)
    assert not unexpected, unexpected
    assert all("rotary" in k for k in missing), missing
    return hf


def _hf_glm4(eng):
    from transformers import Glm4Config, Glm4ForCausalLM

    spec = eng.cfg.spec
    hf_cfg = Glm4Config(
        vocab_size=spec.vocab_size, hidden_size=spec.hidden_size,
        intermediate_size=spec.intermediate_size,
        num_hidden_layers=spec.num_layers,
        num_attention_heads=spec.num_heads,
        num_key_value_heads=spec.num_kv_heads,
        head_dim=spec.head_dim, rms_norm_eps=spec.rms_norm_eps,
        max_position_embeddings=spec.max_position_embeddings,
        rope_theta=spec.rope_theta, tie_word_embeddings=False,
        attention_bias=True, partial_rotary_factor=0.5,
        eos_token_id=1, pad_token_id=0, attn_implementation="eager",
    )
    hf = Glm4ForCausalLM(hf_cfg).eval().float()
    m = eng.runner.model
    sd = _base_sd(eng)
    for li, layer in enumerate(m.layers):
        p = f"model.layers.{li}."
        # four-norm sandwich flow, GLM-4 names
        sd[p + "input_layernorm.weight"] = layer.input_norm.data
        sd[p + "post_self_attn_layernorm.weight"] = \
            layer.post_attn_norm.data
        sd[p + "post_attention_layernorm.weight"] = layer.pre_ff_norm.data
        sd[p + "post_mlp_layernorm.weight"] = layer.post_ff_norm.data
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


@pytest.mark.parametrize("model,hf_fn", [("tiny-glm", _hf_glm),
                                         ("tiny-glm4", _hf_glm4)])
def test_glm_dense_matches_hf_transformers_logits(model, hf_fn):
    eng = _engine(model)
    hf = hf_fn(eng)
    prompt = [3, 1, 4, 1, 5, 9, 2, 6, 5, 3]
    with torch.inference_mode():
        want = hf(torch.tensor([prompt])).logits[0, -1]
    got = _prefill_logits(eng, prompt)
    assert torch.allclose(got, want, atol=3e-4, rtol=1e-3), \
        (got - want).abs().max()


@pytest.mark.parametrize("model,hf_fn", [("tiny-glm", _hf_glm),
                                         ("tiny-glm4", _hf_glm4)])
def test_glm_dense_decode_matches_hf_generation(model, hf_fn):
    eng = _engine(model)
    hf = hf_fn(eng)
    prompt = [2, 7, 1, 8, 2, 8]
    with torch.inference_mode():
        out = hf.generate(torch.tensor([prompt]), max_new_tokens=8,
                          do_sample=False, eos_token_id=None)
    want = out[0, len(prompt):].tolist()
    got = eng.generate([prompt], SamplingParams(max_tokens=8,
                                                ignore_eos=True))[0]
    assert got == want


def test_glm4_spec_from_hf_config():
    spec = ModelSpec.from_hf_config({
        "architectures": ["Glm4ForCausalLM"], "vocab_size": 151552,
        "hidden_size": 4096, "intermediate_size": 13696,
        "num_hidden_layers": 40, "num_attention_heads": 32,
        "num_key_value_heads": 2, "head_dim": 128,
        "partial_rotary_factor": 0.5, "attention_bias": True,
        "rope_theta": 10000.0, "max_position_embeddings": 32768,
    })  # glm-4-9b-0414 config shape
    assert spec.sandwich_norms and spec.attention_bias
    assert spec.rope_mode == "pairwise"
    assert spec.partial_rotary_factor == 0.5

    moe = ModelSpec.from_hf_config({
        "architectures": ["Glm4MoeForCausalLM"], "vocab_size": 151552,
        "hidden_size": 4096, "num_hidden_layers": 4,
        "num_attention_heads": 32, "num_key_value_heads": 2,
        "head_dim": 128, "n_routed_experts": 8,
        "num_experts_per_tok": 2, "moe_intermediate_size": 128,
    })  # the MoE sibling keeps NEOX rope + no sandwich (unchanged)
    assert not moe.sandwich_norms and moe.rope_mode == "neox"
    assert not moe.attention_bias
