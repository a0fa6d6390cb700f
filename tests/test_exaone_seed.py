"""EXAONE-4 (Exaone4ForCausalLM) and Seed-OSS (SeedOssForCausalLM)
families — both pure compositions of existing mechanisms, logits-exact
vs HF transformers on CPU.

EXAONE-4: OLMo-2-style norm-after flow + Qwen3-style per-head qk-norm
(before rope) + hybrid global-NoPE (3:1 sliding layers rope, full
layers are unroped global attention).
Seed-OSS: llama graph with qkv bias (the Qwen2 bias pattern), no o bias.
"""
import dataclasses

import pytest
import torch

import gpustack_amd.engine.config as C
from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams
from gpustack_amd.engine.config import ModelSpec

_LT = tuple("sliding_attention" if (i + 1) % 4 else "full_attention"
            for i in range(4))

TINY_EXA = ModelSpec(
    architecture="Exaone4ForCausalLM", vocab_size=512, hidden_size=128,
    intermediate_size=256, num_layers=4, num_heads=4, num_kv_heads=2,
    head_dim=32, max_position_embeddings=512, rope_theta=10000.0,
    rms_norm_eps=1e-6, eos_token_id=1, qk_norm=True, norm_after=True,
    sliding_window=8, layer_types=_LT, no_rope_layers=(1, 1, 1, 0),
)

TINY_SEED = ModelSpec(
    architecture="SeedOssForCausalLM", vocab_size=512, hidden_size=128,
    intermediate_size=256, num_layers=3, num_heads=4, num_kv_heads=2,
    head_dim=32, max_position_embeddings=512, rope_theta=10000.0,
    eos_token_id=1, attention_bias=True,
)


@pytest.fixture(autouse=True)
def _presets():
    C.PRESETS["tiny-exa"] = dataclasses.replace(TINY_EXA)
    C.PRESETS["tiny-seed"] = dataclasses.replace(TINY_SEED)
    yield
    C.PRESETS.pop("tiny-exa", None)
    C.PRESETS.pop("tiny-seed", None)


def _engine(model, **kw):
    return LLMEngine(EngineConfig(model=model, device="cpu",
                                  dtype="float32", kv_cache_blocks=64, **kw))


def _common_sd(m, spec):
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
        gu = layer.mlp.gate_up_w.data
        ii = spec.intermediate_size
        sd[p + "mlp.gate_proj.weight"] = gu[:ii]
        sd[p + "mlp.up_proj.weight"] = gu[ii:]
        sd[p + "mlp.down_proj.weight"] = layer.mlp.down_w.data
    return sd


def _hf_exa(eng):
    from transformers import Exaone4Config, Exaone4ForCausalLM

    spec = eng.cfg.spec
    hf_cfg = Exaone4Config(
        vocab_size=spec.vocab_size, hidden_size=spec.hidden_size,
        intermediate_size=spec.intermediate_size,
        num_hidden_layers=spec.num_layers,
        num_attention_heads=spec.num_heads,
        num_key_value_heads=spec.num_kv_heads,
        head_dim=spec.head_dim, rms_norm_eps=spec.rms_norm_eps,
        max_position_embeddings=spec.max_position_embeddings,
        rope_parameters={"rope_type": "default",
                         "rope_theta": spec.rope_theta},
        sliding_window=spec.sliding_window,
        layer_types=list(spec.layer_types),
        tie_word_embeddings=False, eos_token_id=1, pad_token_id=0,
        attn_implementation="eager",
    )
    hf = Exaone4ForCausalLM(hf_cfg).eval().float()
    m = eng.runner.model
    sd = _common_sd(m, spec)
    for li, layer in enumerate(m.layers):
        p = f"model.layers.{li}."
        sd[p + "self_attn.q_norm.weight"] = layer.attn.q_norm.data
        sd[p + "self_attn.k_norm.weight"] = layer.attn.k_norm.data
        # norm-after: our input_norm slot holds post_attention_layernorm
        sd[p + "post_attention_layernorm.weight"] = layer.input_norm.data
        sd[p + "post_feedforward_layernorm.weight"] = \
            layer.post_attn_norm.data
    missing, unexpected = hf.load_state_dict(sd, strict=False)
    assert not unexpected, unexpected
    assert all("rotary" in k for k in missing), missing
    return hf


def _hf_seed(eng):
    from transformers import SeedOssConfig, SeedOssForCausalLM

    spec = eng.cfg.spec
    hf_cfg = SeedOssConfig(
        vocab_size=spec.vocab_size, hidden_size=spec.hidden_size,
        intermediate_size=spec.intermediate_size,
        num_hidden_layers=spec.num_layers,
        num_attention_heads=spec.num_heads,
        num_key_value_heads=spec.num_kv_heads,
        head_dim=spec.head_dim, rms_norm_eps=spec.rms_norm_eps,
        max_position_embeddings=spec.max_position_embeddings,
        rope_theta=spec.rope_theta, tie_word_embeddings=False,
        attention_bias=True, attention_out_bias=False, mlp_bias=False,
        eos_token_id=1, pad_token_id=0, attn_implementation="eager",
    )
    hf = SeedOssForCausalLM(hf_cfg).eval().float()
    m = eng.runner.model
    spec_ = spec
    sd = _common_sd(m, spec_)
    d = spec.head_dim
    nq, nk = spec.num_heads * d, spec.num_kv_heads * d
    for li, layer in enumerate(m.layers):
        p = f"model.layers.{li}."
        qkv_b = layer.attn.qkv_b.data
        sd[p + "self_attn.q_proj.bias"] = qkv_b[:nq]
        sd[p + "self_attn.k_proj.bias"] = qkv_b[nq:nq + nk]
        sd[p + "self_attn.v_proj.bias"] = qkv_b[nq + nk:]
        sd[p + "input_layernorm.weight"] = layer.input_norm.data
        sd[p + "post_attention_layernorm.weight"] = layer.post_attn_norm.data
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


def test_exaone4_matches_hf_transformers_logits():
    eng = _engine("tiny-exa")
    m = eng.runner.model
    assert m.layers[0].attn.window == 8 and m.layers[0].attn.use_rope
    assert m.layers[3].attn.window == 0 and not m.layers[3].attn.use_rope
    hf = _hf_exa(eng)
    prompt = [3, 1, 4, 1, 5, 9, 2, 6, 5, 3, 5, 8, 9, 7]  # > window
    with torch.inference_mode():
        want = hf(torch.tensor([prompt])).logits[0, -1]
    got = _prefill_logits(eng, prompt)
    assert torch.allclose(got, want, atol=3e-4, rtol=1e-3), \
        (got - want).abs().max()


def test_exaone4_decode_matches_hf_generation():
    eng = _engine("tiny-exa")
    hf = _hf_exa(eng)
    prompt = [2, 7, 1, 8, 2, 8, 1, 8, 2, 8]
    with torch.inference_mode():
        out = hf.generate(torch.tensor([prompt]), max_new_tokens=8,
                          do_sample=False, eos_token_id=None)
    want = out[0, len(prompt):].tolist()
    got = eng.generate([prompt], SamplingParams(max_tokens=8,
                                                ignore_eos=True))[0]
    assert got == want


def test_seed_oss_matches_hf_transformers_logits():
    eng = _engine("tiny-seed")
    hf = _hf_seed(eng)
    prompt = [3, 1, 4, 1, 5, 9, 2, 6, 5, 3]
    with torch.inference_mode():
        want = hf(torch.tensor([prompt])).logits[0, -1]
    got = _prefill_logits(eng, prompt)
    assert torch.allclose(got, want, atol=3e-4, rtol=1e-3), \
        (got - want).abs().max()


def test_seed_oss_decode_matches_hf_generation():
    eng = _engine("tiny-seed")
    hf = _hf_seed(eng)
    prompt = [2, 7, 1, 8, 2, 8]
    with torch.inference_mode():
        out = hf.generate(torch.tensor([prompt]), max_new_tokens=8,
                          do_sample=False, eos_token_id=None)
    want = out[0, len(prompt):].tolist()
    got = eng.generate([prompt], SamplingParams(max_tokens=8,
                                                ignore_eos=True))[0]
    assert got == want


def test_exaone4_and_seed_spec_from_hf_config():
    spec = ModelSpec.from_hf_config({
        "architectures": ["Exaone4ForCausalLM"], "vocab_size": 102400,
        "hidden_size": 5120, "num_hidden_layers": 8,
        "num_attention_heads": 40, "num_key_value_heads": 8,
        "head_dim": 128, "sliding_window": 4096,
        "layer_types": ["sliding_attention"] * 3 + ["full_attention"]
        + ["sliding_attention"] * 3 + ["full_attention"],
        "rope_parameters": {"rope_theta": 1000000.0,
                            "rope_type": "default"},
    })  # exaone-4.0-32b config shape
    assert spec.qk_norm and spec.norm_after and not spec.qk_norm_full
    assert spec.sliding_window == 4096
    assert spec.no_rope_layers == (1, 1, 1, 0, 1, 1, 1, 0)

    seed = ModelSpec.from_hf_config({
        "architectures": ["SeedOssForCausalLM"], "vocab_size": 155136,
        "hidden_size": 5120, "num_hidden_layers": 64,
        "num_attention_heads": 80, "num_key_value_heads": 8,
        "head_dim": 128, "rope_theta": 10000000.0,
        "attention_bias": True,
    })  # seed-oss-36b config shape
    assert seed.attention_bias and seed.sliding_window == 0
