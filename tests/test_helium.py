"""Helium-1 family (HeliumForCausalLM, Kyutai): llama graph with
PAIRWISE (GPT-J interleaved) rotary — logits-exact vs HF transformers.
Also carries the property test that a truly UNKNOWN arch string maps
to plain llama defaults."""
import dataclasses

import pytest
import torch

import gpustack_amd.engine.config as C
from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams
from gpustack_amd.engine.config import ModelSpec

TINY_HELIUM = ModelSpec(
    architecture="HeliumForCausalLM", vocab_size=512, hidden_size=128,
    intermediate_size=256, num_layers=3, num_heads=4, num_kv_heads=2,
    head_dim=32, max_position_embeddings=512, rope_theta=100000.0,
    rms_norm_eps=1e-8, eos_token_id=1, rope_mode="pairwise",
)


@pytest.fixture(autouse=True)
def _preset():
    C.PRESETS["tiny-helium"] = dataclasses.replace(TINY_HELIUM)
    yield
    C.PRESETS.pop("tiny-helium", None)


def _engine(**kw):
    return LLMEngine(EngineConfig(model="tiny-helium", device="cpu",
                                  dtype="float32", kv_cache_blocks=64, **kw))


def _hf_from(eng):
    from transformers import HeliumConfig, HeliumForCausalLM

    spec = eng.cfg.spec
    hf_cfg = HeliumConfig(
        vocab_size=spec.vocab_size, hidden_size=spec.hidden_size,
        intermediate_size=spec.intermediate_size,
        num_hidden_layers=spec.num_layers,
        num_attention_heads=spec.num_heads,
        num_key_value_heads=spec.num_kv_heads,
        head_dim=spec.head_dim, rms_norm_eps=spec.rms_norm_eps,
        max_position_embeddings=spec.max_position_embeddings,
        rope_theta=spec.rope_theta, tie_word_embeddings=False,
        attention_bias=False, mlp_bias=False,
        eos_token_id=1, pad_token_id=0, attn_implementation="eager",
    )
    hf = HeliumForCausalLM(hf_cfg).eval().float()
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
        sd[p + "self_attn.o_proj.weight"] = layer.attn.o_w.data
        sd[p + "input_layernorm.weight"] = layer.input_norm.data
        sd[p + "post_attention_layernorm.weight"] = layer.post_attn_norm.data
        gu = layer.mlp.gate_up_w.data
        ii = spec.intermediate_size
        sd[p + "mlp.gate_proj.weight"] = gu[:ii]
        sd[p + "mlp.up_proj.weight"] = gu[ii:]
        sd[p + "mlp.down_proj.weight"] = layer.mlp.down_w.data
    missing, unexpected = hf.load_state_dict(sd, strict=False)
    assert not unexpected, unexpected
    assert all("rotary" in k for k in missing), missing
    return hf


def test_helium_matches_hf_transformers_logits():
    eng = _engine()
    hf = _hf_from(eng)
    prompt = [3, 1, 4, 1, 5, 9, 2, 6, 5, 3]
    with torch.inference_mode():
        want = hf(torch.tensor([prompt])).logits[0, -1]
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
    got = eng.runner.model(tokens, meta, eng.runner.kv)[0]
    assert torch.allclose(got, want, atol=3e-4, rtol=1e-3), \
        (got - want).abs().max()


def test_helium_decode_matches_hf_generation():
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


def test_unknown_llama_like_arch_maps_to_plain_defaults():
    """The load-bearing property: an arch string from_hf_config has
    never seen maps to the plain llama graph (no special modes)."""
    spec = ModelSpec.from_hf_config({
        "architectures": ["TotallyNovelForCausalLM"], "vocab_size": 48000,
        "hidden_size": 2560, "intermediate_size": 7040,
        "num_hidden_layers": 24, "num_attention_heads": 20,
        "num_key_value_heads": 20, "head_dim": 128,
        "rope_theta": 100000.0, "max_position_embeddings": 4096,
        "rms_norm_eps": 1e-8,
    })
    assert not (spec.qk_norm or spec.norm_after or spec.sandwich_norms
                or spec.parallel_block or spec.mlp_no_gate
                or spec.attention_bias or spec.sliding_window
                or spec.no_rope_layers)
    assert spec.rope_mode == "neox" and spec.norm_type == "rmsnorm"
    assert spec.num_experts == 0

    helium = ModelSpec.from_hf_config(
        {"architectures": ["HeliumForCausalLM"]})
    assert helium.rope_mode == "pairwise"  # interleaved rotate_half
