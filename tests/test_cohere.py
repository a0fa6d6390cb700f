"""Cohere families (Command-R / Command-R7B): mean-centered LayerNorm
(no RMS), PARALLEL residual block (one shared input norm feeds attn AND
mlp, outputs added together), pairwise rotary, logit_scale multiplier,
tied embeddings. Cohere2 adds 3:1 sliding-window layers with NoPE full
layers — logits-exact vs HF transformers on CPU."""
import dataclasses

import pytest
import torch

import gpustack_amd.engine.config as C
from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams
from gpustack_amd.engine.config import ModelSpec

TINY_COHERE = ModelSpec(
    architecture="CohereForCausalLM", vocab_size=512, hidden_size=128,
    intermediate_size=256, num_layers=3, num_heads=4, num_kv_heads=2,
    head_dim=32, max_position_embeddings=512, rope_theta=10000.0,
    rms_norm_eps=1e-5, eos_token_id=1, tie_word_embeddings=True,
    norm_type="layernorm", parallel_block=True, rope_mode="pairwise",
    logits_multiplier=0.0625,
)

_LT2 = tuple("sliding_attention" if (i + 1) % 4 else "full_attention"
             for i in range(4))

TINY_COHERE2 = dataclasses.replace(
    TINY_COHERE, architecture="Cohere2ForCausalLM", num_layers=4,
    sliding_window=8, layer_types=_LT2, no_rope_layers=(1, 1, 1, 0),
)


@pytest.fixture(autouse=True)
def _presets():
    C.PRESETS["tiny-cohere"] = dataclasses.replace(TINY_COHERE)
    C.PRESETS["tiny-cohere2"] = dataclasses.replace(TINY_COHERE2)
    yield
    C.PRESETS.pop("tiny-cohere", None)
    C.PRESETS.pop("tiny-cohere2", None)


def _engine(model, **kw):
    return LLMEngine(EngineConfig(model=model, device="cpu",
                                  dtype="float32", kv_cache_blocks=64, **kw))


def _load_sd(hf, eng):
    spec = eng.cfg.spec
    m = eng.runner.model
    d = spec.head_dim
    nq, nk = spec.num_heads * d, spec.num_kv_heads * d
    sd = {
        "model.embed_tokens.weight": m.embed.data,
        "model.norm.weight": m.final_norm.data,
    }
    for li, layer in enumerate(m.layers):
        p = f"model.layers.{li}."
        qkv = layer.attn.qkv_w.data
        sd[p + "self_attn.q_proj.weight"] = qkv[:nq]
        sd[p + "self_attn.k_proj.weight"] = qkv[nq:nq + nk]
        sd[p + "self_attn.v_proj.weight"] = qkv[nq + nk:]
        sd[p + "self_attn.o_proj.weight"] = layer.attn.o_w.data
        sd[p + "input_layernorm.weight"] = layer.input_norm.data
        gu = layer.mlp.gate_up_w.data
        ii = spec.intermediate_size
        sd[p + "mlp.gate_proj.weight"] = gu[:ii]
        sd[p + "mlp.up_proj.weight"] = gu[ii:]
        sd[p + "mlp.down_proj.weight"] = layer.mlp.down_w.data
    missing, unexpected = hf.load_state_dict(sd, strict=False)
    assert not unexpected, unexpected
    assert all("rotary" in k or "lm_head" in k for k in missing), missing
    hf.tie_weights()
    return hf


def _hf_cohere(eng):
    from transformers import CohereConfig, CohereForCausalLM

    spec = eng.cfg.spec
    hf_cfg = CohereConfig(
        vocab_size=spec.vocab_size, hidden_size=spec.hidden_size,
        intermediate_size=spec.intermediate_size,
        num_hidden_layers=spec.num_layers,
        num_attention_heads=spec.num_heads,
        num_key_value_heads=spec.num_kv_heads,
        layer_norm_eps=spec.rms_norm_eps,
        max_position_embeddings=spec.max_position_embeddings,
        rope_theta=spec.rope_theta, tie_word_embeddings=True,
        attention_bias=False, use_qk_norm=False,
        logit_scale=spec.logits_multiplier,
        eos_token_id=1, pad_token_id=0, bos_token_id=2,
        attn_implementation="eager",
    )
    return _load_sd(CohereForCausalLM(hf_cfg).eval().float(), eng)


def _hf_cohere2(eng):
    from transformers import Cohere2Config, Cohere2ForCausalLM

    spec = eng.cfg.spec
    hf_cfg = Cohere2Config(
        vocab_size=spec.vocab_size, hidden_size=spec.hidden_size,
        intermediate_size=spec.intermediate_size,
        num_hidden_layers=spec.num_layers,
        num_attention_heads=spec.num_heads,
        num_key_value_heads=spec.num_kv_heads,
        layer_norm_eps=spec.rms_norm_eps,
        max_position_embeddings=spec.max_position_embeddings,
        rope_theta=spec.rope_theta, tie_word_embeddings=True,
        sliding_window=spec.sliding_window, layer_types=list(_LT2),
        logit_scale=spec.logits_multiplier,
        eos_token_id=1, pad_token_id=0, bos_token_id=2,
        attn_implementation="eager",
    )
    return _load_sd(Cohere2ForCausalLM(hf_cfg).eval().float(), eng)


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


def test_cohere_matches_hf_transformers_logits():
    eng = _engine("tiny-cohere")
    hf = _hf_cohere(eng)
    prompt = [3, 1, 4, 1, 5, 9, 2, 6, 5, 3]
    with torch.inference_mode():
        want = hf(torch.tensor([prompt])).logits[0, -1]
    got = _prefill_logits(eng, prompt)
    assert torch.allclose(got, want, atol=3e-4, rtol=1e-3), \
        (got - want).abs().max()


def test_cohere_decode_matches_hf_generation():
    eng = _engine("tiny-cohere")
    hf = _hf_cohere(eng)
    prompt = [2, 7, 1, 8, 2, 8]
    with torch.inference_mode():
        out = hf.generate(torch.tensor([prompt]), max_new_tokens=8,
                          do_sample=False, eos_token_id=None)
    want = out[0, len(prompt):].tolist()
    got = eng.generate([prompt], SamplingParams(max_tokens=8,
                                                ignore_eos=True))[0]
    assert got == want


def test_cohere2_matches_hf_transformers_logits():
    eng = _engine("tiny-cohere2")
    m = eng.runner.model
    assert m.layers[0].attn.window == 8 and m.layers[0].attn.use_rope
    assert m.layers[3].attn.window == 0 and not m.layers[3].attn.use_rope
    hf = _hf_cohere2(eng)
    prompt = [3, 1, 4, 1, 5, 9, 2, 6, 5, 3, 5, 8, 9, 7]  # > window
    with torch.inference_mode():
        want = hf(torch.tensor([prompt])).logits[0, -1]
    got = _prefill_logits(eng, prompt)
    assert torch.allclose(got, want, atol=3e-4, rtol=1e-3), \
        (got - want).abs().max()


def test_cohere2_decode_matches_hf_generation():
    eng = _engine("tiny-cohere2")
    hf = _hf_cohere2(eng)
    prompt = [2, 7, 1, 8, 2, 8, 1, 8, 2, 8]
    with torch.inference_mode():
        out = hf.generate(torch.tensor([prompt]), max_new_tokens=8,
                          do_sample=False, eos_token_id=None)
    want = out[0, len(prompt):].tolist()
    got = eng.generate([prompt], SamplingParams(max_tokens=8,
                                                ignore_eos=True))[0]
    assert got == want


def test_cohere_spec_from_hf_config():
    spec = ModelSpec.from_hf_config({
        "architectures": ["CohereForCausalLM"], "vocab_size": 256000,
        "hidden_size": 8192, "intermediate_size": 22528,
        "num_hidden_layers": 40, "num_attention_heads": 64,
        "num_key_value_heads": 64, "rope_theta": 8000000.0,
        "layer_norm_eps": 1e-5, "logit_scale": 0.0625,
        "tie_word_embeddings": True, "max_position_embeddings": 131072,
    })  # command-r-v01 config shape
    assert spec.norm_type == "layernorm" and spec.parallel_block
    assert spec.rope_mode == "pairwise"
    assert spec.logits_multiplier == 0.0625
    assert spec.rms_norm_eps == 1e-5

    with pytest.raises(NotImplementedError, match="use_qk_norm"):
        ModelSpec.from_hf_config({
            "architectures": ["CohereForCausalLM"], "use_qk_norm": True})


def test_cohere_gpu_serving_refuses_without_gate(monkeypatch):
    """LayerNorm specs need the r3 kernel; GPU init must refuse loudly."""
    monkeypatch.delenv("GPUSTACK_AMD_OSS_KERNELS", raising=False)
    from gpustack_amd.engine.model_runner import ModelRunner

    cfg = EngineConfig(model="tiny-cohere", device="cuda",
                       kv_cache_blocks=16)
    with pytest.raises(NotImplementedError, match="layernorm"):
        ModelRunner(cfg)
