"""Gemma-2 family (Gemma2ForCausalLM): sandwich layer norms ((1+w)
RMSNorm before the residual adds), sqrt(h)-scaled tied embeddings,
attention + final logit softcapping, custom attention scale, GeGLU MLP,
alternating sliding-window layers — logits-exact vs HF transformers at
fp32 on CPU (the family oracle; the D-256 + softcap CDNA4 kernel work is
r3 and verifies against these tests)."""
import dataclasses

import pytest
import torch

import gpustack_amd.engine.config as C
from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams
from gpustack_amd.engine.config import ModelSpec

TINY_G2 = ModelSpec(
    architecture="Gemma2ForCausalLM", vocab_size=512, hidden_size=128,
    intermediate_size=256, num_layers=4, num_heads=4, num_kv_heads=2,
    head_dim=32, max_position_embeddings=512, rope_theta=10000.0,
    rms_norm_eps=1e-6, eos_token_id=1, tie_word_embeddings=True,
    sliding_window=8, sandwich_norms=True, embed_scale=128 ** 0.5,
    attn_logit_softcap=50.0, final_logit_softcap=30.0,
    attn_scale=24 ** -0.5, mlp_act="gelu_tanh",
)


@pytest.fixture(autouse=True)
def _tiny_g2_preset():
    C.PRESETS["tiny-g2"] = dataclasses.replace(TINY_G2)
    yield
    C.PRESETS.pop("tiny-g2", None)


def _engine(**kw):
    return LLMEngine(EngineConfig(model="tiny-g2", device="cpu",
                                  dtype=kw.pop("dtype", "float32"),
                                  kv_cache_blocks=64, **kw))


def _hf_from(eng):
    from transformers import Gemma2Config, Gemma2ForCausalLM

    spec = eng.cfg.spec
    hf_cfg = Gemma2Config(
        vocab_size=spec.vocab_size, hidden_size=spec.hidden_size,
        intermediate_size=spec.intermediate_size,
        num_hidden_layers=spec.num_layers,
        num_attention_heads=spec.num_heads,
        num_key_value_heads=spec.num_kv_heads,
        head_dim=spec.head_dim, rms_norm_eps=spec.rms_norm_eps,
        max_position_embeddings=spec.max_position_embeddings,
        rope_theta=spec.rope_theta, tie_word_embeddings=True,
        attention_bias=False, sliding_window=spec.sliding_window,
        attn_logit_softcapping=spec.attn_logit_softcap,
        final_logit_softcapping=spec.final_logit_softcap,
        query_pre_attn_scalar=24, hidden_activation="gelu_pytorch_tanh",
        attn_implementation="eager",
    )
    assert hf_cfg.layer_types[0] == "sliding_attention"  # even-layer SWA
    hf = Gemma2ForCausalLM(hf_cfg).eval().float()
    m = eng.runner.model
    d = spec.head_dim
    nq, nk = spec.num_heads * d, spec.num_kv_heads * d
    sd = {
        "model.embed_tokens.weight": m.embed.data,
        "model.norm.weight": m.final_norm.data - 1,  # we store (1+w)
    }
    for li, layer in enumerate(m.layers):
        p = f"model.layers.{li}."
        qkv = layer.attn.qkv_w.data
        sd[p + "self_attn.q_proj.weight"] = qkv[:nq]
        sd[p + "self_attn.k_proj.weight"] = qkv[nq:nq + nk]
        sd[p + "self_attn.v_proj.weight"] = qkv[nq + nk:]
        sd[p + "self_attn.o_proj.weight"] = layer.attn.o_w.data
        sd[p + "input_layernorm.weight"] = layer.input_norm.data - 1
        sd[p + "post_attention_layernorm.weight"] = \
            layer.post_attn_norm.data - 1
        sd[p + "pre_feedforward_layernorm.weight"] = \
            layer.pre_ff_norm.data - 1
        sd[p + "post_feedforward_layernorm.weight"] = \
            layer.post_ff_norm.data - 1
        gu = layer.mlp.gate_up_w.data
        ii = spec.intermediate_size
        sd[p + "mlp.gate_proj.weight"] = gu[:ii]
        sd[p + "mlp.up_proj.weight"] = gu[ii:]
        sd[p + "mlp.down_proj.weight"] = layer.mlp.down_w.data
    missing, unexpected = hf.load_state_dict(sd, strict=False)
    assert not unexpected, unexpected
    assert all("rotary" in k or k == "lm_head.weight" for k in missing), \
        missing
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


def test_gemma2_matches_hf_transformers_logits():
    eng = _engine()
    hf = _hf_from(eng)
    # crosses the sliding window (8) so both layer types matter
    prompt = [3, 1, 4, 1, 5, 9, 2, 6, 5, 3, 5, 8, 9, 7]
    with torch.inference_mode():
        want = hf(torch.tensor([prompt])).logits[0, -1]
    got = _prefill_logits(eng, prompt)
    assert torch.allclose(got, want, atol=3e-4, rtol=1e-3), \
        (got - want).abs().max()


def test_gemma2_decode_matches_hf_generation():
    eng = _engine()
    hf = _hf_from(eng)
    prompt = [2, 7, 1, 8, 2, 8]
    n = 10  # decode crosses the window boundary
    with torch.inference_mode():
        out = hf.generate(torch.tensor([prompt]), max_new_tokens=n,
                          do_sample=False, eos_token_id=None)
    want = out[0, len(prompt):].tolist()
    got = eng.generate([prompt], SamplingParams(max_tokens=n,
                                                ignore_eos=True))[0]
    assert got == want


def test_gemma2_chunked_prefill_matches_plain():
    prompt = list(range(2, 40))
    p = SamplingParams(max_tokens=6, ignore_eos=True)
    plain = _engine(enable_chunked_prefill=False).generate([prompt], p)[0]
    chunked = _engine(enable_chunked_prefill=True,
                      max_prefill_tokens=16).generate([prompt], p)[0]
    assert chunked == plain


def test_gemma2_spec_from_hf_config():
    spec = ModelSpec.from_hf_config({
        "architectures": ["Gemma2ForCausalLM"], "vocab_size": 256000,
        "hidden_size": 3584, "intermediate_size": 14336,
        "num_hidden_layers": 42, "num_attention_heads": 16,
        "num_key_value_heads": 8, "head_dim": 256,
        "rope_theta": 10000.0, "max_position_embeddings": 8192,
        "tie_word_embeddings": True, "sliding_window": 4096,
        "attn_logit_softcapping": 50.0, "final_logit_softcapping": 30.0,
        "query_pre_attn_scalar": 224,
        "hidden_activation": "gelu_pytorch_tanh",
    })
    assert spec.sandwich_norms and spec.mlp_act == "gelu_tanh"
    assert spec.attn_logit_softcap == 50.0
    assert abs(spec.attn_scale - 224 ** -0.5) < 1e-9
    assert abs(spec.embed_scale - 3584 ** 0.5) < 1e-6
    assert spec.sliding_window == 4096


def test_gemma2_gpu_softcap_fails_loudly():
    import gpustack_amd.ops as O

    q = torch.randn(1, 4, 32)
    orig = O._backend
    O._backend = lambda t: object()
    try:
        with pytest.raises(NotImplementedError, match="softcap"):
            O.paged_attn_decode(q, q, q, q, None, None, 1.0, softcap=50.0)
        with pytest.raises(NotImplementedError, match="softcap"):
            O.varlen_prefill_attn(q, q, q, q, [1], 1.0, softcap=50.0)
    finally:
        O._backend = orig


TINY_G3 = dataclasses.replace(
    TINY_G2, architecture="Gemma3ForCausalLM", qk_norm=True,
    attn_logit_softcap=0.0, final_logit_softcap=0.0,
    rope_theta=1000000.0, rope_local_theta=10000.0,
    rope_scaling={"rope_type": "linear", "factor": 8.0},
    layer_types=("sliding_attention", "sliding_attention",
                 "sliding_attention", "full_attention"),  # 3:1 tiny pattern
)


@pytest.fixture()
def _tiny_g3_preset():
    C.PRESETS["tiny-g3"] = dataclasses.replace(TINY_G3)
    yield
    C.PRESETS.pop("tiny-g3", None)


def _engine_g3(**kw):
    return LLMEngine(EngineConfig(model="tiny-g3", device="cpu",
                                  dtype="float32", kv_cache_blocks=64, **kw))


def _hf_g3(eng):
    from transformers import Gemma3ForCausalLM, Gemma3TextConfig

    spec = eng.cfg.spec
    hf_cfg = Gemma3TextConfig(
        vocab_size=spec.vocab_size, hidden_size=spec.hidden_size,
        intermediate_size=spec.intermediate_size,
        num_hidden_layers=spec.num_layers,
        num_attention_heads=spec.num_heads,
        num_key_value_heads=spec.num_kv_heads,
        head_dim=spec.head_dim, rms_norm_eps=spec.rms_norm_eps,
        max_position_embeddings=spec.max_position_embeddings,
        tie_word_embeddings=True, attention_bias=False,
        sliding_window=spec.sliding_window,
        layer_types=list(spec.layer_types),
        query_pre_attn_scalar=24, hidden_activation="gelu_pytorch_tanh",
        rope_parameters={
            "sliding_attention": {"rope_type": "default",
                                  "rope_theta": spec.rope_local_theta},
            "full_attention": {"rope_type": "linear", "factor": 8.0,
                               "rope_theta": spec.rope_theta},
        },
        attn_implementation="eager",
    )
    hf = Gemma3ForCausalLM(hf_cfg).eval().float()
    m = eng.runner.model
    d = spec.head_dim
    nq, nk = spec.num_heads * d, spec.num_kv_heads * d
    sd = {
        "model.embed_tokens.weight": m.embed.data,
        "model.norm.weight": m.final_norm.data - 1,
    }
    for li, layer in enumerate(m.layers):
        p = f"model.layers.{li}."
        qkv = layer.attn.qkv_w.data
        sd[p + "self_attn.q_proj.weight"] = qkv[:nq]
        sd[p + "self_attn.k_proj.weight"] = qkv[nq:nq + nk]
        sd[p + "self_attn.v_proj.weight"] = qkv[nq + nk:]
        sd[p + "self_attn.o_proj.weight"] = layer.attn.o_w.data
        sd[p + "self_attn.q_norm.weight"] = layer.attn.q_norm.data - 1
        sd[p + "self_attn.k_norm.weight"] = layer.attn.k_norm.data - 1
        sd[p + "input_layernorm.weight"] = layer.input_norm.data - 1
        sd[p + "post_attention_layernorm.weight"] = \
            layer.post_attn_norm.data - 1
        sd[p + "pre_feedforward_layernorm.weight"] = \
            layer.pre_ff_norm.data - 1
        sd[p + "post_feedforward_layernorm.weight"] = \
            layer.post_ff_norm.data - 1
        gu = layer.mlp.gate_up_w.data
        ii = spec.intermediate_size
        sd[p + "mlp.gate_proj.weight"] = gu[:ii]
        sd[p + "mlp.up_proj.weight"] = gu[ii:]
        sd[p + "mlp.down_proj.weight"] = layer.mlp.down_w.data
    missing, unexpected = hf.load_state_dict(sd, strict=False)
    assert not unexpected, unexpected
    assert all("rotary" in k or k == "lm_head.weight" for k in missing), \
        missing
    return hf


def test_gemma3_matches_hf_transformers_logits(_tiny_g3_preset):
    """Gemma-3: dual rope (local theta on sliding layers, linear-scaled
    global theta on full layers), (1+w) qk-norm, no softcapping."""
    eng = _engine_g3()
    hf = _hf_g3(eng)
    prompt = [3, 1, 4, 1, 5, 9, 2, 6, 5, 3, 5, 8, 9, 7]
    with torch.inference_mode():
        want = hf(torch.tensor([prompt])).logits[0, -1]
    got = _prefill_logits(eng, prompt)
    assert torch.allclose(got, want, atol=3e-4, rtol=1e-3), \
        (got - want).abs().max()


def test_gemma3_decode_matches_hf_generation(_tiny_g3_preset):
    eng = _engine_g3()
    hf = _hf_g3(eng)
    prompt = [2, 7, 1, 8, 2, 8]
    with torch.inference_mode():
        out = hf.generate(torch.tensor([prompt]), max_new_tokens=10,
                          do_sample=False, eos_token_id=None)
    want = out[0, len(prompt):].tolist()
    got = eng.generate([prompt], SamplingParams(max_tokens=10,
                                                ignore_eos=True))[0]
    assert got == want


def test_gemma3_spec_from_hf_config():
    spec = ModelSpec.from_hf_config({
        "architectures": ["Gemma3ForCausalLM"], "vocab_size": 262208,
        "hidden_size": 2560, "intermediate_size": 10240,
        "num_hidden_layers": 34, "num_attention_heads": 8,
        "num_key_value_heads": 4, "head_dim": 256,
        "rope_theta": 1000000.0, "rope_local_base_freq": 10000.0,
        "max_position_embeddings": 131072, "tie_word_embeddings": True,
        "sliding_window": 1024, "sliding_window_pattern": 6,
        "query_pre_attn_scalar": 256,
        "rope_scaling": {"rope_type": "linear", "factor": 8.0},
    })  # gemma-3-4b-it config shape
    assert spec.sandwich_norms and spec.qk_norm
    assert spec.rope_local_theta == 10000.0
    assert spec.attn_logit_softcap == 0.0
    assert spec.layer_types is not None
    assert spec.layer_types[5] == "full_attention"
    assert spec.layer_types[0] == "sliding_attention"
    assert sum(t == "full_attention" for t in spec.layer_types) == 5
