"""OLMo-3 family (Olmo3ForCausalLM): OLMo-2's norm-after flow and
full-projection qk-norm PLUS 3:1 sliding-window layers with per-layer-
type rope (sliding layers rope at their own theta, unscaled — the
Gemma-3 dual-cache path) — logits-exact vs HF transformers."""
import dataclasses

import pytest
import torch

import gpustack_amd.engine.config as C
from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams
from gpustack_amd.engine.config import ModelSpec

_LT = tuple("sliding_attention" if (i + 1) % 4 else "full_attention"
            for i in range(4))

TINY_OLMO3 = ModelSpec(
    architecture="Olmo3ForCausalLM", vocab_size=512, hidden_size=128,
    intermediate_size=256, num_layers=4, num_heads=4, num_kv_heads=2,
    head_dim=32, max_position_embeddings=512, rope_theta=10000.0,
    rms_norm_eps=1e-6, eos_token_id=1, qk_norm=True, norm_after=True,
    qk_norm_full=True, sliding_window=8, layer_types=_LT,
    rope_local_theta=5000.0,  # sliding layers rope at a DIFFERENT base
)


@pytest.fixture(autouse=True)
def _preset():
    C.PRESETS["tiny-olmo3"] = dataclasses.replace(TINY_OLMO3)
    yield
    C.PRESETS.pop("tiny-olmo3", None)


def _engine(**kw):
    return LLMEngine(EngineConfig(model="tiny-olmo3", device="cpu",
                                  dtype="float32", kv_cache_blocks=64, **kw))


def _hf_from(eng):
    from transformers import Olmo3Config, Olmo3ForCausalLM

    spec = eng.cfg.spec
    hf_cfg = Olmo3Config(
        vocab_size=spec.vocab_size, hidden_size=spec.hidden_size,
        intermediate_size=spec.intermediate_size,
        num_hidden_layers=spec.num_layers,
        num_attention_heads=spec.num_heads,
        num_key_value_heads=spec.num_kv_heads,
        rms_norm_eps=spec.rms_norm_eps,
        max_position_embeddings=spec.max_position_embeddings,
        rope_parameters={
            "full_attention": {"rope_type": "default",
                               "rope_theta": spec.rope_theta},
            "sliding_attention": {"rope_type": "default",
                                  "rope_theta": spec.rope_local_theta},
        },
        sliding_window=spec.sliding_window, layer_types=list(_LT),
        tie_word_embeddings=False, attention_bias=False,
        eos_token_id=1, pad_token_id=0, attn_implementation="eager",
    )
    hf = Olmo3ForCausalLM(hf_cfg).eval().float()
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
        sd[p + "self_attn.q_norm.weight"] = layer.attn.q_norm.data
        sd[p + "self_attn.k_norm.weight"] = layer.attn.k_norm.data
        # norm-after flow: our input_norm slot holds HF's
        # post_attention_layernorm, post_attn_norm holds
        # post_feedforward_layernorm (same as OLMo-2)
        sd[p + "post_attention_layernorm.weight"] = layer.input_norm.data
        sd[p + "post_feedforward_layernorm.weight"] = \
            layer.post_attn_norm.data
        gu = layer.mlp.gate_up_w.data
        ii = spec.intermediate_size
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


def test_olmo3_matches_hf_transformers_logits():
    eng = _engine()
    m = eng.runner.model
    assert m.layers[0].attn.window == 8      # sliding layer
    assert m.layers[3].attn.window == 0      # every 4th layer is full
    assert m.cos_sin_local is not None       # dual rope caches built
    hf = _hf_from(eng)
    # prompt LONGER than the window so the sliding mask actually binds
    prompt = [3, 1, 4, 1, 5, 9, 2, 6, 5, 3, 5, 8, 9, 7]
    with torch.inference_mode():
        want = hf(torch.tensor([prompt])).logits[0, -1]
    got = _prefill_logits(eng, prompt)
    assert torch.allclose(got, want, atol=3e-4, rtol=1e-3), \
        (got - want).abs().max()


def test_olmo3_decode_matches_hf_generation():
    eng = _engine()
    hf = _hf_from(eng)
    prompt = [2, 7, 1, 8, 2, 8, 1, 8, 2, 8]
    with torch.inference_mode():
        out = hf.generate(torch.tensor([prompt]), max_new_tokens=8,
                          do_sample=False, eos_token_id=None)
    want = out[0, len(prompt):].tolist()
    got = eng.generate([prompt], SamplingParams(max_tokens=8,
                                                ignore_eos=True))[0]
    assert got == want


def test_olmo3_spec_from_hf_config():
    spec = ModelSpec.from_hf_config({
        "architectures": ["Olmo3ForCausalLM"], "vocab_size": 100352,
        "hidden_size": 4096, "intermediate_size": 11008,
        "num_hidden_layers": 32, "num_attention_heads": 32,
        "num_key_value_heads": 32, "sliding_window": 4096,
        "layer_types": ["sliding_attention" if (i + 1) % 4 else
                        "full_attention" for i in range(32)],
        "rope_parameters": {
            "full_attention": {"rope_type": "default",
                               "rope_theta": 500000.0},
            "sliding_attention": {"rope_type": "default",
                                  "rope_theta": 500000.0},
        },
        "max_position_embeddings": 65536,
    })  # olmo-3-7b config shape
    assert spec.norm_after and spec.qk_norm_full and spec.qk_norm
    assert spec.sliding_window == 4096
    assert spec.layer_types[0] == "sliding_attention"
    assert spec.layer_types[3] == "full_attention"
    assert spec.rope_theta == 500000.0
    assert spec.rope_local_theta == 500000.0  # sliding cache, unscaled
