"""Phi-3 / Phi-4 family (Phi3ForCausalLM): fused qkv/gate_up projections
and LongRoPE scaling — logits-exact vs HF transformers at fp32 on CPU."""
import dataclasses

import pytest
import torch

import gpustack_amd.engine.config as C
from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams
from gpustack_amd.engine.config import ModelSpec

TINY_PHI = ModelSpec(
    architecture="Phi3ForCausalLM", vocab_size=512, hidden_size=128,
    intermediate_size=256, num_layers=3, num_heads=4, num_kv_heads=2,
    head_dim=32, max_position_embeddings=512, rope_theta=10000.0,
    eos_token_id=1,
)


@pytest.fixture(autouse=True)
def _tiny_phi_preset():
    C.PRESETS["tiny-phi"] = dataclasses.replace(TINY_PHI)
    yield
    C.PRESETS.pop("tiny-phi", None)


def _engine(**kw):
    spec_over = kw.pop("spec_over", {})
    if spec_over:
        C.PRESETS["tiny-phi"] = dataclasses.replace(TINY_PHI, **spec_over)
    return LLMEngine(EngineConfig(model="tiny-phi", device="cpu",
                                  dtype="float32", kv_cache_blocks=64, **kw))


def _hf_from(eng, rope_parameters=None):
    from transformers import Phi3Config, Phi3ForCausalLM

    spec = eng.cfg.spec
    kw = {}
    if rope_parameters is not None:
        kw["rope_parameters"] = rope_parameters
    hf_cfg = Phi3Config(
        vocab_size=spec.vocab_size, hidden_size=spec.hidden_size,
        intermediate_size=spec.intermediate_size,
        num_hidden_layers=spec.num_layers,
        num_attention_heads=spec.num_heads,
        num_key_value_heads=spec.num_kv_heads,
        rms_norm_eps=spec.rms_norm_eps,
        max_position_embeddings=spec.max_position_embeddings,
        original_max_position_embeddings=32,
        rope_theta=spec.rope_theta, tie_word_embeddings=False,
        eos_token_id=1, pad_token_id=0, bos_token_id=2,
        attn_implementation="eager", **kw,
    )
    hf = Phi3ForCausalLM(hf_cfg).eval().float()
    m = eng.runner.model
    sd = {
        "model.embed_tokens.weight": m.embed.data,
        "model.norm.weight": m.final_norm.data,
        "lm_head.weight": m.lm_head.data,
    }
    for li, layer in enumerate(m.layers):
        p = f"model.layers.{li}."
        # our fused layouts ARE the phi checkpoint layouts
        sd[p + "self_attn.qkv_proj.weight"] = layer.attn.qkv_w.data
        sd[p + "self_attn.o_proj.weight"] = layer.attn.o_w.data
        sd[p + "mlp.gate_up_proj.weight"] = layer.mlp.gate_up_w.data
        sd[p + "mlp.down_proj.weight"] = layer.mlp.down_w.data
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
    seq.block_table = eng.scheduler.kv.allocator.allocate(4)
    batch = ScheduledBatch(
        is_prefill=True, seqs=[seq], token_ids=prompt,
        positions=list(range(len(prompt))),
        slot_mapping=eng.scheduler.kv.slots_for(seq.block_table, 0,
                                                len(prompt)),
        seq_lens=[len(prompt)],
    )
    tokens, meta = eng.runner._meta(batch)
    return eng.runner.model(tokens, meta, eng.runner.kv)[0]


def test_phi3_matches_hf_transformers_logits():
    eng = _engine()
    hf = _hf_from(eng)
    prompt = [3, 1, 4, 1, 5, 9, 2, 6, 5, 3]
    with torch.inference_mode():
        want = hf(torch.tensor([prompt])).logits[0, -1]
    got = _prefill_logits(eng, prompt)
    assert torch.allclose(got, want, atol=3e-4, rtol=1e-3), \
        (got - want).abs().max()


def test_phi3_decode_matches_hf_generation():
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


def test_phi3_longrope_matches_hf_beyond_original():
    """LongRoPE long-factor regime: prompt longer than the pretraining
    length (32) makes HF's dynamic rope pick the long factors — our
    static max_model_len-sized cache must agree there."""
    half = 16  # rot_dim/2 = head_dim/2
    long_f = [1.0 + 0.25 * i for i in range(half)]
    short_f = [1.0] * half
    scaling = {"rope_type": "longrope", "long_factor": long_f,
               "short_factor": short_f,
               "original_max_position_embeddings": 32}
    eng = _engine(spec_over={"rope_scaling": dict(scaling)})
    hf = _hf_from(eng, rope_parameters={
        "rope_type": "longrope", "long_factor": long_f,
        "short_factor": short_f, "rope_theta": 10000.0,
    })
    prompt = list(range(2, 50))  # 48 tokens > original 32
    with torch.inference_mode():
        want = hf(torch.tensor([prompt])).logits[0, -1]
    got = _prefill_logits(eng, prompt)
    assert torch.allclose(got, want, atol=4e-4, rtol=1e-3), \
        (got - want).abs().max()


def test_phi_spec_from_hf_config():
    spec = ModelSpec.from_hf_config({
        "architectures": ["Phi3ForCausalLM"], "vocab_size": 32064,
        "hidden_size": 3072, "intermediate_size": 8192,
        "num_hidden_layers": 32, "num_attention_heads": 32,
        "num_key_value_heads": 32, "rope_theta": 10000.0,
        "max_position_embeddings": 131072,
        "original_max_position_embeddings": 4096,
        "rope_scaling": {"type": "longrope",
                         "long_factor": [1.0] * 48,
                         "short_factor": [1.0] * 48},
    })  # phi-3-mini-128k config shape
    assert spec.rope_scaling["original_max_position_embeddings"] == 4096
    assert spec.rope_scaling["type"] == "longrope"


def test_phi_fused_checkpoint_loader_roundtrip(tmp_path):
    """Phi checkpoints store FUSED qkv_proj/gate_up_proj tensors; the
    loader splits them — logits match HF after loading from disk."""
    from safetensors.torch import save_file

    from gpustack_amd.models.weights import load_safetensors

    eng = _engine()
    hf = _hf_from(eng)
    save_file({k: v.contiguous().clone() for k, v in hf.state_dict().items()},
              str(tmp_path / "model.safetensors"))
    eng2 = _engine(seed=99)
    load_safetensors(eng2.runner.model, eng2.cfg, tmp_path)
    prompt = [9, 8, 7, 3, 2, 6]
    with torch.inference_mode():
        want = hf(torch.tensor([prompt])).logits[0, -1]
    got = _prefill_logits(eng2, prompt)
    assert torch.allclose(got, want, atol=3e-4, rtol=1e-3), \
        (got - want).abs().max()
