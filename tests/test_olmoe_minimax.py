"""OLMoE (OlmoeForCausalLM) and MiniMax-M2 (MiniMaxM2ForCausalLM)
families — MoE models with FULL-PROJECTION qk-norm (the OLMo-2 norm
over the whole q/k projection, TP-group-reduced) on the standard
pre-norm llama flow — logits-exact vs HF transformers on CPU.

OLMoE: softmax router, norm_topk_prob=False (full-softmax probs kept).
MiniMax-M2: sigmoid scores + learned e_score_correction_bias for the
top-k CHOICE only, gathered sigmoid weights renormalized by their sum
(our sigmoid_bias router mode with n_group=1)."""
import dataclasses

import pytest
import torch

import gpustack_amd.engine.config as C
from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams
from gpustack_amd.engine.config import ModelSpec

TINY_OLMOE = ModelSpec(
    architecture="OlmoeForCausalLM", vocab_size=512, hidden_size=128,
    intermediate_size=256, num_layers=3, num_heads=4, num_kv_heads=2,
    head_dim=32, max_position_embeddings=512, rope_theta=10000.0,
    rms_norm_eps=1e-5, eos_token_id=1, qk_norm=True, qk_norm_full=True,
    num_experts=8, num_experts_per_tok=2, moe_intermediate_size=64,
    norm_topk_prob=False,
)

TINY_M2 = dataclasses.replace(
    TINY_OLMOE, architecture="MiniMaxM2ForCausalLM", rms_norm_eps=1e-6,
    router_mode="sigmoid_bias", norm_topk_prob=True,
)


@pytest.fixture(autouse=True)
def _presets():
    C.PRESETS["tiny-olmoe"] = dataclasses.replace(TINY_OLMOE)
    C.PRESETS["tiny-m2"] = dataclasses.replace(TINY_M2)
    yield
    C.PRESETS.pop("tiny-olmoe", None)
    C.PRESETS.pop("tiny-m2", None)


def _engine(model, **kw):
    return LLMEngine(EngineConfig(model=model, device="cpu",
                                  dtype="float32", kv_cache_blocks=64, **kw))


def _sd_from(eng):
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
        sd[p + "self_attn.o_proj.weight"] = layer.attn.o_w.data
        # full-projection qk-norm weights
        sd[p + "self_attn.q_norm.weight"] = layer.attn.q_norm.data
        sd[p + "self_attn.k_norm.weight"] = layer.attn.k_norm.data
        sd[p + "input_layernorm.weight"] = layer.input_norm.data
        sd[p + "post_attention_layernorm.weight"] = layer.post_attn_norm.data
        # stacked 3D expert tensors: HF [E, 2i, h]/[E, h, i] == ours
        sd[p + "mlp.experts.gate_up_proj"] = layer.mlp.gate_up_w.data
        sd[p + "mlp.experts.down_proj"] = layer.mlp.down_w.data
        sd[p + "mlp.gate.weight"] = layer.mlp.router_w.data
        if layer.mlp.router_bias is not None:
            sd[p + "mlp.e_score_correction_bias"] = \
                layer.mlp.router_bias.data
    return sd


def _hf_olmoe(eng):
    from transformers import OlmoeConfig, OlmoeForCausalLM

    spec = eng.cfg.spec
    hf_cfg = OlmoeConfig(
        vocab_size=spec.vocab_size, hidden_size=spec.hidden_size,
        intermediate_size=spec.moe_intermediate_size,
        num_hidden_layers=spec.num_layers,
        num_attention_heads=spec.num_heads,
        num_key_value_heads=spec.num_kv_heads,
        rms_norm_eps=spec.rms_norm_eps,
        max_position_embeddings=spec.max_position_embeddings,
        rope_theta=spec.rope_theta, tie_word_embeddings=False,
        attention_bias=False, num_experts=spec.num_experts,
        num_experts_per_tok=spec.num_experts_per_tok,
        norm_topk_prob=False, eos_token_id=1, pad_token_id=0,
        attn_implementation="eager",
    )
    hf = OlmoeForCausalLM(hf_cfg).eval().float()
    missing, unexpected = hf.load_state_dict(_sd_from(eng), strict=False)
    assert not unexpected, unexpected
    assert all("rotary" in k for k in missing), missing
    return hf


def _hf_m2(eng):
    from transformers import MiniMaxM2Config, MiniMaxM2ForCausalLM

    spec = eng.cfg.spec
    hf_cfg = MiniMaxM2Config(
        vocab_size=spec.vocab_size, hidden_size=spec.hidden_size,
        intermediate_size=spec.moe_intermediate_size,
        num_hidden_layers=spec.num_layers,
        num_attention_heads=spec.num_heads,
        num_key_value_heads=spec.num_kv_heads,
        head_dim=spec.head_dim, rms_norm_eps=spec.rms_norm_eps,
        max_position_embeddings=spec.max_position_embeddings,
        rope_theta=spec.rope_theta, tie_word_embeddings=False,
        num_local_experts=spec.num_experts,
        num_experts_per_tok=spec.num_experts_per_tok,
        eos_token_id=1, pad_token_id=0, attn_implementation="eager",
    )
    hf = MiniMaxM2ForCausalLM(hf_cfg).eval().float()
    missing, unexpected = hf.load_state_dict(_sd_from(eng), strict=False)
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


@pytest.mark.parametrize("model,hf_fn", [("tiny-olmoe", _hf_olmoe),
                                         ("tiny-m2", _hf_m2)])
def test_moe_qknorm_full_matches_hf_logits(model, hf_fn):
    eng = _engine(model)
    hf = hf_fn(eng)
    prompt = [3, 1, 4, 1, 5, 9, 2, 6, 5, 3]
    with torch.inference_mode():
        want = hf(torch.tensor([prompt])).logits[0, -1]
    got = _prefill_logits(eng, prompt)
    assert torch.allclose(got, want, atol=3e-4, rtol=1e-3), \
        (got - want).abs().max()


@pytest.mark.parametrize("model,hf_fn", [("tiny-olmoe", _hf_olmoe),
                                         ("tiny-m2", _hf_m2)])
def test_moe_qknorm_full_decode_matches_hf(model, hf_fn):
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


def test_minimax_m2_spec_from_hf_config():
    m2 = ModelSpec.from_hf_config({
        "architectures": ["MiniMaxM2ForCausalLM"], "vocab_size": 200064,
        "hidden_size": 3072, "num_hidden_layers": 62,
        "num_attention_heads": 48, "num_key_value_heads": 8,
        "head_dim": 128, "rotary_dim": 64, "num_local_experts": 256,
        "num_experts_per_tok": 8, "intermediate_size": 1536,
        "rope_theta": 5000000.0, "max_position_embeddings": 196608,
    })  # minimax-m2 config shape
    assert m2.qk_norm_full and m2.router_mode == "sigmoid_bias"
    assert m2.norm_topk_prob and m2.partial_rotary_factor == 0.5
    assert m2.num_experts == 256 and m2.moe_intermediate_size == 1536

    oe = ModelSpec.from_hf_config({
        "architectures": ["OlmoeForCausalLM"], "vocab_size": 50304,
        "hidden_size": 2048, "num_hidden_layers": 16,
        "num_attention_heads": 16, "num_key_value_heads": 16,
        "num_experts": 64, "num_experts_per_tok": 8,
        "norm_topk_prob": False, "intermediate_size": 1024,
    })  # olmoe-1b-7b config shape
    assert oe.qk_norm_full and oe.router_mode == "softmax"
    assert not oe.norm_topk_prob and not oe.norm_after
