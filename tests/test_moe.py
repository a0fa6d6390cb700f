"""MoE model family (Qwen3-MoE / Mixtral): routing, dispatch, HF parity."""
import pytest
import torch

from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams


def test_moe_generates_and_routes():
    eng = LLMEngine(EngineConfig(model="tiny-moe", device="cpu",
                                 kv_cache_blocks=64, max_model_len=128))
    layer = eng.runner.model.layers[0]
    assert layer.mlp.__class__.__name__ == "MoEMLP"
    assert layer.mlp.e == 8 and layer.mlp.top_k == 2
    out = eng.generate([[1, 2, 3, 4], [5, 6, 7]],
                       SamplingParams(max_tokens=8, ignore_eos=True))
    assert all(len(o) == 8 for o in out)


def test_moe_decode_matches_prefill():
    p = SamplingParams(max_tokens=10, ignore_eos=True)
    full = LLMEngine(EngineConfig(model="tiny-moe", device="cpu",
                                  kv_cache_blocks=64, max_model_len=128)
                     ).generate([[1, 2, 3, 4, 5]], p)[0]
    cont = LLMEngine(EngineConfig(model="tiny-moe", device="cpu",
                                  kv_cache_blocks=64, max_model_len=128)
                     ).generate([[1, 2, 3, 4, 5] + full[:5]],
                                SamplingParams(max_tokens=5, ignore_eos=True))[0]
    assert cont == full[5:]


def test_moe_weight_bytes_counts_experts():
    from gpustack_amd.engine.config import PRESETS

    moe = PRESETS["qwen3-30b-a3b"]
    assert moe.weight_bytes() > 55e9  # ~30B params bf16


def test_moe_matches_hf_transformers_logits():
    from transformers import Qwen3MoeConfig, Qwen3MoeForCausalLM as HFModel

    cfg = EngineConfig(model="tiny-moe", device="cpu", dtype="float32",
                       kv_cache_blocks=64)
    cfg.spec.qk_norm = True  # HF Qwen3Moe always applies q/k norms
    eng = LLMEngine(cfg)
    spec = cfg.spec
    hf_cfg = Qwen3MoeConfig(
        vocab_size=spec.vocab_size, hidden_size=spec.hidden_size,
        intermediate_size=spec.intermediate_size,
        moe_intermediate_size=spec.moe_intermediate_size,
        num_experts=spec.num_experts,
        num_experts_per_tok=spec.num_experts_per_tok,
        norm_topk_prob=True, decoder_sparse_step=1, mlp_only_layers=[],
        num_hidden_layers=spec.num_layers,
        num_attention_heads=spec.num_heads,
        num_key_value_heads=spec.num_kv_heads, head_dim=spec.head_dim,
        rms_norm_eps=spec.rms_norm_eps, rope_theta=spec.rope_theta,
        max_position_embeddings=spec.max_position_embeddings,
        tie_word_embeddings=False, attention_bias=False,
    )
    hf = HFModel(hf_cfg).eval().float()
    m = eng.runner.model
    sd = {
        "model.embed_tokens.weight": m.embed.data,
        "model.norm.weight": m.final_norm.data,
        "lm_head.weight": m.lm_head.data,
    }
    d = spec.head_dim
    for li, layer in enumerate(m.layers):
        p = f"model.layers.{li}."
        qkv = layer.attn.qkv_w.data
        nq, nk = spec.num_heads * d, spec.num_kv_heads * d
        sd[p + "self_attn.q_proj.weight"] = qkv[:nq]
        sd[p + "self_attn.k_proj.weight"] = qkv[nq:nq + nk]
        sd[p + "self_attn.v_proj.weight"] = qkv[nq + nk:]
        sd[p + "self_attn.o_proj.weight"] = layer.attn.o_w.data
        sd[p + "self_attn.q_norm.weight"] = layer.attn.q_norm.data
        sd[p + "self_attn.k_norm.weight"] = layer.attn.k_norm.data
        sd[p + "mlp.gate.weight"] = layer.mlp.router_w.data
        # transformers >= 5 stores experts fused in OUR layout:
        # gate_up_proj [E, 2*mi, h], down_proj [E, h, mi]
        sd[p + "mlp.experts.gate_up_proj"] = layer.mlp.gate_up_w.data
        sd[p + "mlp.experts.down_proj"] = layer.mlp.down_w.data
        sd[p + "input_layernorm.weight"] = layer.input_norm.data
        sd[p + "post_attention_layernorm.weight"] = layer.post_attn_norm.data
    missing, unexpected = hf.load_state_dict(sd, strict=False)
    assert not unexpected, unexpected[:5]
    assert all("rotary" in k or "bias" in k for k in missing), missing[:5]

    prompt = [3, 1, 4, 1, 5, 9, 2, 6]
    with torch.inference_mode():
        hf_logits = hf(torch.tensor([prompt])).logits[0, -1]

    from gpustack_amd.engine.scheduler import ScheduledBatch
    from gpustack_amd.engine.sequence import Sequence

    seq = Sequence("t", prompt)
    seq.block_table = eng.scheduler.kv.allocator.allocate(2)
    batch = ScheduledBatch(is_prefill=True, seqs=[seq], token_ids=prompt,
                           positions=list(range(len(prompt))),
                           slot_mapping=eng.scheduler.kv.slots_for(
                               seq.block_table, 0, len(prompt)),
                           seq_lens=[len(prompt)])
    tokens, meta = eng.runner._meta(batch)
    ours = eng.runner.model(tokens, meta, eng.runner.kv)[0]
    torch.testing.assert_close(ours, hf_logits, atol=2e-4, rtol=2e-4)


def test_moe_from_hf_config():
    from gpustack_amd.engine.config import ModelSpec

    spec = ModelSpec.from_hf_config({
        "architectures": ["Qwen3MoeForCausalLM"], "vocab_size": 151936,
        "hidden_size": 2048, "num_hidden_layers": 48,
        "num_attention_heads": 32, "num_key_value_heads": 4,
        "head_dim": 128, "num_experts": 128, "num_experts_per_tok": 8,
        "moe_intermediate_size": 768, "norm_topk_prob": True,
    })
    assert spec.num_experts == 128 and spec.moe_intermediate_size == 768
    mix = ModelSpec.from_hf_config({
        "architectures": ["MixtralForCausalLM"], "num_local_experts": 8,
        "num_experts_per_tok": 2, "intermediate_size": 14336,
    })
    assert mix.num_experts == 8 and mix.moe_intermediate_size == 14336


def test_moe_dispatch_paths_agree():
    # the padded-bmm decode path must match the per-expert loop exactly
    # (same per-row math; padding rows are discarded)
    eng = LLMEngine(EngineConfig(model="tiny-moe", device="cpu",
                                 kv_cache_blocks=64, max_model_len=128))
    mlp = eng.runner.model.layers[0].mlp
    torch.manual_seed(0)
    x = torch.randn(9, 128, dtype=mlp.gate_up_w.dtype)
    logits = torch.nn.functional.linear(x.float(), mlp.router_w.float())
    w, e = torch.topk(logits, mlp.top_k, dim=-1)
    w = torch.softmax(w, dim=-1)
    fe = e.reshape(-1)
    ft = torch.arange(9).repeat_interleave(mlp.top_k)
    fw = w.reshape(-1).to(x.dtype)
    a = x.new_zeros(x.shape[0] * mlp.top_k, x.shape[1])
    b = x.new_zeros(x.shape[0] * mlp.top_k, x.shape[1])
    mlp._loop_dispatch(x, a, fe, ft, fw)
    mlp._bmm_dispatch(x, b, fe, ft, fw)
    torch.testing.assert_close(a, b, atol=2e-2, rtol=2e-2)
