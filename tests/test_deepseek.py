"""DeepSeek-V3/R1 family (DeepseekV3ForCausalLM): MLA latent attention
(q LoRA + compressed KV + absorbed-attention serving path over the paged
latent cache) + sigmoid/bias grouped MoE — logits-exact vs HF transformers
at fp32 on CPU. The same oracle discipline as the other families; the
CDNA4 absorbed-attention kernels (r3) verify against these tests."""
import dataclasses

import pytest
import torch

import gpustack_amd.engine.config as C
from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams
from gpustack_amd.engine.config import ModelSpec

TINY_DS = ModelSpec(
    architecture="DeepseekV3ForCausalLM", vocab_size=512, hidden_size=128,
    intermediate_size=256, num_layers=3, num_heads=4, num_kv_heads=4,
    head_dim=48, max_position_embeddings=256, rope_theta=10000.0,
    eos_token_id=1, num_experts=8, num_experts_per_tok=2,
    moe_intermediate_size=64, router_mode="sigmoid_bias",
    n_shared_experts=1, first_k_dense_replace=1, routed_scaling_factor=1.5,
    n_group=2, topk_group=1, norm_topk_prob=True,
    q_lora_rank=64, kv_lora_rank=96, qk_nope_head_dim=32,
    qk_rope_head_dim=16, v_head_dim=32, rope_interleave=True,
)


@pytest.fixture(autouse=True)
def _tiny_ds_preset():
    C.PRESETS["tiny-ds"] = dataclasses.replace(TINY_DS)
    yield
    C.PRESETS.pop("tiny-ds", None)


def _engine(**kw):
    spec_over = kw.pop("spec_over", {})
    if spec_over:
        C.PRESETS["tiny-ds"] = dataclasses.replace(TINY_DS, **spec_over)
    return LLMEngine(EngineConfig(model="tiny-ds", device="cpu",
                                  dtype=kw.pop("dtype", "float32"),
                                  kv_cache_blocks=64, **kw))


def _hf_from(eng):
    from transformers import DeepseekV3Config, DeepseekV3ForCausalLM

    spec = eng.cfg.spec
    hf_cfg = DeepseekV3Config(
        vocab_size=spec.vocab_size, hidden_size=spec.hidden_size,
        intermediate_size=spec.intermediate_size,
        num_hidden_layers=spec.num_layers,
        num_attention_heads=spec.num_heads,
        num_key_value_heads=spec.num_kv_heads,
        rms_norm_eps=spec.rms_norm_eps,
        max_position_embeddings=spec.max_position_embeddings,
        tie_word_embeddings=False, attention_bias=False,
        n_routed_experts=spec.num_experts,
        num_experts_per_tok=spec.num_experts_per_tok,
        moe_intermediate_size=spec.moe_intermediate_size,
        n_shared_experts=spec.n_shared_experts,
        first_k_dense_replace=spec.first_k_dense_replace,
        n_group=spec.n_group, topk_group=spec.topk_group,
        routed_scaling_factor=spec.routed_scaling_factor,
        norm_topk_prob=spec.norm_topk_prob,
        q_lora_rank=spec.q_lora_rank or None,
        kv_lora_rank=spec.kv_lora_rank,
        qk_nope_head_dim=spec.qk_nope_head_dim,
        qk_rope_head_dim=spec.qk_rope_head_dim,
        v_head_dim=spec.v_head_dim,
        rope_interleave=spec.rope_interleave,
        rope_parameters={"rope_type": "default",
                         "rope_theta": spec.rope_theta},
        attn_implementation="eager",
    )
    hf = DeepseekV3ForCausalLM(hf_cfg).eval().float()
    m = eng.runner.model
    spec_i = spec.intermediate_size
    sd = {
        "model.embed_tokens.weight": m.embed.data,
        "model.norm.weight": m.final_norm.data,
        "lm_head.weight": m.lm_head.data,
    }
    for li, layer in enumerate(m.layers):
        p = f"model.layers.{li}."
        a = layer.attn
        if spec.q_lora_rank:
            sd[p + "self_attn.q_a_proj.weight"] = a.q_a_w.data
            sd[p + "self_attn.q_a_layernorm.weight"] = a.q_a_norm.data
            sd[p + "self_attn.q_b_proj.weight"] = a.q_b_w.data
        else:
            sd[p + "self_attn.q_proj.weight"] = a.q_w.data
        sd[p + "self_attn.kv_a_proj_with_mqa.weight"] = a.kv_a_w.data
        sd[p + "self_attn.kv_a_layernorm.weight"] = a.kv_a_norm.data
        sd[p + "self_attn.kv_b_proj.weight"] = a.kv_b_w.data
        sd[p + "self_attn.o_proj.weight"] = a.o_w.data
        sd[p + "input_layernorm.weight"] = layer.input_norm.data
        sd[p + "post_attention_layernorm.weight"] = layer.post_attn_norm.data
        if hasattr(layer.mlp, "router_w"):
            sd[p + "mlp.gate.weight"] = layer.mlp.router_w.data
            sd[p + "mlp.gate.e_score_correction_bias"] = \
                layer.mlp.router_bias.data
            sd[p + "mlp.experts.gate_up_proj"] = layer.mlp.gate_up_w.data
            sd[p + "mlp.experts.down_proj"] = layer.mlp.down_w.data
            sgu = layer.mlp.shared_gate_up_w.data
            si = layer.mlp.shared_i
            sd[p + "mlp.shared_experts.gate_proj.weight"] = sgu[:si]
            sd[p + "mlp.shared_experts.up_proj.weight"] = sgu[si:]
            sd[p + "mlp.shared_experts.down_proj.weight"] = \
                layer.mlp.shared_down_w.data
        else:
            gu = layer.mlp.gate_up_w.data
            sd[p + "mlp.gate_proj.weight"] = gu[:spec_i]
            sd[p + "mlp.up_proj.weight"] = gu[spec_i:]
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


@pytest.mark.parametrize("interleave", [True, False])
def test_deepseek_matches_hf_transformers_logits(interleave):
    eng = _engine(spec_over={"rope_interleave": interleave})
    hf = _hf_from(eng)
    prompt = [3, 1, 4, 1, 5, 9, 2, 6, 5, 3, 5, 8]
    with torch.inference_mode():
        want = hf(torch.tensor([prompt])).logits[0, -1]
    got = _prefill_logits(eng, prompt)
    assert torch.allclose(got, want, atol=3e-4, rtol=1e-3), \
        (got - want).abs().max()


def test_deepseek_no_q_lora_matches_hf():
    """V2-Lite-style direct q_proj (q_lora_rank null)."""
    eng = _engine(spec_over={"q_lora_rank": 0})
    hf = _hf_from(eng)
    prompt = [7, 2, 9, 9, 4, 6, 1, 3]
    with torch.inference_mode():
        want = hf(torch.tensor([prompt])).logits[0, -1]
    got = _prefill_logits(eng, prompt)
    assert torch.allclose(got, want, atol=3e-4, rtol=1e-3), \
        (got - want).abs().max()


def test_deepseek_decode_matches_hf_generation():
    """Greedy continuation through the engine (paged latent cache,
    absorbed decode) == HF greedy generation."""
    eng = _engine()
    hf = _hf_from(eng)
    prompt = [2, 7, 1, 8, 2, 8]
    n = 8
    with torch.inference_mode():
        out = hf.generate(torch.tensor([prompt]), max_new_tokens=n,
                          do_sample=False, eos_token_id=None)
    want = out[0, len(prompt):].tolist()
    got = eng.generate([prompt], SamplingParams(max_tokens=n,
                                                ignore_eos=True))[0]
    assert got == want


def test_deepseek_chunked_prefill_matches_plain():
    """Chunk continuations run the suffix/latent-gather path — output
    must equal unchunked serving exactly."""
    prompt = list(range(2, 50))
    p = SamplingParams(max_tokens=6, ignore_eos=True)
    plain = _engine(enable_chunked_prefill=False).generate([prompt], p)[0]
    chunked = _engine(enable_chunked_prefill=True,
                      max_prefill_tokens=16).generate([prompt], p)[0]
    assert chunked == plain


def test_deepseek_kv_cache_is_latent_sized():
    eng = _engine()
    kv = eng.runner.kv
    spec = eng.cfg.spec
    lat = spec.kv_lora_rank + spec.qk_rope_head_dim
    assert kv.k_caches[0].shape[1:] == (1, eng.cfg.block_size, lat)
    assert kv.v_caches[0].numel() == 0  # no V pool under MLA
    # bytes accounting follows the latent layout
    assert spec.kv_bytes_per_token() == spec.num_layers * lat * 2


def test_deepseek_spec_from_hf_config():
    spec = ModelSpec.from_hf_config({
        "architectures": ["DeepseekV3ForCausalLM"], "vocab_size": 129280,
        "hidden_size": 7168, "intermediate_size": 18432,
        "num_hidden_layers": 61, "num_attention_heads": 128,
        "num_key_value_heads": 128, "n_routed_experts": 256,
        "num_experts_per_tok": 8, "moe_intermediate_size": 2048,
        "n_shared_experts": 1, "first_k_dense_replace": 3,
        "routed_scaling_factor": 2.5, "n_group": 8, "topk_group": 4,
        "norm_topk_prob": True, "q_lora_rank": 1536, "kv_lora_rank": 512,
        "qk_nope_head_dim": 128, "qk_rope_head_dim": 64, "v_head_dim": 128,
        "rope_theta": 10000.0,
    })
    assert spec.router_mode == "sigmoid_bias"
    assert spec.kv_lora_rank == 512 and spec.q_lora_rank == 1536
    assert spec.qk_rope_head_dim == 64 and spec.rope_interleave
    # MLA latent cache: 576 elements/token/layer, not 2*128*192
    assert spec.kv_bytes_per_token() == 61 * 576 * 2


def test_deepseek_gpu_fails_loudly(monkeypatch):
    from gpustack_amd.engine.model_runner import ModelRunner

    cfg = EngineConfig(model="tiny-ds", device="cuda", kv_cache_blocks=8)
    with pytest.raises(NotImplementedError, match="MLA"):
        ModelRunner(cfg)


def test_deepseek_checkpoint_loader_roundtrip(tmp_path):
    """load_safetensors maps real DeepSeek checkpoint names (q_a/kv_a/
    kv_b/o + gate.e_score_correction_bias + stacked experts + shared
    experts) onto the engine layout — logits match HF after loading from
    disk."""
    from safetensors.torch import save_file

    from gpustack_amd.models.weights import load_safetensors

    eng = _engine()
    hf = _hf_from(eng)  # HF now holds the engine's weights
    sd = {k: v.contiguous().clone() for k, v in hf.state_dict().items()}
    save_file(sd, str(tmp_path / "model.safetensors"))

    # fresh engine with DIFFERENT weights, then load the checkpoint back
    eng2 = _engine(seed=123)
    load_safetensors(eng2.runner.model, eng2.cfg, tmp_path)
    prompt = [9, 8, 7, 3, 2, 6, 1]
    with torch.inference_mode():
        want = hf(torch.tensor([prompt])).logits[0, -1]
    got = _prefill_logits(eng2, prompt)
    assert torch.allclose(got, want, atol=3e-4, rtol=1e-3), \
        (got - want).abs().max()


def _mla_tp_rank_main(rank, world, port, out_path):
    import json
    import os

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams
    from gpustack_amd.parallel import init_tp

    prompts = [[3, 1, 4, 1, 5, 9, 2, 6], [11, 22, 33]]
    comm = init_tp(world, rank, master_port=port, backend="gloo")
    cfg = EngineConfig(model="tiny-mla", device="cpu", kv_cache_blocks=64,
                       max_model_len=128, seed=0, dtype="float32",
                       tp_size=world, tp_rank=rank)
    eng = LLMEngine(cfg, comm)
    results = {}
    if rank == 0:
        rids = [eng.add_request(p, SamplingParams(max_tokens=6,
                                                  ignore_eos=True))
                for p in prompts]
        results = {r: [] for r in rids}
    while eng.tp_active():
        for o in eng.step():
            if rank == 0 and o.request_id in results:
                results[o.request_id].append(o.token_id)
    if rank == 0:
        with open(out_path, "w") as f:
            json.dump([results[r] for r in rids], f)
    import torch.distributed as dist

    dist.barrier()
    dist.destroy_process_group()


def test_deepseek_tp2_matches_tp1():
    """MLA under TP (gloo world 2): latent projections replicate, q_b/
    kv_b/o shard by head — greedy continuation must be deterministic,
    run-to-run stable, and equal to the single-rank engine."""
    import json
    import multiprocessing as mp
    import socket
    import tempfile

    prompts = [[3, 1, 4, 1, 5, 9, 2, 6], [11, 22, 33]]

    def run2():
        s = socket.socket()
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
        s.close()
        out = tempfile.mktemp(suffix=".json")
        ctx = mp.get_context("spawn")
        procs = [ctx.Process(target=_mla_tp_rank_main,
                             args=(r, 2, port, out)) for r in range(2)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=240)
            assert p.exitcode == 0
        with open(out) as f:
            return json.load(f)

    a = run2()
    b = run2()
    assert a == b and all(len(x) == 6 for x in a)
    single = LLMEngine(EngineConfig(model="tiny-mla", device="cpu",
                                    kv_cache_blocks=64, max_model_len=128,
                                    seed=0, dtype="float32")).generate(
        prompts, SamplingParams(max_tokens=6, ignore_eos=True))
    assert a == single


def test_deepseek_prefix_cache_hit_matches_cold():
    """Automatic prefix caching over the LATENT cache: a shared-prefix
    second prompt recomputes only the suffix (the MLA suffix/absorbed
    path) and must produce the cold-start continuation exactly."""
    p = SamplingParams(max_tokens=6, ignore_eos=True)
    shared = list(range(2, 40))
    prompts = [shared + [41, 42], shared + [43, 44, 45]]
    cold = _engine().generate(prompts, p)

    eng = _engine(enable_prefix_caching=True)
    warm1 = eng.generate([prompts[0]], p)[0]
    warm2 = eng.generate([prompts[1]], p)[0]  # hits prompts[0]'s prefix
    assert [warm1, warm2] == cold
    alloc = eng.scheduler.kv.allocator
    assert alloc.hits > 0  # the second prompt actually reused blocks


def test_deepseek_dynamic_lora_rejected(tmp_path):
    """MLA attention has no adapter hook yet — a dynamic LoRA add must
    refuse loudly instead of silently serving a half-applied adapter."""
    eng = _engine()
    with pytest.raises(ValueError, match="MLA"):
        eng.runner.add_lora("a", str(tmp_path))
