"""OLMo-2 family (Olmo2ForCausalLM): norm-AFTER layer flow (RMSNorm on
each sublayer's output before the residual add, no input norms) and
full-projection qk-norm — logits-exact vs HF transformers at fp32 on CPU,
plus TP2 exactness for the group-reduced full-dim norm."""
import dataclasses

import pytest
import torch

import gpustack_amd.engine.config as C
from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams
from gpustack_amd.engine.config import ModelSpec

TINY_OLMO = ModelSpec(
    architecture="Olmo2ForCausalLM", vocab_size=512, hidden_size=128,
    intermediate_size=256, num_layers=3, num_heads=4, num_kv_heads=2,
    head_dim=32, max_position_embeddings=512, rope_theta=10000.0,
    rms_norm_eps=1e-6, eos_token_id=1, qk_norm=True, norm_after=True,
    qk_norm_full=True,
)


@pytest.fixture(autouse=True)
def _tiny_olmo_preset():
    C.PRESETS["tiny-olmo"] = dataclasses.replace(TINY_OLMO)
    yield
    C.PRESETS.pop("tiny-olmo", None)


def _engine(**kw):
    return LLMEngine(EngineConfig(model="tiny-olmo", device="cpu",
                                  dtype="float32", kv_cache_blocks=64, **kw))


def _hf_from(eng):
    from transformers import Olmo2Config, Olmo2ForCausalLM

    spec = eng.cfg.spec
    hf_cfg = Olmo2Config(
        vocab_size=spec.vocab_size, hidden_size=spec.hidden_size,
        intermediate_size=spec.intermediate_size,
        num_hidden_layers=spec.num_layers,
        num_attention_heads=spec.num_heads,
        num_key_value_heads=spec.num_kv_heads,
        rms_norm_eps=spec.rms_norm_eps,
        max_position_embeddings=spec.max_position_embeddings,
        rope_theta=spec.rope_theta, tie_word_embeddings=False,
        attention_bias=False, eos_token_id=1, pad_token_id=0,
        attn_implementation="eager",
    )
    hf = Olmo2ForCausalLM(hf_cfg).eval().float()
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
        # our input_norm slot holds HF's post_attention_layernorm,
        # post_attn_norm holds post_feedforward_layernorm
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


def test_olmo2_matches_hf_transformers_logits():
    eng = _engine()
    hf = _hf_from(eng)
    prompt = [3, 1, 4, 1, 5, 9, 2, 6, 5, 3]
    with torch.inference_mode():
        want = hf(torch.tensor([prompt])).logits[0, -1]
    got = _prefill_logits(eng, prompt)
    assert torch.allclose(got, want, atol=3e-4, rtol=1e-3), \
        (got - want).abs().max()


def test_olmo2_decode_matches_hf_generation():
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


def _olmo_tp_rank_main(rank, world, port, out_path):
    import json
    import os

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import dataclasses as _dc

    import gpustack_amd.engine.config as CC
    from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams
    from gpustack_amd.parallel import init_tp

    CC.PRESETS["tiny-olmo"] = _dc.replace(TINY_OLMO)
    prompts = [[3, 1, 4, 1, 5, 9, 2, 6], [11, 22, 33]]
    comm = init_tp(world, rank, master_port=port, backend="gloo")
    cfg = EngineConfig(model="tiny-olmo", device="cpu", kv_cache_blocks=64,
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


def test_olmo2_tp2_matches_tp1():
    """The full-projection qk-norm reduces its mean square across the TP
    group — TP2 output must equal single-rank exactly."""
    import json
    import multiprocessing as mp
    import socket
    import tempfile

    prompts = [[3, 1, 4, 1, 5, 9, 2, 6], [11, 22, 33]]
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    out = tempfile.mktemp(suffix=".json")
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_olmo_tp_rank_main, args=(r, 2, port, out))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=240)
        assert p.exitcode == 0
    with open(out) as f:
        got = json.load(f)
    single = _engine().generate(prompts,
                                SamplingParams(max_tokens=6,
                                               ignore_eos=True))
    assert got == single
