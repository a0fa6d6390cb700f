"""Draft-model (EAGLE-style) speculative decoding.

Key invariant (same as ngram spec): greedy acceptance makes the output
IDENTICAL to plain greedy decoding regardless of draft quality — the
draft only affects how many rows verify per step.
"""
import pytest
import torch

from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams


def _gen(spec=None, prompts=None, n=24):
    cfg = EngineConfig(model="tiny", device="cpu", kv_cache_blocks=256,
                       max_model_len=256, speculative=spec)
    eng = LLMEngine(cfg)
    p = SamplingParams(max_tokens=n, ignore_eos=True)
    return eng, eng.generate(prompts or [[1, 2, 3, 4, 5], [9, 8, 7, 6, 5, 4, 3]], p)


def test_eagle_matches_plain():
    _, plain = _gen(None)
    eng, spec = _gen({"method": "eagle", "num_draft_tokens": 3})
    assert eng.runner.eagle is not None
    assert spec == plain


def test_eagle3_alias_and_k():
    _, plain = _gen(None)
    _, spec = _gen({"method": "eagle3", "num_draft_tokens": 5})
    assert spec == plain


def test_eagle_draft_state_lifecycle():
    eng, _ = _gen({"method": "eagle", "num_draft_tokens": 2})
    eagle = eng.runner.eagle
    assert not eagle.states  # all finished -> all dropped
    assert eagle.allocator.num_free == eagle.allocator.num_blocks


def test_eagle_abort_drops_state():
    cfg = EngineConfig(model="tiny", device="cpu", kv_cache_blocks=256,
                       max_model_len=256,
                       speculative={"method": "eagle", "num_draft_tokens": 2})
    eng = LLMEngine(cfg)
    rid = eng.add_request([1, 2, 3], SamplingParams(max_tokens=50, ignore_eos=True))
    for _ in range(4):
        eng.step()
    assert rid in eng.runner.eagle.states
    eng.abort_request(rid)
    assert rid not in eng.runner.eagle.states
    assert eng.runner.eagle.allocator.num_free == eng.runner.eagle.allocator.num_blocks


def test_eagle_draft_proposals_flow():
    # after a few steps, sequences carry fresh draft windows
    cfg = EngineConfig(model="tiny", device="cpu", kv_cache_blocks=256,
                       max_model_len=256,
                       speculative={"method": "eagle", "num_draft_tokens": 3})
    eng = LLMEngine(cfg)
    rid = eng.add_request([1, 2, 3], SamplingParams(max_tokens=50, ignore_eos=True))
    eng.step()  # prefill + seed
    seq = eng.seqs[rid]
    assert seq.next_draft is not None and len(seq.next_draft) == 3
    eng.step()  # decode consumes the draft, proposes the next
    assert seq.next_draft is not None and len(seq.next_draft) == 3


def test_eagle_survives_preemption():
    # tiny KV pool forces preemption; output must still match plain greedy
    plain_cfg = EngineConfig(model="tiny", device="cpu", kv_cache_blocks=24,
                             max_model_len=128)
    p = SamplingParams(max_tokens=16, ignore_eos=True)
    prompts = [[1, 2, 3, 4, 5, 6, 7, 8], [9, 8, 7, 6, 5, 4, 3, 2],
               [11, 12, 13, 14, 15, 16, 17, 18]]
    plain = LLMEngine(plain_cfg).generate(prompts, p)
    cfg = EngineConfig(model="tiny", device="cpu", kv_cache_blocks=24,
                       max_model_len=128,
                       speculative={"method": "eagle", "num_draft_tokens": 2})
    eng = LLMEngine(cfg)
    out = eng.generate(prompts, p)
    assert out == plain


def test_mtp_matches_plain():
    # MTP: k chained draft heads with distinct weights
    _, plain = _gen(None)
    eng, out = _gen({"method": "mtp", "num_draft_tokens": 3})
    assert eng.runner.eagle.n_heads == 3
    assert out == plain


def test_draft_checkpoint_loading(tmp_path):
    """speculative_config draft_dir: published EAGLE-naming checkpoint loads
    into the draft head; outputs stay exact vs plain greedy."""
    import torch
    from safetensors.torch import save_file

    from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams

    spec = EngineConfig(model="tiny").spec
    d = spec.head_dim
    torch.manual_seed(9)
    ckpt = {
        "fc.weight": torch.randn(spec.hidden_size, 2 * spec.hidden_size) * 0.02,
        "layers.0.self_attn.q_proj.weight":
            torch.randn(spec.num_heads * d, spec.hidden_size) * 0.02,
        "layers.0.self_attn.k_proj.weight":
            torch.randn(spec.num_kv_heads * d, spec.hidden_size) * 0.02,
        "layers.0.self_attn.v_proj.weight":
            torch.randn(spec.num_kv_heads * d, spec.hidden_size) * 0.02,
        "layers.0.self_attn.o_proj.weight":
            torch.randn(spec.hidden_size, spec.num_heads * d) * 0.02,
        "layers.0.mlp.gate_proj.weight":
            torch.randn(spec.intermediate_size, spec.hidden_size) * 0.02,
        "layers.0.mlp.up_proj.weight":
            torch.randn(spec.intermediate_size, spec.hidden_size) * 0.02,
        "layers.0.mlp.down_proj.weight":
            torch.randn(spec.hidden_size, spec.intermediate_size) * 0.02,
        "layers.0.input_layernorm.weight": torch.ones(spec.hidden_size),
        "layers.0.post_attention_layernorm.weight": torch.ones(spec.hidden_size),
    }
    save_file(ckpt, str(tmp_path / "draft.safetensors"))

    def cfg(spec_cfg=None):
        return EngineConfig(model="tiny", device="cpu", kv_cache_blocks=96,
                            max_model_len=256, speculative=spec_cfg)

    eng = LLMEngine(cfg({"method": "eagle", "num_draft_tokens": 2,
                         "draft_dir": str(tmp_path)}))
    # loaded weights match the checkpoint (TP=1: no shard offset)
    got = eng.runner.eagle.layers[0].attn.qkv_w.float()
    want = torch.cat([ckpt["layers.0.self_attn.q_proj.weight"],
                      ckpt["layers.0.self_attn.k_proj.weight"],
                      ckpt["layers.0.self_attn.v_proj.weight"]]).to(torch.bfloat16).float()
    assert torch.equal(got, want)
    assert torch.equal(eng.runner.eagle.fc_ws[0].float(),
                       ckpt["fc.weight"].to(torch.bfloat16).float())
    # exactness invariant holds with a loaded draft too
    p = SamplingParams(max_tokens=12, ignore_eos=True)
    plain = LLMEngine(cfg()).generate([[3, 4, 5]], p)
    assert eng.generate([[3, 4, 5]], p) == plain


def test_draft_checkpoint_bad_dir(tmp_path):
    import pytest as _pytest

    from gpustack_amd.engine import EngineConfig, LLMEngine

    with _pytest.raises(FileNotFoundError):
        LLMEngine(EngineConfig(model="tiny", device="cpu", kv_cache_blocks=64,
                               speculative={"method": "eagle",
                                            "draft_dir": str(tmp_path)}))
