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
