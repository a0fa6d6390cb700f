"""Speculative decoding (n-gram draft-verify) correctness on CPU.

The invariant: spec decoding must be OUTPUT-IDENTICAL to plain greedy
decoding — acceptance only keeps tokens the target model would have
produced anyway."""
import pytest

from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams
from gpustack_amd.engine.spec import NgramProposer, accept_tokens
from gpustack_amd.engine.sequence import Sequence


def test_ngram_proposer_copies_continuation():
    p = NgramProposer(num_draft_tokens=3, ngram_max=2, ngram_min=1)
    seq = Sequence("r", [5, 6, 7, 8, 9, 5, 6])
    assert p.propose(seq) == [7, 8, 9]


def test_ngram_proposer_pads_when_no_match():
    p = NgramProposer(num_draft_tokens=3)
    seq = Sequence("r", [1, 2, 3])
    d = p.propose(seq)
    assert len(d) == 3


def test_accept_tokens():
    # model agrees with first 2 drafts, disagrees on 3rd
    assert accept_tokens([10, 11, 12], [10, 11, 99, 55]) == [10, 11, 99]
    # disagrees immediately
    assert accept_tokens([10, 11, 12], [44, 11, 12, 13]) == [44]
    # fully accepted
    assert accept_tokens([10, 11], [10, 11, 12]) == [10, 11, 12]


def _gen(spec, prompts, n=20):
    cfg = EngineConfig(model="tiny", device="cpu", kv_cache_blocks=128,
                       max_model_len=256, speculative=spec)
    eng = LLMEngine(cfg)
    return eng.generate(prompts, SamplingParams(max_tokens=n, ignore_eos=True))


@pytest.mark.parametrize("prompts", [
    [[7, 8, 9, 7, 8, 9, 7, 8]],                   # repetitive: drafts accept
    [[3, 1, 4, 1, 5, 9, 2, 6]],                   # random-ish
    [[1, 2, 3, 1, 2, 3], [9, 9, 9, 9], [4, 5]],   # batch
])
def test_spec_matches_plain_greedy(prompts):
    plain = _gen(None, prompts)
    spec = _gen({"method": "ngram", "num_draft_tokens": 3}, prompts)
    assert spec == plain


def test_spec_near_max_model_len():
    # drafting across the max_model_len boundary must not corrupt KV
    prompts = [[1, 2, 3, 1, 2, 3] * 8]
    cfg = EngineConfig(model="tiny", device="cpu", kv_cache_blocks=128,
                       max_model_len=64,
                       speculative={"method": "ngram", "num_draft_tokens": 4})
    eng = LLMEngine(cfg)
    out = eng.generate(prompts, SamplingParams(max_tokens=64, ignore_eos=True))
    cfg2 = EngineConfig(model="tiny", device="cpu", kv_cache_blocks=128,
                        max_model_len=64)
    eng2 = LLMEngine(cfg2)
    out2 = eng2.generate(prompts, SamplingParams(max_tokens=64, ignore_eos=True))
    assert out == out2


def test_spec_with_random_sampling_falls_back():
    prompts = [[1, 2, 3, 1, 2, 3]]
    p = SamplingParams(temperature=0.9, max_tokens=8, ignore_eos=True, seed=3)
    spec = _gen.__wrapped__ if hasattr(_gen, "__wrapped__") else None
    cfg = EngineConfig(model="tiny", device="cpu", kv_cache_blocks=128,
                       max_model_len=256,
                       speculative={"method": "ngram", "num_draft_tokens": 3})
    eng = LLMEngine(cfg)
    out = eng.generate(prompts, p)
    assert len(out[0]) == 8  # one token per step, no spec acceptance
