"""Automatic prefix caching: shared KV blocks, suffix-only recompute,
eviction, and output equivalence."""
import pytest

from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams


def _cfg(**kw):
    kw.setdefault("model", "tiny")
    kw.setdefault("device", "cpu")
    kw.setdefault("kv_cache_blocks", 128)
    kw.setdefault("max_model_len", 512)
    return EngineConfig(**kw)


PREFIX = list(range(2, 50))  # 48 tokens = 3 full blocks


def test_cached_outputs_match_uncached():
    p = SamplingParams(max_tokens=8, ignore_eos=True)
    prompts = [PREFIX + [100], PREFIX + [101, 102]]
    plain_eng = LLMEngine(_cfg())
    plain = [plain_eng.generate([q], p)[0] for q in prompts]
    eng = LLMEngine(_cfg(enable_prefix_caching=True))
    a = eng.generate([prompts[0]], p)[0]
    b = eng.generate([prompts[1]], p)[0]    # hits the cached prefix
    assert [a, b] == plain
    assert eng.scheduler.kv.allocator.hits > 0


def test_prefix_sharing_reuses_blocks():
    p = SamplingParams(max_tokens=4, ignore_eos=True)
    eng = LLMEngine(_cfg(enable_prefix_caching=True))
    eng.generate([PREFIX + [100]], p)
    alloc = eng.scheduler.kv.allocator
    hits0 = alloc.hits
    rid = eng.add_request(PREFIX + [101], p)
    eng.step()  # suffix admission
    seq = eng.seqs[rid]
    # after the suffix step, KV covers the whole prompt; the shared part
    # shows up as 3 cache hits (one per full prefix block)
    assert seq.num_cached_tokens == len(seq.prompt_token_ids)
    assert alloc.hits == hits0 + 3
    while eng.has_unfinished():
        eng.step()


def test_identical_prompt_caches_all_but_last_block():
    p = SamplingParams(max_tokens=4, ignore_eos=True)
    eng = LLMEngine(_cfg(enable_prefix_caching=True))
    q = PREFIX + [100]  # 49 tokens
    first = eng.generate([q], p)[0]
    second = eng.generate([q], p)[0]
    assert first == second


def test_eviction_under_pressure():
    p = SamplingParams(max_tokens=4, ignore_eos=True)
    eng = LLMEngine(_cfg(enable_prefix_caching=True, kv_cache_blocks=24,
                         max_model_len=128))
    outs = []
    for base in range(5):  # distinct 4-block prompts overflow 24 blocks
        q = [1000 * 0 + base * 90 + t for t in range(60)]
        q = [v % 500 for v in q]
        outs.append(eng.generate([q], p)[0])
    # still correct after evictions; allocator accounting consistent
    alloc = eng.scheduler.kv.allocator
    assert alloc.num_free + 0 <= alloc.num_blocks
    assert len(outs) == 5


def test_decode_matches_prefill_with_caching():
    p = SamplingParams(max_tokens=10, ignore_eos=True)
    full = LLMEngine(_cfg(enable_prefix_caching=True)).generate(
        [PREFIX + [7]], p)[0]
    cont = LLMEngine(_cfg(enable_prefix_caching=True)).generate(
        [PREFIX + [7] + full[:5]], SamplingParams(max_tokens=5,
                                                  ignore_eos=True))[0]
    assert cont == full[5:]
