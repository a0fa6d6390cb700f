"""Chunked prefill: budget-sized prompt admission, decode interleaving,
output equivalence, block accounting (reference: vLLM --enable-chunked-prefill,
surfaced through backend_parameters passthrough in gpustack)."""
import pytest

from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams


def _cfg(**kw):
    kw.setdefault("model", "tiny")
    kw.setdefault("device", "cpu")
    kw.setdefault("kv_cache_blocks", 128)
    kw.setdefault("max_model_len", 512)
    kw.setdefault("max_prefill_tokens", 64)
    return EngineConfig(**kw)


LONG = [(7 * t + 3) % 500 for t in range(200)]  # 200 tokens > 3x budget


def test_chunked_matches_unchunked():
    p = SamplingParams(max_tokens=8, ignore_eos=True)
    plain = LLMEngine(_cfg()).generate([LONG], p)[0]
    chunked = LLMEngine(_cfg(enable_chunked_prefill=True)).generate([LONG], p)[0]
    assert chunked == plain


def test_chunk_steps_respect_budget():
    eng = LLMEngine(_cfg(enable_chunked_prefill=True))
    sizes = []
    orig = eng.runner.execute

    def rec(batch):
        if batch.is_prefill or batch.is_suffix:
            sizes.append(batch.num_tokens)
        return orig(batch)

    eng.runner.execute = rec
    eng.generate([LONG], SamplingParams(max_tokens=4, ignore_eos=True))
    assert sizes and max(sizes) <= 64
    assert len(sizes) >= 4  # 200 tokens / 64 budget -> >= 4 admission steps


def test_decode_progresses_between_chunks():
    """A running sequence keeps emitting tokens while a long prompt is being
    admitted chunk by chunk (the entire point of chunked prefill)."""
    eng = LLMEngine(_cfg(enable_chunked_prefill=True))
    p = SamplingParams(max_tokens=400, ignore_eos=True)
    short = eng.add_request([5, 6, 7], p)
    for _ in range(3):
        eng.step()  # short is prefilled and decoding
    long_id = eng.add_request(LONG, SamplingParams(max_tokens=4, ignore_eos=True))
    short_tokens_during_admission = 0
    for _ in range(50):
        outs = eng.step()
        short_tokens_during_admission += sum(
            1 for o in outs if o.request_id == short)
        if any(o.request_id == long_id for o in outs):
            break  # long prompt produced its first token
    else:
        pytest.fail("long prompt never produced a token")
    assert short_tokens_during_admission >= 2
    eng.abort_request(short)


def test_block_accounting_and_abort_mid_chunk():
    eng = LLMEngine(_cfg(enable_chunked_prefill=True))
    free0 = eng.scheduler.kv.allocator.num_free
    rid = eng.add_request(LONG, SamplingParams(max_tokens=4, ignore_eos=True))
    eng.step()  # chunk 0
    assert eng.scheduler._chunking is not None
    assert eng.has_unfinished()
    assert eng.abort_request(rid)
    assert not eng.has_unfinished()
    assert eng.scheduler.kv.allocator.num_free == free0

    # full run also returns every block
    eng.generate([LONG], SamplingParams(max_tokens=4, ignore_eos=True))
    assert eng.scheduler.kv.allocator.num_free == free0


def test_chunked_with_spec_model_rejected():
    # chunked prefill (default-on since r2) yields to an explicit
    # draft-model speculative config instead of raising
    cfg = _cfg(enable_chunked_prefill=True,
               speculative={"method": "eagle", "num_draft_tokens": 3})
    assert cfg.enable_chunked_prefill is False


def test_chunked_with_ngram_spec_ok():
    p = SamplingParams(max_tokens=8, ignore_eos=True)
    eng = LLMEngine(_cfg(enable_chunked_prefill=True,
                         speculative={"method": "ngram",
                                      "num_draft_tokens": 2}))
    out = eng.generate([LONG], p)[0]
    plain = LLMEngine(_cfg()).generate([LONG], p)[0]
    assert out == plain


def test_chunked_with_prefix_caching():
    """Chunked prefill + automatic prefix caching coexist: cache-hit
    prompts still take the suffix path; cache-miss long prompts chunk."""
    p = SamplingParams(max_tokens=6, ignore_eos=True)
    plain = LLMEngine(_cfg())
    want_long = plain.generate([LONG], p)[0]
    short = [5, 6, 7] * 20  # 60 tokens, shares no prefix with LONG
    want_short = plain.generate([short], p)[0]

    eng = LLMEngine(_cfg(enable_chunked_prefill=True,
                         enable_prefix_caching=True))
    assert eng.generate([LONG], p)[0] == want_long
    assert eng.generate([short], p)[0] == want_short
    # re-running the long prompt hits the prefix cache (suffix path)
    hits0 = eng.scheduler.kv.allocator.hits
    assert eng.generate([LONG], p)[0] == want_long
    assert eng.scheduler.kv.allocator.hits > hits0
