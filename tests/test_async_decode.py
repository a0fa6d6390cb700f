"""Async decode pipeline: output-identical to synchronous stepping."""
import os

import pytest

from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams


def _gen(async_on, prompts, p, **cfg_kw):
    os.environ["GPUSTACK_AMD_ASYNC"] = "1" if async_on else "0"
    try:
        cfg_kw.setdefault("kv_cache_blocks", 128)
        cfg_kw.setdefault("max_model_len", 256)
        eng = LLMEngine(EngineConfig(model="tiny", device="cpu", **cfg_kw))
        assert eng._async_enabled == async_on
        return eng.generate(prompts, p)
    finally:
        os.environ.pop("GPUSTACK_AMD_ASYNC", None)


def test_async_matches_sync_greedy():
    prompts = [[3, 1, 4, 1, 5], [9, 2, 6], [7] * 20]
    p = SamplingParams(max_tokens=20, ignore_eos=True)
    assert _gen(True, prompts, p) == _gen(False, prompts, p)


def test_async_matches_sync_with_eos():
    # eos-sensitive seqs force per-step drains but must stay correct
    prompts = [[5, 6, 7, 8]]
    p = SamplingParams(max_tokens=30, ignore_eos=False)
    assert _gen(True, prompts, p) == _gen(False, prompts, p)


def test_async_under_memory_pressure():
    prompts = [[1, 2, 3], [4, 5, 6]]
    p = SamplingParams(max_tokens=40, ignore_eos=True)
    a = _gen(True, prompts, p, kv_cache_blocks=8, max_model_len=64)
    b = _gen(False, prompts, p, kv_cache_blocks=8, max_model_len=64)
    assert a == b


def test_async_staggered_arrivals():
    os.environ["GPUSTACK_AMD_ASYNC"] = "1"
    try:
        eng = LLMEngine(EngineConfig(model="tiny", device="cpu",
                                     kv_cache_blocks=128, max_model_len=256,
                                     admission_max_wait_s=0.0))
        p = SamplingParams(max_tokens=12, ignore_eos=True)
        r1 = eng.add_request([1, 2, 3], p)
        results = {r1: []}
        added_second = False
        r2 = None
        while eng.has_unfinished() or eng._pending is not None:
            outs = eng.step()
            for o in outs:
                results.setdefault(o.request_id, []).append(o.token_id)
            if not added_second and len(results[r1]) >= 4:
                r2 = eng.add_request([7, 8, 9], p)
                added_second = True
        assert len(results[r1]) == 12
        assert r2 is not None and len(results[r2]) == 12
    finally:
        os.environ.pop("GPUSTACK_AMD_ASYNC", None)
    # cross-check against an isolated run of the second prompt
    iso = LLMEngine(EngineConfig(model="tiny", device="cpu", kv_cache_blocks=128,
                                 max_model_len=256)).generate(
        [[7, 8, 9]], SamplingParams(max_tokens=12, ignore_eos=True))[0]
    assert results[r2] == iso


def test_async_abort_mid_flight():
    os.environ["GPUSTACK_AMD_ASYNC"] = "1"
    try:
        eng = LLMEngine(EngineConfig(model="tiny", device="cpu",
                                     kv_cache_blocks=128, max_model_len=256))
        p = SamplingParams(max_tokens=50, ignore_eos=True)
        r1 = eng.add_request([1, 2, 3], p)
        r2 = eng.add_request([4, 5, 6], p)
        for _ in range(5):
            eng.step()
        assert eng.abort_request(r1)
        results = []
        while eng.has_unfinished() or eng._pending is not None:
            results += [o for o in eng.step() if o.request_id == r2]
        assert sum(1 for o in results) > 0
        free = eng.scheduler.kv.allocator.num_free
        # all blocks released at the end
        while eng.has_unfinished():
            eng.step()
        assert eng.scheduler.kv.allocator.num_free == eng.scheduler.kv.allocator.num_blocks
    finally:
        os.environ.pop("GPUSTACK_AMD_ASYNC", None)
