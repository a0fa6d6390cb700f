"""Request priority scheduling (vLLM priority analog): admission order and
preemption victim selection."""
from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams


def _cfg(**kw):
    kw.setdefault("model", "tiny")
    kw.setdefault("device", "cpu")
    kw.setdefault("kv_cache_blocks", 64)
    kw.setdefault("max_model_len", 128)
    return EngineConfig(**kw)


def test_priority_admission_order():
    eng = LLMEngine(_cfg(max_num_seqs=64))
    p_lo = SamplingParams(max_tokens=2, ignore_eos=True, priority=0)
    p_hi = SamplingParams(max_tokens=2, ignore_eos=True, priority=5)
    lo = eng.add_request([1, 2, 3], p_lo)
    hi = eng.add_request([4, 5, 6], p_hi)
    waiting = [s.request_id for s in eng.scheduler.waiting]
    assert waiting == [hi, lo]  # high priority jumped the queue
    while eng.has_unfinished():
        eng.step()


def test_priority_fifo_within_class():
    eng = LLMEngine(_cfg())
    p = SamplingParams(max_tokens=2, ignore_eos=True)
    a = eng.add_request([1], p)
    b = eng.add_request([2], p)
    assert [s.request_id for s in eng.scheduler.waiting] == [a, b]
    eng.abort_request(a)
    eng.abort_request(b)


def test_preemption_evicts_lowest_priority():
    # tiny pool: 8 blocks of 16 = 128 token slots force preemption
    eng = LLMEngine(_cfg(kv_cache_blocks=9, max_num_seqs=8,
                         max_prefill_tokens=64))
    hi = eng.add_request([1] * 40, SamplingParams(max_tokens=80,
                                                  ignore_eos=True,
                                                  priority=5))
    lo = eng.add_request([2] * 40, SamplingParams(max_tokens=80,
                                                  ignore_eos=True,
                                                  priority=0))
    preempted = None
    for _ in range(300):
        eng.step()
        for rid in (hi, lo):
            seq = eng.seqs.get(rid)
            if seq is not None and seq.preemptions + seq.swap_outs > 0:
                preempted = rid
                break
        if preempted:
            break
        if not eng.has_unfinished():
            break
    assert preempted == lo  # the low-priority request got evicted
    # high-priority request runs to completion
    while eng.has_unfinished():
        eng.step()
    assert eng.seqs == {} or all(s.status.value == "finished"
                                 for s in eng.seqs.values())
