"""Native embedding serving (/v1/embeddings, reference model category)."""
import math

from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams


def _engine():
    return LLMEngine(EngineConfig(model="tiny", device="cpu", kv_cache_blocks=64,
                                  max_model_len=128))


def test_embed_shapes_and_normalization():
    eng = _engine()
    vecs = eng.runner.embed([[1, 2, 3], [4, 5, 6, 7, 8]])
    assert len(vecs) == 2
    assert len(vecs[0]) == eng.cfg.spec.hidden_size
    for v in vecs:
        assert abs(math.sqrt(sum(x * x for x in v)) - 1.0) < 1e-4


def test_embed_deterministic_and_discriminative():
    eng = _engine()
    a1, a2, b = eng.runner.embed([[1, 2, 3]]) + eng.runner.embed([[1, 2, 3], [9, 8, 7]])
    dot_same = sum(x * y for x, y in zip(a1, a2))
    dot_diff = sum(x * y for x, y in zip(a1, b))
    assert dot_same > 0.999
    assert dot_diff < 0.999


def test_embed_pooling_modes_differ():
    eng = _engine()
    last = eng.runner.embed([[1, 2, 3, 4]], pooling="last")[0]
    mean = eng.runner.embed([[1, 2, 3, 4]], pooling="mean")[0]
    assert last != mean


def test_embed_does_not_disturb_generation():
    eng = _engine()
    before = eng.generate([[5, 6, 7]], SamplingParams(max_tokens=5, ignore_eos=True))
    eng.runner.embed([[1, 2, 3]])
    after = eng.generate([[5, 6, 7]], SamplingParams(max_tokens=5, ignore_eos=True))
    assert before == after
    # KV pool untouched (slot -1 writes skipped)
    assert eng.scheduler.kv.allocator.num_free == eng.scheduler.kv.allocator.num_blocks
