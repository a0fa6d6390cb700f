"""Engine-level GPU tests: the full HIP path (graphs, paged KV, spec
decode) must reproduce the same tokens as recomputation from scratch."""
import os

import pytest
import torch

pytestmark = pytest.mark.gpu

from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams


def _cfg(**kw):
    kw.setdefault("model", "llama-3-8b")
    kw.setdefault("device", "cuda")
    kw.setdefault("max_model_len", 1024)
    kw.setdefault("max_num_seqs", 16)
    kw.setdefault("gpu_memory_utilization", 0.2)
    cfg = EngineConfig(**kw)
    cfg.spec.num_layers = 4  # small depth: fast, same code path
    return cfg


PROMPTS = [[11, 12, 13, 14, 15, 16] * 8, [101, 102, 103] * 5, [7] * 33]


@pytest.mark.parametrize("model", ["llama-3-8b", "qwen3-14b", "qwen2.5-7b"])
def test_decode_matches_prefill_gpu(model):
    # covers GQA ratios 4/5/7, qwen2 attention bias, qwen3 per-head qk-norm
    eng = LLMEngine(_cfg(model=model))
    full = eng.generate(PROMPTS[:1], SamplingParams(max_tokens=8, ignore_eos=True))[0]
    del eng
    torch.cuda.empty_cache()
    eng2 = LLMEngine(_cfg(model=model))
    cont = eng2.generate([PROMPTS[0] + full[:4]],
                         SamplingParams(max_tokens=4, ignore_eos=True))[0]
    assert cont == full[4:], f"{cont} != {full[4:]}"
    del eng2
    torch.cuda.empty_cache()


def test_graphs_match_eager_gpu():
    eng = LLMEngine(_cfg())
    with_graphs = eng.generate(PROMPTS, SamplingParams(max_tokens=6, ignore_eos=True))
    del eng
    torch.cuda.empty_cache()
    os.environ["GPUSTACK_AMD_NO_GRAPHS"] = "1"
    try:
        eng2 = LLMEngine(_cfg())
        eager = eng2.generate(PROMPTS, SamplingParams(max_tokens=6, ignore_eos=True))
        del eng2
        torch.cuda.empty_cache()
    finally:
        os.environ.pop("GPUSTACK_AMD_NO_GRAPHS", None)
    assert with_graphs == eager


def test_spec_matches_plain_gpu():
    eng = LLMEngine(_cfg())
    plain = eng.generate(PROMPTS, SamplingParams(max_tokens=10, ignore_eos=True))
    del eng
    torch.cuda.empty_cache()
    eng2 = LLMEngine(_cfg(speculative={"method": "ngram", "num_draft_tokens": 3}))
    spec = eng2.generate(PROMPTS, SamplingParams(max_tokens=10, ignore_eos=True))
    del eng2
    torch.cuda.empty_cache()
    assert spec == plain


def test_embed_gpu():
    eng = LLMEngine(_cfg())
    vecs = eng.runner.embed([[1, 2, 3, 4] * 10, [7, 8, 9]])
    assert len(vecs) == 2 and len(vecs[0]) == eng.cfg.spec.hidden_size
    import math

    for v in vecs:
        assert abs(math.sqrt(sum(x * x for x in v)) - 1.0) < 1e-3
    # KV pool untouched
    assert eng.scheduler.kv.allocator.num_free == eng.scheduler.kv.allocator.num_blocks
    del eng
    torch.cuda.empty_cache()


def test_eagle_matches_plain_gpu():
    # draft-model speculative on the HIP path: identical output to plain
    eng = LLMEngine(_cfg())
    plain = eng.generate(PROMPTS, SamplingParams(max_tokens=10, ignore_eos=True))
    del eng
    torch.cuda.empty_cache()
    eng2 = LLMEngine(_cfg(speculative={"method": "eagle", "num_draft_tokens": 3}))
    assert eng2.runner.eagle is not None
    out = eng2.generate(PROMPTS, SamplingParams(max_tokens=10, ignore_eos=True))
    del eng2
    torch.cuda.empty_cache()
    assert out == plain


def test_moe_decode_matches_prefill_gpu():
    # MoE routing/dispatch on the HIP path (D=128 decode kernel; 4 layers
    # of the real qwen3-30b-a3b expert geometry)
    eng = LLMEngine(_cfg(model="qwen3-30b-a3b", max_model_len=256))
    full = eng.generate(PROMPTS[:1], SamplingParams(max_tokens=8, ignore_eos=True))[0]
    del eng
    torch.cuda.empty_cache()
    eng2 = LLMEngine(_cfg(model="qwen3-30b-a3b", max_model_len=256))
    cont = eng2.generate([PROMPTS[0] + full[:4]],
                         SamplingParams(max_tokens=4, ignore_eos=True))[0]
    assert cont == full[4:], f"{cont} != {full[4:]}"
    del eng2
    torch.cuda.empty_cache()
