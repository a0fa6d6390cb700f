"""fp8-e4m3 KV cache (opt-in): engine runs, outputs close to bf16 KV."""
import torch

from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams


def test_fp8_kv_cache_cpu():
    p = SamplingParams(max_tokens=12, ignore_eos=True)
    prompts = [[1, 2, 3, 4, 5, 6], [9, 8, 7]]
    bf = LLMEngine(EngineConfig(model="tiny", device="cpu", kv_cache_blocks=64,
                                max_model_len=128)).generate(prompts, p)
    eng = LLMEngine(EngineConfig(model="tiny", device="cpu", kv_cache_blocks=64,
                                 max_model_len=128, kv_cache_dtype="fp8"))
    assert eng.scheduler.kv.k_caches[0].dtype == torch.float8_e4m3fn
    fp8 = eng.generate(prompts, p)
    assert all(len(o) == 12 for o in fp8)
    # decode==prefill consistency holds under fp8 quantization too
    eng2 = LLMEngine(EngineConfig(model="tiny", device="cpu", kv_cache_blocks=64,
                                  max_model_len=128, kv_cache_dtype="fp8"))
    cont = eng2.generate([prompts[0] + fp8[0][:6]],
                         SamplingParams(max_tokens=6, ignore_eos=True))[0]
    assert cont == fp8[0][6:]


def test_fp8_capacity_doubles():
    from gpustack_amd.engine.kv_cache import KVCache

    a = KVCache.compute_num_blocks(EngineConfig(model="tiny", device="cpu"), 1 << 30)
    b = KVCache.compute_num_blocks(
        EngineConfig(model="tiny", device="cpu", kv_cache_dtype="fp8"), 1 << 30)
    assert b in (2 * a, 2 * a + 1)  # integer division rounding
