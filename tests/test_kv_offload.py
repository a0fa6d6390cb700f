"""Host-DRAM KV offload tier (extended_kv_cache parity) on CPU."""
import torch

from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams
from gpustack_amd.engine.kv_cache import KVCache


def test_swap_roundtrip_preserves_kv():
    cfg = EngineConfig(model="tiny", device="cpu", kv_cache_blocks=8,
                       kv_offload_gb=0.001)
    kv = KVCache(cfg, 8, "cpu")
    assert kv.host_blocks > 0
    blocks = kv.allocator.allocate(2)
    for li in range(kv.num_layers):
        kv.k_caches[li][blocks[0]].normal_()
        kv.v_caches[li][blocks[1]].normal_()
    snap_k = [kv.k_caches[li][blocks[0]].clone() for li in range(kv.num_layers)]
    snap_v = [kv.v_caches[li][blocks[1]].clone() for li in range(kv.num_layers)]
    hb = kv.swap_out(blocks)
    # dirty the (now free) gpu blocks
    for li in range(kv.num_layers):
        kv.k_caches[li].zero_()
        kv.v_caches[li].zero_()
    back = kv.swap_in(hb)
    for li in range(kv.num_layers):
        assert torch.equal(kv.k_caches[li][back[0]], snap_k[li])
        assert torch.equal(kv.v_caches[li][back[1]], snap_v[li])


def test_offload_preserves_outputs_under_pressure():
    prompts = [[1, 2, 3], [4, 5, 6]]
    p = SamplingParams(max_tokens=40, ignore_eos=True)
    big = LLMEngine(EngineConfig(model="tiny", device="cpu", kv_cache_blocks=128,
                                 max_model_len=64)).generate(prompts, p)
    eng = LLMEngine(EngineConfig(model="tiny", device="cpu", kv_cache_blocks=5,
                                 max_model_len=64, kv_offload_gb=0.01))
    out = eng.generate(prompts, p)
    assert out == big


def test_offload_swaps_instead_of_recompute():
    # with a big host tier, pressure leads to swap-outs, not recompute
    eng = LLMEngine(EngineConfig(model="tiny", device="cpu", kv_cache_blocks=5,
                                 max_model_len=64, kv_offload_gb=0.01))
    rids = [eng.add_request([i, i + 1, i + 2], SamplingParams(max_tokens=40, ignore_eos=True))
            for i in range(2)]
    seqs = [eng.seqs[r] for r in rids]
    while eng.has_unfinished():
        eng.step()
    assert sum(s.swap_outs for s in seqs) > 0
    assert all(s.preemptions == 0 for s in seqs)
