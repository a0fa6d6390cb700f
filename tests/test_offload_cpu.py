"""CPU weight offload (engine/offload.py) + offload placement policy."""
import torch

from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams


def test_offload_engine_output_identical():
    """Streamed-weight execution must be bit-identical to resident
    execution (same kernels, same data — only residency differs)."""
    p = SamplingParams(max_tokens=10, ignore_eos=True)
    plain = LLMEngine(EngineConfig(model="tiny", device="cpu",
                                   kv_cache_blocks=64))
    want = plain.generate([[1, 2, 3, 4, 5]], p)[0]
    eng = LLMEngine(EngineConfig(model="tiny", device="cpu",
                                 kv_cache_blocks=64, cpu_offload_gb=1.0))
    off = eng.runner.model.offload
    assert off is not None and off.first >= 1
    # offloaded layers' parameters were freed
    lay = eng.runner.model.layers[off.first]
    assert lay.attn.qkv_w.numel() == 0
    assert eng.generate([[1, 2, 3, 4, 5]], p)[0] == want
    # second run re-binds cleanly
    assert eng.generate([[9, 8, 7]], p)[0] == plain.generate([[9, 8, 7]], p)[0]


def test_offload_placement_policy():
    from gpustack_amd.scheduler.policies import select_candidates

    dev = {"index": 0, "type": "rocm",
           "memory": {"total": 12 << 30}}
    worker = {"id": 1, "name": "w1", "state": "ready",
              "status": {"gpu_devices": [dev],
                         "memory": {"total": 512 << 30}},
              "system_reserved": {}}
    # llama-3-8b bf16 claim (~25 GiB) never fits a 12 GiB GPU
    model = {"id": 1, "source": "preset", "model_ref": "llama-3-8b",
             "gpus_per_replica": 1, "backend_parameters": {}}
    assert select_candidates(model, [worker], []) == []
    model["backend_parameters"] = {"cpu_offload": True}
    cands = select_candidates(model, [worker], [])
    assert len(cands) == 1
    c = cands[0]
    assert c.offload_gb > 0 and c.offload_layers >= 1
    assert c.ram_claim > c.offload_gb * (2 << 29)
    assert c.vram_claim[0] <= int((12 << 30) * 0.9)


def test_offload_refused_when_too_little_resident():
    from gpustack_amd.scheduler.policies import select_candidates

    dev = {"index": 0, "type": "rocm", "memory": {"total": 7 << 30}}
    worker = {"id": 1, "name": "w1", "state": "ready",
              "status": {"gpu_devices": [dev], "memory": {"total": 512 << 30}},
              "system_reserved": {}}
    # 70B bf16 (~140 GiB): <10% would stay resident on a 7 GiB GPU
    model = {"id": 1, "source": "preset", "model_ref": "llama-3-70b",
             "gpus_per_replica": 1,
             "backend_parameters": {"cpu_offload": True}}
    assert select_candidates(model, [worker], []) == []


def test_offload_composes_with_w4_runtime():
    """W4-packed layers have freed (0-element) parameters; the offload
    streamer must skip them and still stream any remaining bf16 weights."""
    p = SamplingParams(max_tokens=6, ignore_eos=True)
    eng = LLMEngine(EngineConfig(model="tiny", device="cpu",
                                 kv_cache_blocks=64, quantize_runtime="w4",
                                 cpu_offload_gb=1.0))
    ref = LLMEngine(EngineConfig(model="tiny", device="cpu",
                                 kv_cache_blocks=64, quantize_runtime="w4"))
    out = eng.generate([[3, 4, 5]], p)[0]
    assert out == ref.generate([[3, 4, 5]], p)[0]


def test_engine_server_accepts_w4_and_offload_backend_params():
    """serve_manager passes quantize_runtime / cpu_offload_gb through
    backend_parameters into EngineConfig (engine_server filters on
    dataclass fields)."""
    from gpustack_amd.engine import EngineConfig

    extra = {"quantize_runtime": "w4", "cpu_offload_gb": 1.5,
             "bogus_key": 1}
    kept = {k: v for k, v in extra.items()
            if k in EngineConfig.__dataclass_fields__}
    assert kept == {"quantize_runtime": "w4", "cpu_offload_gb": 1.5}
    cfg = EngineConfig(model="tiny", device="cpu", **kept)
    assert cfg.quantize_runtime == "w4" and cfg.cpu_offload_gb == 1.5
