"""Dynamic multi-LoRA serving: per-request adapters, row isolation,
add/remove at runtime (reference: vLLM dynamic adapters surfaced by
gpustack's per-LoRA model routes, gpustack/server/lora_model_routes.py)."""
import json
import tempfile
from pathlib import Path

import pytest
import torch
from safetensors.torch import save_file

from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams


def _cfg(**kw):
    kw.setdefault("model", "tiny")
    kw.setdefault("device", "cpu")
    kw.setdefault("kv_cache_blocks", 64)
    return EngineConfig(**kw)


def _make_adapter(tmp: Path, spec, seed=7, r=4, alpha=8):
    torch.manual_seed(seed)
    tensors = {}
    d = spec.head_dim
    for li in range(spec.num_layers):
        pre = f"base_model.model.model.layers.{li}.self_attn.q_proj"
        tensors[f"{pre}.lora_A.weight"] = torch.randn(r, spec.hidden_size) * 0.05
        tensors[f"{pre}.lora_B.weight"] = torch.randn(spec.num_heads * d, r) * 0.05
        pre = f"base_model.model.model.layers.{li}.mlp.down_proj"
        tensors[f"{pre}.lora_A.weight"] = torch.randn(r, spec.intermediate_size) * 0.05
        tensors[f"{pre}.lora_B.weight"] = torch.randn(spec.hidden_size, r) * 0.05
    save_file(tensors, str(tmp / "adapter_model.safetensors"))
    (tmp / "adapter_config.json").write_text(json.dumps({"r": r, "lora_alpha": alpha}))


PROMPT = [1, 2, 3, 4, 5]
P = SamplingParams(max_tokens=8, ignore_eos=True)


def test_dynamic_lora_changes_output_and_isolates_rows():
    tmp = Path(tempfile.mkdtemp())
    spec = EngineConfig(model="tiny").spec
    _make_adapter(tmp, spec)
    base_out = LLMEngine(_cfg()).generate([PROMPT], P)[0]

    eng = LLMEngine(_cfg())
    eng.add_lora("tuned", str(tmp))
    lp = SamplingParams(max_tokens=8, ignore_eos=True, lora_name="tuned")
    # one batch, mixed adapters: base row must be untouched by the adapter row
    rid_base = eng.add_request(PROMPT, P)
    rid_lora = eng.add_request(PROMPT, lp)
    results = {rid_base: [], rid_lora: []}
    while eng.has_unfinished():
        for o in eng.step():
            results[o.request_id].append(o.token_id)
    assert results[rid_base] == base_out          # isolation
    assert results[rid_lora] != base_out          # adapter took effect


def test_dynamic_matches_merged():
    tmp = Path(tempfile.mkdtemp())
    spec = EngineConfig(model="tiny").spec
    _make_adapter(tmp, spec)
    merged = LLMEngine(_cfg(lora_dirs=[str(tmp)])).generate([PROMPT], P)[0]
    eng = LLMEngine(_cfg())
    eng.add_lora("tuned", str(tmp))
    dyn = eng.generate(
        [PROMPT], SamplingParams(max_tokens=8, ignore_eos=True,
                                 lora_name="tuned"))[0]
    assert dyn == merged


def test_two_adapters_concurrently():
    spec = EngineConfig(model="tiny").spec
    t1, t2 = Path(tempfile.mkdtemp()), Path(tempfile.mkdtemp())
    _make_adapter(t1, spec, seed=7)
    _make_adapter(t2, spec, seed=11)

    def solo(tmp):
        e = LLMEngine(_cfg())
        e.add_lora("x", str(tmp))
        return e.generate([PROMPT], SamplingParams(
            max_tokens=8, ignore_eos=True, lora_name="x"))[0]

    solo1, solo2 = solo(t1), solo(t2)
    assert solo1 != solo2
    eng = LLMEngine(_cfg())
    eng.add_lora("a", str(t1))
    eng.add_lora("b", str(t2))
    r1 = eng.add_request(PROMPT, SamplingParams(max_tokens=8, ignore_eos=True,
                                                lora_name="a"))
    r2 = eng.add_request(PROMPT, SamplingParams(max_tokens=8, ignore_eos=True,
                                                lora_name="b"))
    results = {r1: [], r2: []}
    while eng.has_unfinished():
        for o in eng.step():
            results[o.request_id].append(o.token_id)
    assert results[r1] == solo1
    assert results[r2] == solo2


def test_add_remove_lifecycle():
    tmp = Path(tempfile.mkdtemp())
    spec = EngineConfig(model="tiny").spec
    _make_adapter(tmp, spec)
    eng = LLMEngine(_cfg())
    base_out = eng.generate([PROMPT], P)[0]
    eng.add_lora("t", str(tmp))
    assert "t" in eng.lora_names()
    lp = SamplingParams(max_tokens=8, ignore_eos=True, lora_name="t")
    assert eng.generate([PROMPT], lp)[0] != base_out
    assert eng.remove_lora("t")
    assert "t" not in eng.lora_names()
    # unknown adapter falls back to the base model (with a warning)
    assert eng.generate([PROMPT], lp)[0] == base_out
    assert not eng.remove_lora("t")


def test_moe_mlp_adapter_rejected():
    tmp = Path(tempfile.mkdtemp())
    spec = EngineConfig(model="tiny-moe").spec
    _make_adapter(tmp, spec)  # includes down_proj -> must be rejected
    eng = LLMEngine(_cfg(model="tiny-moe"))
    with pytest.raises(ValueError):
        eng.add_lora("bad", str(tmp))


@pytest.mark.timeout(240)
def test_engine_server_dynamic_lora_endpoints():
    """vLLM-compatible /v1/load_lora_adapter + adapter-name request routing
    through the engine server (single rank, CPU)."""
    import socket
    import subprocess
    import sys
    import time

    import httpx

    tmp = Path(tempfile.mkdtemp())
    spec = EngineConfig(model="tiny").spec
    _make_adapter(tmp, spec)

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    proc = subprocess.Popen([
        sys.executable, "-m", "gpustack_amd.worker.engine_server",
        "--served-name", "tiny-l", "--source", "preset", "--model-ref", "tiny",
        "--port", str(port), "--max-model-len", "256",
        "--device", "cpu", "--kv-cache-blocks", "64",
    ])
    base_url = f"http://127.0.0.1:{port}"
    try:
        t0 = time.time()
        while time.time() - t0 < 90:
            if proc.poll() is not None:
                raise AssertionError(f"engine server exited {proc.returncode}")
            try:
                if httpx.get(f"{base_url}/health", timeout=2).status_code == 200:
                    break
            except httpx.HTTPError:
                pass
            time.sleep(0.5)
        else:
            raise AssertionError("engine server never became healthy")

        body = {"model": "tiny-l", "prompt": "hello", "max_tokens": 8,
                "ignore_eos": True, "temperature": 0}
        base_text = httpx.post(f"{base_url}/v1/completions", json=body,
                               timeout=60).json()["choices"][0]["text"]
        r = httpx.post(f"{base_url}/v1/load_lora_adapter", json={
            "lora_name": "tuned", "lora_path": str(tmp)}, timeout=60)
        assert r.status_code == 200
        ids = [m["id"] for m in httpx.get(f"{base_url}/v1/models",
                                          timeout=10).json()["data"]]
        assert "tuned" in ids
        lora_text = httpx.post(f"{base_url}/v1/completions", json={
            **body, "model": "tuned"}, timeout=60).json()["choices"][0]["text"]
        assert lora_text != base_text
        # base name still serves the unmodified model
        again = httpx.post(f"{base_url}/v1/completions", json=body,
                           timeout=60).json()["choices"][0]["text"]
        assert again == base_text
        r = httpx.post(f"{base_url}/v1/unload_lora_adapter",
                       json={"lora_name": "tuned"}, timeout=60)
        assert r.status_code == 200
        r = httpx.post(f"{base_url}/v1/unload_lora_adapter",
                       json={"lora_name": "tuned"}, timeout=60)
        assert r.status_code == 404
        # bad path is a 400, not a server crash
        r = httpx.post(f"{base_url}/v1/load_lora_adapter", json={
            "lora_name": "x", "lora_path": "/nonexistent"}, timeout=60)
        assert r.status_code == 400
    finally:
        proc.terminate()
        try:
            proc.wait(timeout=15)
        except subprocess.TimeoutExpired:
            proc.kill()
