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


def test_generative_rerank_mode():
    """Qwen3-Reranker-style scoring on a generative engine: relevance =
    sigmoid(logit_yes - logit_no) at the judgment position; results are
    sorted, bounded to (0,1), and deterministic."""
    import socket
    import subprocess
    import sys
    import time

    import httpx

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    proc = subprocess.Popen([
        sys.executable, "-m", "gpustack_amd.worker.engine_server",
        "--served-name", "tiny-gr", "--source", "preset",
        "--model-ref", "tiny", "--port", str(port), "--device", "cpu",
        "--kv-cache-blocks", "64", "--max-model-len", "256",
    ])
    try:
        t0 = time.time()
        while time.time() - t0 < 90:
            if proc.poll() is not None:
                raise AssertionError(f"server exited {proc.returncode}")
            try:
                if httpx.get(f"http://127.0.0.1:{port}/health",
                             timeout=2).status_code == 200:
                    break
            except httpx.HTTPError:
                time.sleep(0.5)
        docs = ["alpha beta", "gamma", "delta epsilon zeta"]
        r = httpx.post(f"http://127.0.0.1:{port}/v1/rerank", json={
            "model": "tiny-gr", "query": "alpha", "documents": docs,
            "mode": "generative"}, timeout=60)
        assert r.status_code == 200, r.text
        res = r.json()["results"]
        assert len(res) == 3
        assert all(0.0 < x["relevance_score"] < 1.0 for x in res)
        scores = [x["relevance_score"] for x in res]
        assert scores == sorted(scores, reverse=True)
        r2 = httpx.post(f"http://127.0.0.1:{port}/v1/rerank", json={
            "model": "tiny-gr", "query": "alpha", "documents": docs,
            "mode": "generative"}, timeout=60)
        assert r2.json()["results"] == res  # deterministic
    finally:
        proc.terminate()
        proc.wait(timeout=10)
