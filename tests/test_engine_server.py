"""Engine server process tests (OpenAI surface + TP spawn) on CPU."""
import socket
import subprocess
import sys
import time

import httpx
import pytest


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _wait_health(port: int, proc, timeout=90) -> None:
    t0 = time.time()
    while time.time() - t0 < timeout:
        if proc.poll() is not None:
            raise AssertionError(f"engine server exited {proc.returncode}")
        try:
            r = httpx.get(f"http://127.0.0.1:{port}/health", timeout=2)
            if r.status_code == 200:
                return
        except httpx.HTTPError:
            pass
        time.sleep(0.5)
    raise AssertionError("engine server never became healthy")


@pytest.mark.timeout(240)
def test_engine_server_tp2_cpu():
    """gpus_per_replica=2 path: rank 0 spawns a follower, both step in
    lockstep over gloo, OpenAI endpoint answers."""
    port = _free_port()
    proc = subprocess.Popen([
        sys.executable, "-m", "gpustack_amd.worker.engine_server",
        "--served-name", "tiny-tp", "--source", "preset", "--model-ref", "tiny",
        "--port", str(port), "--max-model-len", "256", "--tp", "2",
        "--device", "cpu", "--kv-cache-blocks", "64",
    ])
    try:
        _wait_health(port, proc)
        r = httpx.post(f"http://127.0.0.1:{port}/v1/completions", json={
            "model": "tiny-tp", "prompt": "ab", "max_tokens": 6,
            "ignore_eos": True, "temperature": 0,
        }, timeout=60)
        assert r.status_code == 200, r.text
        assert r.json()["usage"]["completion_tokens"] == 6
        # second request exercises the idle->active->idle transition
        r = httpx.post(f"http://127.0.0.1:{port}/v1/completions", json={
            "model": "tiny-tp", "prompt": "xyz", "max_tokens": 4,
            "ignore_eos": True, "temperature": 0,
        }, timeout=60)
        assert r.status_code == 200
    finally:
        proc.terminate()
        try:
            proc.wait(timeout=15)
        except subprocess.TimeoutExpired:
            proc.kill()


@pytest.mark.timeout(240)
def test_engine_server_stop_strings():
    port = _free_port()
    proc = subprocess.Popen([
        sys.executable, "-m", "gpustack_amd.worker.engine_server",
        "--served-name", "tiny-s", "--source", "preset", "--model-ref", "tiny",
        "--port", str(port), "--max-model-len", "256",
        "--device", "cpu", "--kv-cache-blocks", "64",
    ])
    try:
        _wait_health(port, proc)
        # find what the model actually generates, then stop on a substring
        r = httpx.post(f"http://127.0.0.1:{port}/v1/completions", json={
            "model": "tiny-s", "prompt": "hello", "max_tokens": 20,
            "ignore_eos": True, "temperature": 0,
        }, timeout=60)
        full = r.json()["choices"][0]["text"]
        if len(full) > 4:
            stop = full[2:4]
            r = httpx.post(f"http://127.0.0.1:{port}/v1/completions", json={
                "model": "tiny-s", "prompt": "hello", "max_tokens": 20,
                "ignore_eos": True, "temperature": 0, "stop": [stop],
            }, timeout=60)
            text = r.json()["choices"][0]["text"]
            assert stop not in text
            assert len(text) <= len(full)
        # embeddings endpoint responds too
        r = httpx.post(f"http://127.0.0.1:{port}/v1/embeddings", json={
            "model": "tiny-s", "input": ["abc", "def"]}, timeout=60)
        assert r.status_code == 200
        assert len(r.json()["data"]) == 2
        # rerank: identical doc must out-score an unrelated one
        r = httpx.post(f"http://127.0.0.1:{port}/v1/rerank", json={
            "model": "tiny-s", "query": "abcabc",
            "documents": ["abcabc", "zzqqwwx"], "top_n": 2}, timeout=60)
        assert r.status_code == 200
        res = r.json()["results"]
        assert len(res) == 2
        assert res[0]["index"] == 0  # self-similarity ranks first
        assert res[0]["relevance_score"] >= res[1]["relevance_score"]
        # n>1 choices (independent samples through the batcher)
        r = httpx.post(f"http://127.0.0.1:{port}/v1/completions", json={
            "model": "tiny-s", "prompt": "hi", "max_tokens": 5, "n": 3,
            "ignore_eos": True, "temperature": 0.8, "seed": 1}, timeout=60)
        assert r.status_code == 200
        ch = r.json()["choices"]
        assert [c["index"] for c in ch] == [0, 1, 2]
        assert r.json()["usage"]["completion_tokens"] == 15
        assert len({c["text"] for c in ch}) > 1  # seeds differ per choice

        # guided_choice: output constrained to one of the given strings
        r = httpx.post(f"http://127.0.0.1:{port}/v1/completions", json={
            "model": "tiny-s", "prompt": "pick", "max_tokens": 20,
            "guided_choice": ["yes", "no"]}, timeout=60)
        assert r.status_code == 200
        assert r.json()["choices"][0]["text"] in ("yes", "no")

        # score: same-text pair scores ~1.0
        r = httpx.post(f"http://127.0.0.1:{port}/v1/score", json={
            "text_1": "abcabc", "text_2": ["abcabc", "zzqq"]}, timeout=60)
        assert r.status_code == 200
        d = r.json()["data"]
        assert d[0]["score"] > 0.99 and d[0]["score"] >= d[1]["score"]
    finally:
        proc.terminate()
        try:
            proc.wait(timeout=15)
        except subprocess.TimeoutExpired:
            proc.kill()


@pytest.mark.timeout(240)
def test_anthropic_messages_endpoint():
    port = _free_port()
    proc = subprocess.Popen([
        sys.executable, "-m", "gpustack_amd.worker.engine_server",
        "--served-name", "tiny-a", "--source", "preset", "--model-ref", "tiny",
        "--port", str(port), "--max-model-len", "256",
        "--device", "cpu", "--kv-cache-blocks", "64",
    ])
    try:
        _wait_health(port, proc)
        r = httpx.post(f"http://127.0.0.1:{port}/v1/messages", json={
            "model": "tiny-a", "max_tokens": 8,
            "system": "be brief",
            "messages": [{"role": "user", "content": "hello"}],
        }, timeout=60)
        assert r.status_code == 200
        j = r.json()
        assert j["type"] == "message" and j["role"] == "assistant"
        assert j["content"][0]["type"] == "text"
        assert j["stop_reason"] in ("end_turn", "max_tokens")
        assert j["usage"]["output_tokens"] > 0
        # streaming: anthropic event sequence
        with httpx.stream("POST", f"http://127.0.0.1:{port}/v1/messages", json={
            "model": "tiny-a", "max_tokens": 6, "stream": True,
            "messages": [{"role": "user",
                          "content": [{"type": "text", "text": "hi"}]}],
        }, timeout=60) as r:
            events = [ln.split(" ", 1)[1] for ln in r.iter_lines()
                      if ln.startswith("event: ")]
        assert events[0] == "message_start"
        assert "content_block_delta" in events
        assert events[-1] == "message_stop"
    finally:
        proc.terminate()
        try:
            proc.wait(timeout=15)
        except subprocess.TimeoutExpired:
            proc.kill()


@pytest.mark.timeout(240)
def test_responses_and_count_tokens():
    """OpenAI Responses API (minimal) + Anthropic count_tokens."""
    import httpx

    port = _free_port()
    proc = subprocess.Popen([
        sys.executable, "-m", "gpustack_amd.worker.engine_server",
        "--served-name", "tiny-r", "--source", "preset", "--model-ref", "tiny",
        "--port", str(port), "--max-model-len", "256",
        "--device", "cpu", "--kv-cache-blocks", "64",
    ])
    base = f"http://127.0.0.1:{port}"
    try:
        _wait_health(port, proc)
        r = httpx.post(f"{base}/v1/responses", json={
            "model": "tiny-r", "input": "hello there",
            "max_output_tokens": 6, "temperature": 0, "ignore_eos": True,
        }, timeout=60)
        assert r.status_code == 200, r.text
        d = r.json()
        assert d["object"] == "response" and d["status"] == "completed"
        assert d["output"][0]["content"][0]["type"] == "output_text"
        assert d["output_text"] == d["output"][0]["content"][0]["text"]
        assert d["usage"]["output_tokens"] == 6

        # message-list input with typed content parts
        r = httpx.post(f"{base}/v1/responses", json={
            "model": "tiny-r", "instructions": "be brief",
            "input": [{"role": "user",
                       "content": [{"type": "input_text", "text": "hi"}]}],
            "max_output_tokens": 4, "temperature": 0, "ignore_eos": True,
        }, timeout=60)
        assert r.status_code == 200
        assert r.json()["usage"]["output_tokens"] == 4

        r = httpx.post(f"{base}/v1/messages/count_tokens", json={
            "model": "tiny-r",
            "messages": [{"role": "user", "content": "hello"}]}, timeout=30)
        assert r.status_code == 200
        assert r.json()["input_tokens"] == 5  # ByteTokenizer: 1 token/byte
    finally:
        proc.terminate()
        try:
            proc.wait(timeout=15)
        except subprocess.TimeoutExpired:
            proc.kill()


@pytest.mark.timeout(240)
def test_streaming_stop_strings():
    """Stop strings truncate STREAMED output too (non-stream path already
    covered by test_engine_server_stop_strings)."""
    import json as _json

    import httpx

    port = _free_port()
    proc = subprocess.Popen([
        sys.executable, "-m", "gpustack_amd.worker.engine_server",
        "--served-name", "tiny-ss", "--source", "preset", "--model-ref", "tiny",
        "--port", str(port), "--max-model-len", "256",
        "--device", "cpu", "--kv-cache-blocks", "64",
    ])
    base = f"http://127.0.0.1:{port}"
    try:
        _wait_health(port, proc)
        r = httpx.post(f"{base}/v1/completions", json={
            "model": "tiny-ss", "prompt": "hello", "max_tokens": 20,
            "ignore_eos": True, "temperature": 0}, timeout=60)
        full = r.json()["choices"][0]["text"]
        if len(full) <= 4:
            pytest.skip("output too short to carve a stop string")
        stop = full[2:4]
        chunks = []
        with httpx.stream("POST", f"{base}/v1/completions", json={
            "model": "tiny-ss", "prompt": "hello", "max_tokens": 20,
            "ignore_eos": True, "temperature": 0, "stream": True,
            "stop": [stop],
        }, timeout=60) as resp:
            for line in resp.iter_lines():
                if line.startswith("data:") and "[DONE]" not in line:
                    chunks.append(_json.loads(line[5:]))
        text = "".join(c["choices"][0].get("text", "") for c in chunks)
        assert stop not in text
        assert text == full[:2]
        assert chunks[-1]["choices"][0]["finish_reason"] == "stop"
    finally:
        proc.terminate()
        try:
            proc.wait(timeout=15)
        except subprocess.TimeoutExpired:
            proc.kill()


@pytest.mark.timeout(240)
def test_streaming_logprobs():
    import httpx
    import json as _json

    port = _free_port()
    proc = subprocess.Popen([
        sys.executable, "-m", "gpustack_amd.worker.engine_server",
        "--served-name", "tiny-lp", "--source", "preset", "--model-ref", "tiny",
        "--port", str(port), "--max-model-len", "256",
        "--device", "cpu", "--kv-cache-blocks", "64",
    ])
    try:
        _wait_health(port, proc)
        lps = []
        with httpx.stream("POST", f"http://127.0.0.1:{port}/v1/completions",
                          json={"model": "tiny-lp", "prompt": "x",
                                "max_tokens": 5, "ignore_eos": True,
                                "temperature": 0, "stream": True,
                                "logprobs": True}, timeout=60) as resp:
            for line in resp.iter_lines():
                if line.startswith("data:") and "[DONE]" not in line:
                    ch = _json.loads(line[5:])["choices"][0]
                    if ch.get("logprobs"):
                        lps.extend(ch["logprobs"]["token_logprobs"])
        assert len(lps) >= 4
        assert all(isinstance(v, float) and v <= 0 for v in lps)
    finally:
        proc.terminate()
        try:
            proc.wait(timeout=15)
        except subprocess.TimeoutExpired:
            proc.kill()


def test_engine_metrics_histograms():
    """vLLM-parity latency observability: TTFT and inter-token-gap
    histograms appear in the engine /metrics after serving requests."""
    port = _free_port()
    proc = subprocess.Popen([
        sys.executable, "-m", "gpustack_amd.worker.engine_server",
        "--served-name", "tiny-h", "--source", "preset", "--model-ref",
        "tiny", "--port", str(port), "--max-model-len", "256",
        "--device", "cpu", "--kv-cache-blocks", "64",
    ])
    try:
        _wait_health(port, proc)
        for _ in range(3):
            r = httpx.post(f"http://127.0.0.1:{port}/v1/completions", json={
                "model": "tiny-h", "prompt": "hello", "max_tokens": 6,
                "ignore_eos": True}, timeout=60)
            assert r.status_code == 200
        m = httpx.get(f"http://127.0.0.1:{port}/metrics", timeout=10).text
        assert "gpustack_engine_ttft_seconds_bucket" in m
        assert "gpustack_engine_time_per_output_token_seconds_count" in m
        import re as _re

        ttft_count = int(_re.search(
            r"gpustack_engine_ttft_seconds_count (\d+)", m).group(1))
        assert ttft_count >= 3
        tpot_count = int(_re.search(
            r"gpustack_engine_time_per_output_token_seconds_count (\d+)",
            m).group(1))
        assert tpot_count >= 3 * 4  # >=5 gaps per 6-token request, 3 reqs
        # cumulative histogram sanity: +Inf bucket equals count
        inf = int(_re.search(
            r'gpustack_engine_ttft_seconds_bucket\{le="\+Inf"\} (\d+)',
            m).group(1))
        assert inf == ttft_count
    finally:
        proc.terminate()
        proc.wait(timeout=10)
