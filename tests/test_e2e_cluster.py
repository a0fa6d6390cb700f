"""End-to-end cluster test on CPU (BASELINE.json config 1: server +
scheduler + gateway with 1 CPU worker serving a tiny model greedily).

Boots a real uvicorn server, a real worker agent (static GPU override so
the fit filter passes on a GPU-less host), deploys a model through the API,
waits for the full PENDING -> SCHEDULED -> STARTING -> RUNNING loop
(engine subprocess included), then round-trips /v1/chat/completions and
/v1/completions through the server-side proxy, checks usage metering and
instance logs, and scales to zero.
"""
import socket
import tempfile
import threading
import time

import httpx
import pytest


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


@pytest.fixture(scope="module")
def cluster():
    import uvicorn

    from gpustack_amd.config import Config
    from gpustack_amd.server.app import create_app
    from gpustack_amd.worker.agent import WorkerAgent

    wport = _free_port()
    lo = _free_port()
    # retry the server bind: _free_port() closes the probe socket before
    # uvicorn rebinds it, so a parallel process can steal the port (seen
    # as a transient fixture error under full-suite load)
    app = server = None
    base = ""
    for _attempt in range(3):
        sport = _free_port()
        cfg = Config(
            data_dir=tempfile.mkdtemp(), bootstrap_password="pw",
            host="127.0.0.1", port=sport,
        )
        app = create_app(cfg, start_background=True)
        server = uvicorn.Server(uvicorn.Config(app, host="127.0.0.1",
                                               port=sport,
                                               log_level="warning"))
        st = threading.Thread(target=server.run, daemon=True)
        st.start()
        base = f"http://127.0.0.1:{sport}"
        up = False
        for _ in range(100):
            try:
                httpx.get(base + "/healthz", timeout=1)
                up = True
                break
            except httpx.HTTPError:
                time.sleep(0.1)
        if up:
            break
        from gpustack_amd.server.app import stop_background_tasks

        stop_background_tasks(app)
        server.should_exit = True
    else:
        raise RuntimeError("e2e server never became healthy")

    wcfg = Config(
        data_dir=tempfile.mkdtemp(),
        server_url=base,
        token=app.state.bootstrap["registration_token"],
        worker_name="cpu-worker-0",
        worker_ip="127.0.0.1",
        worker_port=wport,
        port_range=f"{lo}-{lo + 50}",
        gpu_devices=[{"index": 0, "name": "AMD Instinct MI355X",
                      "memory": {"total": 288 * 1024**3}}],
        heartbeat_interval=2.0,
        worker_status_interval=5.0,
    )
    agent = WorkerAgent(wcfg)
    wt = threading.Thread(target=agent.start, daemon=True)
    wt.start()

    client = httpx.Client(base_url=base, timeout=30)
    tok = client.post("/auth/login", json={"username": "admin", "password": "pw"}).json()["token"]
    client.headers["Authorization"] = f"Bearer {tok}"
    for _ in range(100):
        if client.get("/v2/workers").json()["items"]:
            break
        time.sleep(0.2)
    yield client, agent
    from gpustack_amd.server.app import stop_background_tasks

    stop_background_tasks(app)
    agent.stop()
    server.should_exit = True


@pytest.mark.timeout(180)
def test_deploy_and_chat(cluster):
    client, agent = cluster
    r = client.post("/v2/models", json={
        "name": "tiny-chat", "source": "preset", "model_ref": "tiny",
        "replicas": 1, "max_model_len": 256,
    })
    assert r.status_code == 201, r.text

    state = None
    for _ in range(240):  # engine subprocess needs to import torch
        insts = client.get("/v2/model_instances").json()["items"]
        if insts:
            state = insts[0]["state"]
            if state == "running":
                break
            assert state != "error", insts[0]["state_message"]
        time.sleep(0.5)
    assert state == "running", f"instance never ran (last state: {state})"

    r = client.post("/v1/chat/completions", json={
        "model": "tiny-chat",
        "messages": [{"role": "user", "content": "hello"}],
        "max_tokens": 8, "ignore_eos": True,
    })
    assert r.status_code == 200, r.text
    data = r.json()
    assert data["choices"][0]["message"]["content"] is not None
    assert data["usage"]["completion_tokens"] == 8

    # streaming
    with client.stream("POST", "/v1/chat/completions", json={
        "model": "tiny-chat",
        "messages": [{"role": "user", "content": "stream test"}],
        "max_tokens": 4, "stream": True, "ignore_eos": True,
    }) as resp:
        assert resp.status_code == 200
        frames = [l for l in resp.iter_lines() if l.startswith("data:")]
    assert frames[-1].strip() == "data: [DONE]"

    # completions endpoint
    r = client.post("/v1/completions", json={
        "model": "tiny-chat", "prompt": "abc", "max_tokens": 4, "ignore_eos": True,
    })
    assert r.status_code == 200
    assert r.json()["usage"]["completion_tokens"] == 4

    # anthropic-style messages endpoint through the gateway
    r = client.post("/v1/messages", json={
        "model": "tiny-chat", "max_tokens": 4,
        "messages": [{"role": "user", "content": "hello"}],
    })
    assert r.status_code == 200, r.text
    assert r.json()["type"] == "message"
    assert r.json()["usage"]["output_tokens"] > 0

    # rerank through the gateway
    r = client.post("/v1/rerank", json={
        "model": "tiny-chat", "query": "abc", "documents": ["abc", "zzz"],
    })
    assert r.status_code == 200, r.text
    assert len(r.json()["results"]) == 2

    # usage metering recorded
    usage = client.get("/v2/usage").json()["items"]
    assert usage and usage[0]["completion_tokens"] >= 12

    # worker /metrics re-exports engine runtime metrics with instance label
    wm = httpx.get(f"http://127.0.0.1:{agent.cfg.worker_port}/metrics", timeout=10)
    assert wm.status_code == 200
    assert 'gpustack_engine_generated_tokens_total{instance="' in wm.text

    # instance logs reachable on the worker API
    insts = client.get("/v2/model_instances").json()["items"]
    wr = httpx.get(
        f"http://127.0.0.1:{agent.cfg.worker_port}/logs/{insts[0]['name']}",
        timeout=10)
    assert wr.status_code == 200

    # /v1/models lists it
    ids = [m["id"] for m in client.get("/v1/models").json()["data"]]
    assert "tiny-chat" in ids

    # scale to zero -> instance deleted -> engine stopped
    mid = client.get("/v2/models").json()["items"][0]["id"]
    client.patch(f"/v2/models/{mid}", json={"replicas": 0})
    for _ in range(60):
        if not client.get("/v2/model_instances").json()["items"]:
            break
        time.sleep(0.5)
    assert not client.get("/v2/model_instances").json()["items"]
    for _ in range(40):
        if not agent.serve_manager.processes:
            break
        time.sleep(0.5)
    assert not agent.serve_manager.processes


@pytest.mark.timeout(180)
def test_benchmark_flow(cluster):
    """In-product benchmark subsystem: fixed-concurrency load against the
    deployed instance, aggregated TTFT/TPOT/TPS posted back."""
    client, agent = cluster
    mid = client.get("/v2/models").json()["items"][0]["id"]
    client.patch(f"/v2/models/{mid}", json={"replicas": 1})
    for _ in range(240):
        insts = client.get("/v2/model_instances").json()["items"]
        if insts and insts[0]["state"] == "running":
            break
        time.sleep(0.5)
    assert insts and insts[0]["state"] == "running"

    r = client.post("/v2/benchmarks", json={
        "name": "b1", "model_name": "tiny-chat", "mode": "concurrency",
        "value": 2, "duration_s": 3.0, "isl": 32, "osl": 8,
    })
    assert r.status_code == 201, r.text
    bid = r.json()["id"]
    state = None
    for _ in range(120):
        b = [x for x in client.get("/v2/benchmarks").json()["items"] if x["id"] == bid][0]
        state = b["state"]
        if state in ("completed", "error"):
            break
        time.sleep(1.0)
    assert state == "completed", b.get("state_message")
    res = b["results"]
    assert res["successful_requests"] > 0
    assert res["output_tps"] > 0
    assert res["ttft_p50_ms"] is not None
    # sweep profile: one point per value, curve in results.profile
    r = client.post("/v2/benchmarks", json={
        "name": "b-sweep", "model_name": "tiny-chat", "mode": "concurrency",
        "sweep": [1, 2], "duration_s": 2.0, "isl": 16, "osl": 4,
    })
    assert r.status_code == 201, r.text
    bid2 = r.json()["id"]
    for _ in range(120):
        b2 = [x for x in client.get("/v2/benchmarks").json()["items"]
              if x["id"] == bid2][0]
        if b2["state"] in ("completed", "error"):
            break
        time.sleep(1.0)
    assert b2["state"] == "completed", b2.get("state_message")
    prof = b2["results"]["profile"]
    assert len(prof) == 2 and {p["value"] for p in prof} == {1.0, 2.0}

    # benchmark against a model with no instance is rejected
    r = client.post("/v2/benchmarks", json={
        "name": "b2", "model_name": "missing", "duration_s": 1})
    assert r.status_code == 404


@pytest.mark.timeout(180)
def test_crash_restart(cluster):
    """Failure detection: killed engine -> ERROR -> automatic restart with
    backoff -> RUNNING again (reference: serve_manager.py:1842-1885)."""
    import os
    import signal

    from gpustack_amd.worker import serve_manager as sm_mod

    client, agent = cluster
    mid = client.get("/v2/models").json()["items"][0]["id"]
    client.patch(f"/v2/models/{mid}", json={"replicas": 1})
    for _ in range(240):
        insts = client.get("/v2/model_instances").json()["items"]
        if insts and insts[0]["state"] == "running":
            break
        time.sleep(0.5)
    assert insts[0]["state"] == "running"

    # shrink backoff so the test is fast
    old_base = sm_mod.RESTART_BASE
    sm_mod.RESTART_BASE = 0.5
    try:
        iid = insts[0]["id"]
        ip = agent.serve_manager.processes[iid]
        os.killpg(ip.proc.pid, signal.SIGKILL)
        saw_error = False
        state = None
        for _ in range(240):
            inst = [i for i in client.get("/v2/model_instances").json()["items"]
                    if i["id"] == iid][0]
            state = inst["state"]
            if state == "error":
                saw_error = True
            if saw_error and state == "running":
                break
            time.sleep(0.5)
        assert saw_error, "crash never surfaced as ERROR"
        assert state == "running", f"never restarted (state={state})"
        assert inst["restart_count"] >= 1
        # still serves
        r = client.post("/v1/chat/completions", json={
            "model": "tiny-chat", "messages": [{"role": "user", "content": "x"}],
            "max_tokens": 2, "ignore_eos": True})
        assert r.status_code == 200
    finally:
        sm_mod.RESTART_BASE = old_base


@pytest.mark.timeout(240)
def test_deploy_with_lora_adapters(cluster, tmp_path_factory):
    """Full dynamic-LoRA path: Model.lora_adapters -> serve manager backend
    params -> engine server mounts the adapter -> gateway resolves the
    adapter NAME as a servable model and routes it to the parent."""
    import sys
    sys.path.insert(0, "tests")
    from test_lora_dynamic import _make_adapter

    from gpustack_amd.engine import EngineConfig

    client, agent = cluster
    tmp = tmp_path_factory.mktemp("adapter")
    _make_adapter(tmp, EngineConfig(model="tiny").spec)

    r = client.post("/v2/models", json={
        "name": "tiny-lora-host", "source": "preset", "model_ref": "tiny",
        "replicas": 1, "max_model_len": 256,
        "lora_adapters": [{"name": "tuned-x", "path": str(tmp)}],
    })
    assert r.status_code == 201, r.text
    state = None
    for _ in range(240):
        insts = [i for i in client.get("/v2/model_instances").json()["items"]
                 if i["model_name"] == "tiny-lora-host"]
        if insts:
            state = insts[0]["state"]
            if state == "running":
                break
            assert state != "error", insts[0]["state_message"]
        time.sleep(0.5)
    assert state == "running", f"instance never ran (last state: {state})"

    # adapter name is listed and routable through the gateway
    ids = {m["id"] for m in client.get("/v1/models").json()["data"]}
    assert "tuned-x" in ids
    body = {"prompt": "hello", "max_tokens": 8, "ignore_eos": True,
            "temperature": 0}
    base_text = client.post("/v1/completions", json={
        **body, "model": "tiny-lora-host"}).json()["choices"][0]["text"]
    r = client.post("/v1/completions", json={**body, "model": "tuned-x"})
    assert r.status_code == 200, r.text
    lora_text = r.json()["choices"][0]["text"]
    assert lora_text != base_text  # adapter rows actually applied
    # parent stays clean
    again = client.post("/v1/completions", json={
        **body, "model": "tiny-lora-host"}).json()["choices"][0]["text"]
    assert again == base_text
    mid = next(m["id"] for m in client.get("/v2/models").json()["items"]
               if m["name"] == "tiny-lora-host")
    client.delete(f"/v2/models/{mid}")


def test_deploy_w4_runtime_model(cluster):
    """Full control-plane path for a W4-runtime model: backend_parameters
    flow scheduler -> serve manager -> engine_server -> EngineConfig, the
    engine packs its weights int4 and serves through the gateway."""
    client, agent = cluster
    r = client.post("/v2/models", json={
        "name": "tiny-w4", "source": "preset", "model_ref": "tiny",
        "replicas": 1, "max_model_len": 256,
        "backend_parameters": {"quantize_runtime": "w4"},
    })
    assert r.status_code == 201, r.text
    state = None
    for _ in range(240):
        insts = [i for i in client.get("/v2/model_instances").json()["items"]
                 if i["model_name"] == "tiny-w4"]
        if insts:
            state = insts[0]["state"]
            if state == "running":
                break
            assert state != "error", insts[0]["state_message"]
        time.sleep(0.5)
    assert state == "running", f"instance never ran (last state: {state})"
    r = client.post("/v1/completions", json={
        "model": "tiny-w4", "prompt": "xyz", "max_tokens": 5,
        "ignore_eos": True,
    })
    assert r.status_code == 200, r.text
    assert r.json()["usage"]["completion_tokens"] == 5
    client.delete(f"/v2/models/{[m for m in client.get('/v2/models').json()['items'] if m['name'] == 'tiny-w4'][0]['id']}")


def test_deploy_moe_model(cluster):
    """MoE family end-to-end through the cluster (router + expert bank
    in the engine subprocess)."""
    client, agent = cluster
    r = client.post("/v2/models", json={
        "name": "tiny-moe-e2e", "source": "preset", "model_ref": "tiny-moe",
        "replicas": 1, "max_model_len": 256,
    })
    assert r.status_code == 201, r.text
    state = None
    for _ in range(240):
        insts = [i for i in client.get("/v2/model_instances").json()["items"]
                 if i["model_name"] == "tiny-moe-e2e"]
        if insts:
            state = insts[0]["state"]
            if state == "running":
                break
            assert state != "error", insts[0]["state_message"]
        time.sleep(0.5)
    assert state == "running", f"instance never ran (last state: {state})"
    r = client.post("/v1/chat/completions", json={
        "model": "tiny-moe-e2e",
        "messages": [{"role": "user", "content": "hi"}],
        "max_tokens": 4, "ignore_eos": True,
    })
    assert r.status_code == 200, r.text
    client.delete(f"/v2/models/{[m for m in client.get('/v2/models').json()['items'] if m['name'] == 'tiny-moe-e2e'][0]['id']}")


@pytest.mark.timeout(240)
def test_deploy_reranker_model(cluster, tmp_path_factory):
    """Cross-encoder reranker checkpoint end-to-end: the model is
    auto-categorized 'reranker' at create, the scheduler sizes it from
    the encoder config, serve_manager launches engine_server which
    detects the sequence-classification architecture and serves the
    pair-scoring /v1/rerank through the gateway."""
    import json as _json

    import torch
    from safetensors.torch import save_file

    from gpustack_amd.models.encoder import CrossEncoderModel, EncoderSpec

    client, agent = cluster
    d = tmp_path_factory.mktemp("reranker")
    spec = EncoderSpec(
        architecture="BertForSequenceClassification", vocab_size=256,
        hidden_size=64, intermediate_size=128, num_layers=2, num_heads=4,
        max_position_embeddings=96, type_vocab_size=2, num_labels=1,
        pad_token_id=0)
    (d / "config.json").write_text(_json.dumps({
        "architectures": [spec.architecture],
        "vocab_size": spec.vocab_size, "hidden_size": spec.hidden_size,
        "num_hidden_layers": spec.num_layers,
        "num_attention_heads": spec.num_heads,
        "intermediate_size": spec.intermediate_size,
        "max_position_embeddings": spec.max_position_embeddings,
        "type_vocab_size": spec.type_vocab_size,
        "id2label": {"0": "LABEL_0"}, "pad_token_id": 0,
    }))
    from transformers import BertConfig, BertForSequenceClassification

    torch.manual_seed(11)
    hf = BertForSequenceClassification(BertConfig(
        vocab_size=spec.vocab_size, hidden_size=spec.hidden_size,
        num_hidden_layers=spec.num_layers,
        num_attention_heads=spec.num_heads,
        intermediate_size=spec.intermediate_size,
        max_position_embeddings=spec.max_position_embeddings,
        type_vocab_size=spec.type_vocab_size, num_labels=1,
        pad_token_id=0)).eval()
    save_file({k: v.contiguous() for k, v in hf.state_dict().items()
               if "position_ids" not in k}, str(d / "model.safetensors"))

    r = client.post("/v2/models", json={
        "name": "bge-e2e", "source": "local", "model_ref": str(d),
        "replicas": 1,
    })
    assert r.status_code == 201, r.text
    assert r.json()["categories"] == ["reranker"]
    state = None
    for _ in range(240):
        insts = [i for i in client.get("/v2/model_instances").json()["items"]
                 if i["model_name"] == "bge-e2e"]
        if insts:
            state = insts[0]["state"]
            if state == "running":
                break
            assert state != "error", insts[0]["state_message"]
        time.sleep(0.5)
    assert state == "running", f"instance never ran (last state: {state})"
    r = client.post("/v1/rerank", json={
        "model": "bge-e2e", "query": "gpu kernels",
        "documents": ["mfma tiles", "pasta recipe", "hip streams"],
        "top_n": 2,
    })
    assert r.status_code == 200, r.text
    res = r.json()["results"]
    assert len(res) == 2
    assert res[0]["relevance_score"] >= res[1]["relevance_score"]
    client.delete(f"/v2/models/{[m for m in client.get('/v2/models').json()['items'] if m['name'] == 'bge-e2e'][0]['id']}")


@pytest.mark.timeout(240)
def test_deploy_gemma_model(cluster):
    """Gemma-2-class model (sandwich norms, softcapping, GeGLU) served
    end-to-end through the cluster on the CPU oracle path."""
    client, agent = cluster
    r = client.post("/v2/models", json={
        "name": "tiny-g2-e2e", "source": "preset", "model_ref": "tiny-gemma",
        "replicas": 1, "max_model_len": 256,
    })
    assert r.status_code == 201, r.text
    state = None
    for _ in range(240):
        insts = [i for i in client.get("/v2/model_instances").json()["items"]
                 if i["model_name"] == "tiny-g2-e2e"]
        if insts:
            state = insts[0]["state"]
            if state == "running":
                break
            assert state != "error", insts[0]["state_message"]
        time.sleep(0.5)
    assert state == "running", f"instance never ran (last state: {state})"
    r = client.post("/v1/completions", json={
        "model": "tiny-g2-e2e", "prompt": "abc", "max_tokens": 5,
        "ignore_eos": True,
    })
    assert r.status_code == 200, r.text
    assert r.json()["usage"]["completion_tokens"] == 5
    client.delete(f"/v2/models/{[m for m in client.get('/v2/models').json()['items'] if m['name'] == 'tiny-g2-e2e'][0]['id']}")


@pytest.mark.timeout(120)
def test_instance_logs_via_server(cluster):
    """Instance logs proxied through the server to the worker's log API
    (reference: model-instance logs route)."""
    client, agent = cluster
    r = client.post("/v2/models", json={
        "name": "tiny-logs", "source": "preset", "model_ref": "tiny",
        "replicas": 1, "max_model_len": 256,
    })
    assert r.status_code == 201, r.text
    inst = None
    for _ in range(240):
        insts = [i for i in client.get("/v2/model_instances").json()["items"]
                 if i["model_name"] == "tiny-logs"]
        if insts and insts[0]["state"] == "running":
            inst = insts[0]
            break
        time.sleep(0.5)
    assert inst is not None
    r = client.get(f"/v2/model_instances/{inst['id']}/logs",
                   params={"tail": 50})
    assert r.status_code == 200, r.text
    assert "engine" in r.text or len(r.text) > 0  # some log content
    assert client.get("/v2/model_instances/999999/logs").status_code == 404
    client.delete(f"/v2/models/{[m for m in client.get('/v2/models').json()['items'] if m['name'] == 'tiny-logs'][0]['id']}")


@pytest.mark.timeout(180)
def test_instance_restart_action(cluster):
    """POST /restart tears the instance down and the controller brings a
    fresh one back to RUNNING."""
    client, agent = cluster
    r = client.post("/v2/models", json={
        "name": "tiny-restart", "source": "preset", "model_ref": "tiny",
        "replicas": 1, "max_model_len": 256,
    })
    assert r.status_code == 201
    inst = None
    for _ in range(240):
        insts = [i for i in client.get("/v2/model_instances").json()["items"]
                 if i["model_name"] == "tiny-restart"]
        if insts and insts[0]["state"] == "running":
            inst = insts[0]
            break
        time.sleep(0.5)
    assert inst is not None
    r = client.post(f"/v2/model_instances/{inst['id']}/restart")
    assert r.status_code == 200, r.text
    # the instance is torn down (row gone or non-running) and a fresh
    # process comes back to RUNNING (SQLite may reuse the row id)
    saw_down = False
    recovered = None
    for _ in range(240):
        insts = [i for i in client.get("/v2/model_instances").json()["items"]
                 if i["model_name"] == "tiny-restart"]
        if not insts or insts[0]["state"] != "running":
            saw_down = True
        elif saw_down and insts[0]["state"] == "running":
            recovered = insts[0]
            break
        time.sleep(0.25)
    assert saw_down, "restart never tore the instance down"
    assert recovered is not None, "replacement instance never ran"
    assert recovered.get("pid") != inst.get("pid")  # fresh process
    r = client.post("/v1/completions", json={
        "model": "tiny-restart", "prompt": "x", "max_tokens": 3,
        "ignore_eos": True})
    assert r.status_code == 200
    client.delete(f"/v2/models/{[m for m in client.get('/v2/models').json()['items'] if m['name'] == 'tiny-restart'][0]['id']}")


@pytest.mark.timeout(120)
def test_worker_log_follow_stream(cluster):
    """Worker /logs?follow=true streams the tail then appended lines
    until timeout_s (reference: log_sources follow)."""
    client, agent = cluster
    r = client.post("/v2/models", json={
        "name": "tiny-follow", "source": "preset", "model_ref": "tiny",
        "replicas": 1, "max_model_len": 256,
    })
    assert r.status_code == 201
    inst = None
    for _ in range(240):
        insts = [i for i in client.get("/v2/model_instances").json()["items"]
                 if i["model_name"] == "tiny-follow"]
        if insts and insts[0]["state"] == "running":
            inst = insts[0]
            break
        time.sleep(0.5)
    assert inst is not None
    wport = agent.cfg.worker_port
    t0 = time.time()
    r = httpx.get(f"http://127.0.0.1:{wport}/logs/{inst['name']}",
                  params={"follow": "true", "timeout_s": 1.5, "tail": 20},
                  timeout=30)
    took = time.time() - t0
    assert r.status_code == 200
    assert len(r.text) > 0          # tail delivered
    assert 1.0 < took < 10.0        # held open until timeout_s
    client.delete(f"/v2/models/{[m for m in client.get('/v2/models').json()['items'] if m['name'] == 'tiny-follow'][0]['id']}")
