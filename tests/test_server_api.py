"""Server API tests: auth, CRUD, worker registration, controllers,
scheduler integration, watch streams, exporter."""
import json
import tempfile
import threading
import time

import pytest
from fastapi.testclient import TestClient

from gpustack_amd.config import Config
from gpustack_amd.server.app import create_app


@pytest.fixture()
def server():
    cfg = Config(data_dir=tempfile.mkdtemp(), bootstrap_password="pw123")
    app = create_app(cfg, start_background=False)
    client = TestClient(app)
    r = client.post("/auth/login", json={"username": "admin", "password": "pw123"})
    assert r.status_code == 200
    token = r.json()["token"]
    client.headers["Authorization"] = f"Bearer {token}"
    reg_token = app.state.bootstrap["registration_token"]
    return client, app, cfg, reg_token


def _register_worker(client, reg_token, name="w1", n_gpus=8):
    from fixtures.workers.fixtures import mi355x_8g

    payload = mi355x_8g(1)
    payload["name"] = name
    return client.post(
        "/v2/workers/register", json={
            "name": name, "ip": "127.0.0.1", "port": 10150,
            "status": payload["status"],
            "system_reserved": payload["system_reserved"],
        },
        headers={"Authorization": f"Bearer {reg_token}"},
    )


def test_auth_rejects_bad_credentials(server):
    client, app, cfg, _ = server
    c = TestClient(app)
    assert c.post("/auth/login", json={"username": "admin", "password": "no"}).status_code == 401
    assert c.get("/v2/models").status_code == 401


def test_api_key_roundtrip(server):
    client, app, cfg, _ = server
    r = client.post("/v2/api_keys", json={"name": "k1"})
    assert r.status_code == 201
    key = r.json()["value"]
    assert key.startswith("gsa_")
    c = TestClient(app)
    c.headers["Authorization"] = f"Bearer {key}"
    assert c.get("/v2/models").status_code == 200
    # wrong secret fails
    c.headers["Authorization"] = f"Bearer {key[:-4]}beef"
    assert c.get("/v2/models").status_code == 401


def test_user_crud_admin_only(server):
    client, app, cfg, _ = server
    r = client.post("/v2/users", json={"username": "bob", "password": "pw", "is_admin": False})
    assert r.status_code == 201
    c = TestClient(app)
    r2 = c.post("/auth/login", json={"username": "bob", "password": "pw"})
    c.headers["Authorization"] = f"Bearer {r2.json()['token']}"
    assert c.get("/v2/users").status_code == 403  # not admin
    assert c.get("/v2/models").status_code == 200


def test_worker_register_and_status(server):
    client, app, cfg, reg = server
    r = _register_worker(client, reg)
    assert r.status_code == 200
    wid = r.json()["id"]
    assert r.json()["state"] == "ready"
    r = client.post(f"/v2/workers/{wid}/status", json={"status": {"cpu": {}}},
                    headers={"Authorization": f"Bearer {reg}"})
    assert r.status_code == 200
    # bad token rejected
    r = client.post(f"/v2/workers/{wid}/heartbeat",
                    headers={"Authorization": "Bearer nope"})
    assert r.status_code == 401
    # status posts are batched (reference worker_status_buffer semantics):
    # the write lands after the buffer flushes, coalesced per worker
    from gpustack_amd.server.routes_v2 import _status_buffer

    _status_buffer.flush()
    workers = client.get("/v2/workers").json()["items"]
    assert len(workers) == 1
    assert workers[0]["status"].get("gpu_devices") is None  # status replaced
    # re-register restores devices
    _register_worker(client, reg)
    w = client.get("/v2/workers").json()["items"][0]
    assert len(w["status"]["gpu_devices"]) == 8
    assert w["status"]["gpu_devices"][0]["arch_family"] == "gfx950"


def test_model_controller_sync_replicas(server):
    client, app, cfg, reg = server
    from gpustack_amd.server.controllers import ModelController

    mc = ModelController(cfg)
    r = client.post("/v2/models", json={"name": "m1", "model_ref": "tiny", "replicas": 3})
    mid = r.json()["id"]
    mc.sync_replicas(mid)
    insts = client.get("/v2/model_instances").json()["items"]
    assert len(insts) == 3
    assert all(i["state"] == "pending" for i in insts)
    # scale down
    client.patch(f"/v2/models/{mid}", json={"replicas": 1})
    mc.sync_replicas(mid)
    assert len(client.get("/v2/model_instances").json()["items"]) == 1
    # scale up again
    client.patch(f"/v2/models/{mid}", json={"replicas": 2})
    mc.sync_replicas(mid)
    assert len(client.get("/v2/model_instances").json()["items"]) == 2


def test_scheduler_places_instance(server):
    client, app, cfg, reg = server
    from gpustack_amd.scheduler.scheduler import PlacementScheduler
    from gpustack_amd.server.controllers import ModelController

    _register_worker(client, reg)
    r = client.post("/v2/models", json={"name": "m2", "model_ref": "llama-3-8b", "replicas": 1})
    ModelController(cfg).sync_replicas(r.json()["id"])
    inst = client.get("/v2/model_instances").json()["items"][0]
    sched = PlacementScheduler(cfg)
    assert sched.schedule_one(inst["id"])
    inst = client.get("/v2/model_instances").json()["items"][0]
    assert inst["state"] == "scheduled"
    assert inst["worker_id"] is not None
    assert len(inst["gpu_indexes"]) == 1
    assert inst["computed_resource_claim"]["vram"]


def test_scheduler_analyzing_when_no_worker(server):
    client, app, cfg, reg = server
    from gpustack_amd.scheduler.scheduler import PlacementScheduler
    from gpustack_amd.server.controllers import ModelController

    r = client.post("/v2/models", json={"name": "m3", "model_ref": "llama-3-8b"})
    ModelController(cfg).sync_replicas(r.json()["id"])
    inst = client.get("/v2/model_instances").json()["items"][0]
    assert not PlacementScheduler(cfg).schedule_one(inst["id"])
    inst = client.get("/v2/model_instances").json()["items"][0]
    assert inst["state"] == "analyzing"
    assert "no worker" in inst["state_message"]


def test_worker_monitor_marks_unreachable(server):
    client, app, cfg, reg = server
    from gpustack_amd.db import get_session
    from gpustack_amd.schemas import ModelInstance, Worker
    from gpustack_amd.server.controllers import WorkerMonitor

    _register_worker(client, reg)
    mid = client.post("/v2/models", json={"name": "x", "model_ref": "tiny"}).json()["id"]
    with get_session() as s:
        w = s.query(Worker).first()
        w.heartbeat_time = time.time() - 1000
        s.add(ModelInstance(model_id=mid, model_name="x", name="x-0",
                            worker_id=w.id, state="running"))
        s.commit()
        wid = w.id
    WorkerMonitor(cfg).check_once()
    w = client.get("/v2/workers").json()["items"][0]
    assert w["state"] == "unreachable"
    inst = client.get("/v2/model_instances").json()["items"][0]
    assert inst["state"] == "unreachable"


def test_watch_stream_replays_and_relays(server):
    # (TestClient buffers streaming bodies, so exercise the NDJSON
    # generator directly; the e2e cluster test covers it over real HTTP)
    client, app, cfg, reg = server
    client.post("/v2/models", json={"name": "mw", "model_ref": "tiny"})
    from gpustack_amd.server.routes_v2 import watch_ndjson

    gen = watch_ndjson("models", [{"id": 1, "name": "mw"}], None)
    frames = []

    def reader():
        for line in gen:
            frames.append(json.loads(line))
            if len(frames) >= 2:
                break

    t = threading.Thread(target=reader, daemon=True)
    t.start()
    time.sleep(0.2)
    client.post("/v2/models", json={"name": "mw2", "model_ref": "tiny"})
    t.join(timeout=10)
    assert len(frames) >= 2
    assert frames[0]["type"] == "CREATED" and frames[0]["data"]["name"] == "mw"
    assert frames[1]["type"] == "CREATED" and frames[1]["data"]["name"] == "mw2"


def test_openai_models_and_missing_model(server):
    client, app, cfg, reg = server
    client.post("/v2/models", json={"name": "chat-model", "model_ref": "tiny"})
    data = client.get("/v1/models").json()["data"]
    assert any(m["id"] == "chat-model" for m in data)
    r = client.post("/v1/chat/completions", json={
        "model": "nope", "messages": [{"role": "user", "content": "hi"}]})
    assert r.status_code == 404
    r = client.post("/v1/chat/completions", json={
        "model": "chat-model", "messages": [{"role": "user", "content": "hi"}]})
    assert r.status_code == 503  # no running instances


def test_metrics_exporter(server):
    client, app, cfg, reg = server
    _register_worker(client, reg)
    client.post("/v2/models", json={"name": "mx", "model_ref": "tiny", "replicas": 2})
    text = client.get("/metrics").text
    assert "gpustack_workers" in text
    assert "gpustack_worker_gpu_vram_total_bytes" in text
    assert 'gpustack_model_desired_replicas{model="mx"} 2.0' in text


def test_model_route_resolution(server):
    client, app, cfg, reg = server
    client.post("/v2/models", json={"name": "backend-a", "model_ref": "tiny"})
    r = client.post("/v2/model_routes", json={
        "name": "public-name",
        "targets": [{"model_name": "backend-a", "weight": 1}]})
    assert r.status_code == 201
    from gpustack_amd.server.routes_openai import _resolve_model_name

    assert _resolve_model_name("public-name") == ("backend-a", None)
    assert _resolve_model_name("backend-a") == ("backend-a", None)


def test_lora_adapter_route_resolution(server):
    """Per-LoRA child routes: an adapter name from Model.lora_adapters
    resolves to the parent model with the adapter name attached."""
    client, app, cfg, reg = server
    r = client.post("/v2/models", json={
        "name": "base-m", "model_ref": "tiny",
        "lora_adapters": [{"name": "sql-tuned", "path": "/adapters/sql"}]})
    assert r.status_code == 201
    from gpustack_amd.server.routes_openai import _resolve_model_name

    assert _resolve_model_name("sql-tuned") == ("base-m", "sql-tuned")
    assert _resolve_model_name("base-m") == ("base-m", None)
    # adapters appear in /v1/models with their parent
    items = {m["id"]: m for m in client.get("/v1/models").json()["data"]}
    assert items["sql-tuned"]["parent"] == "base-m"


def test_model_provider_routing(server):
    """External provider fallback: unknown local model routed to a
    registered OpenAI-compatible provider (stub) with its API key."""
    import threading
    import socket
    import time as _t

    import uvicorn
    from fastapi import FastAPI as _F, Request as _R

    client, app, cfg, reg = server
    stub = _F()
    seen = {}

    @stub.post("/v1/chat/completions")
    async def chat(request: _R):
        seen["auth"] = request.headers.get("authorization")
        body = await request.json()
        return {"id": "x", "object": "chat.completion",
                "choices": [{"index": 0, "message": {"role": "assistant",
                                                     "content": "from-provider"},
                             "finish_reason": "stop"}],
                "usage": {"prompt_tokens": 3, "completion_tokens": 2,
                          "total_tokens": 5}}

    s = socket.socket(); s.bind(("127.0.0.1", 0)); port = s.getsockname()[1]; s.close()
    server_u = uvicorn.Server(uvicorn.Config(stub, host="127.0.0.1", port=port,
                                             log_level="warning"))
    threading.Thread(target=server_u.run, daemon=True).start()
    import httpx as _h
    for _ in range(100):
        try:
            _h.post(f"http://127.0.0.1:{port}/v1/chat/completions", json={}, timeout=1)
            break
        except _h.HTTPError:
            _t.sleep(0.1)

    r = client.post("/v2/model_providers", json={
        "name": "ext", "base_url": f"http://127.0.0.1:{port}",
        "api_key": "sk-secret", "models": ["gpt-x"]})
    assert r.status_code == 201
    assert r.json()["api_key"] == "***"  # never echoed

    r = client.post("/v1/chat/completions", json={
        "model": "gpt-x", "messages": [{"role": "user", "content": "hi"}]})
    assert r.status_code == 200, r.text
    assert r.json()["choices"][0]["message"]["content"] == "from-provider"
    assert seen["auth"] == "Bearer sk-secret"
    # listed in /v1/models
    assert any(m["id"] == "gpt-x" for m in client.get("/v1/models").json()["data"])
    # usage metered under provider/model name
    usage = client.get("/v2/usage").json()["items"]
    assert any(u["model_name"] == "ext/gpt-x" and u["completion_tokens"] == 2
               for u in usage)
    server_u.should_exit = True


def test_catalog_and_version(server):
    client, app, cfg, reg = server
    cat = client.get("/v2/catalog").json()
    assert any(m["model_ref"] == "llama-3-8b" for m in cat["models"])
    assert "version" in client.get("/v2/version").json()


def test_model_update_redeploys_instances(server):
    from gpustack_amd.db import get_session
    from gpustack_amd.schemas import ModelInstance
    from gpustack_amd.server.controllers import ModelController

    client, app, cfg, reg = server
    r = client.post("/v2/models", json={"name": "redeploy-m",
                                        "model_ref": "tiny", "replicas": 2})
    assert r.status_code == 201
    mid = r.json()["id"]
    from gpustack_amd.db import EventType, bus

    mc = ModelController(cfg)
    mc.sync_replicas(mid)
    with get_session() as s:
        before = {i.name: i.spec_hash for i in
                  s.query(ModelInstance).filter_by(model_id=mid).all()}
    assert len(before) == 2 and all(before.values())
    # serving-relevant update -> instances torn down and recreated
    # (DELETED + CREATED events drive the worker restart; SQLite may reuse
    # row ids, so assert on the event stream + spec_hash)
    q = bus.subscribe("model_instances")
    client.patch(f"/v2/models/{mid}", json={"max_model_len": 2048})
    mc.sync_replicas(mid)
    events = []
    import queue as _q

    try:
        while True:
            events.append(q.get_nowait())
    except _q.Empty:
        pass
    bus.unsubscribe("model_instances", q)
    assert sum(1 for e in events if e.type == EventType.DELETED) == 2
    assert sum(1 for e in events if e.type == EventType.CREATED) == 2
    with get_session() as s:
        after = s.query(ModelInstance).filter_by(model_id=mid).all()
    assert len(after) == 2
    old_hash = list(before.values())[0]
    assert all(i.spec_hash and i.spec_hash != old_hash for i in after)
    # replicas-only change does NOT replace
    q = bus.subscribe("model_instances")
    client.patch(f"/v2/models/{mid}", json={"replicas": 2})
    mc.sync_replicas(mid)
    try:
        while True:
            assert q.get_nowait().type not in (EventType.DELETED,
                                               EventType.CREATED)
    except _q.Empty:
        pass
    bus.unsubscribe("model_instances", q)


def test_multi_cluster_registration_and_placement(server):
    """Clusters: token-scoped worker registration + scheduler cluster
    filter (reference: schemas/clusters.py per-cluster tokens)."""
    client, app, cfg, reg = server
    # default cluster exists and has the bootstrap token
    clusters = client.get("/v2/clusters").json()["items"]
    assert any(c["is_default"] for c in clusters)
    default_id = next(c["id"] for c in clusters if c["is_default"])

    r = client.post("/v2/clusters", json={"name": "edge"})
    assert r.status_code == 201
    edge = r.json()
    edge_token = edge["registration_token"]
    assert edge_token and edge["id"] != default_id

    # one worker per cluster: default token vs edge token
    _register_worker(client, reg, name="w-default")
    _register_worker(client, edge_token, name="w-edge")
    ws = {w["name"]: w for w in client.get("/v2/workers").json()["items"]}
    assert ws["w-default"]["cluster_id"] == default_id
    assert ws["w-edge"]["cluster_id"] == edge["id"]

    # placement respects the model's cluster pin
    from gpustack_amd.scheduler.policies import cluster_filter

    workers = list(ws.values())
    got = cluster_filter(workers, {"cluster_id": edge["id"]})
    assert [w["name"] for w in got] == ["w-edge"]
    assert len(cluster_filter(workers, {"cluster_id": None})) == 2

    # default cluster cannot be deleted; empty one can
    rid = next(c["id"] for c in clusters if c["is_default"])
    assert client.delete(f"/v2/clusters/{rid}").status_code == 400
    assert client.delete(f"/v2/clusters/{edge['id']}").status_code == 409  # has workers
    wid = ws["w-edge"]["id"]
    client.delete(f"/v2/workers/{wid}")
    assert client.delete(f"/v2/clusters/{edge['id']}").status_code == 200


def test_v1_openai_legacy_mount(server):
    """Frozen legacy subset at /v1-openai (reference: routes/openai.py:81-92)."""
    client, app, cfg, reg = server
    r = client.get("/v1-openai/models")
    assert r.status_code == 200
    assert r.json()["object"] == "list"
    # proxy path resolves models the same way (404 for unknown)
    r = client.post("/v1-openai/chat/completions", json={
        "model": "nope", "messages": [{"role": "user", "content": "hi"}]})
    assert r.status_code == 404


def test_org_tenancy(server):
    """Org-scoped model visibility (reference: UserService
    .model_allowed_for_user, api/tenant.py org gates)."""
    client, app, cfg, reg = server
    org = client.post("/v2/orgs", json={"name": "team-a"}).json()
    client.post("/v2/models", json={"name": "pub-model", "model_ref": "tiny"})
    client.post("/v2/models", json={"name": "team-model", "model_ref": "tiny",
                                    "org_id": org["id"]})
    client.post("/v2/users", json={"username": "ann", "password": "pw",
                                   "org_id": org["id"]})
    client.post("/v2/users", json={"username": "bob", "password": "pw"})

    def as_user(name):
        c = TestClient(app)
        r = c.post("/auth/login", json={"username": name, "password": "pw"})
        c.headers["Authorization"] = f"Bearer {r.json()['token']}"
        return c

    ann, bob = as_user("ann"), as_user("bob")
    assert {m["name"] for m in ann.get("/v2/models").json()["items"]} == \
        {"pub-model", "team-model"}
    assert {m["name"] for m in bob.get("/v2/models").json()["items"]} == \
        {"pub-model"}
    # /v1/models mirrors the same visibility
    assert "team-model" not in {m["id"] for m in
                                bob.get("/v1/models").json()["data"]}
    assert "team-model" in {m["id"] for m in
                            ann.get("/v1/models").json()["data"]}
    # inference path: out-of-org model answers 404 (name disclosure safe)
    r = bob.post("/v1/chat/completions", json={
        "model": "team-model", "messages": [{"role": "user", "content": "x"}]})
    assert r.status_code == 404
    # in-org user reaches placement (503: no running instance yet)
    r = ann.post("/v1/chat/completions", json={
        "model": "team-model", "messages": [{"role": "user", "content": "x"}]})
    assert r.status_code == 503
    # admin sees everything; org with members cannot be deleted
    assert client.delete(f"/v2/orgs/{org['id']}").status_code == 409


def test_unsupported_modalities_return_501(server):
    """Audio/image/moderations: wire-compatible endpoints exist and fail
    with a structured 501 (formal descope, PARITY.md)."""
    client, app, cfg, reg = server
    for path in ("/v1/audio/transcriptions", "/v1/images/generations",
                 "/v1/moderations"):
        r = client.post(path, json={"model": "x"})
        assert r.status_code == 501, path
        assert r.json()["detail"]["error"]["type"] == "unsupported_modality"


def test_resource_event_logger_records_transitions(server):
    """Metering pair (reference resource_events): state transitions append
    hot rows with the claim footprint; archiver drains old rows."""
    import time as _time

    from gpustack_amd.db import get_session
    from gpustack_amd.schemas import ModelInstance
    from gpustack_amd.schemas.tables import ResourceEvent, ResourceEventArchive
    from gpustack_amd.server.controllers import (ResourceEventLogger,
                                                 UsageArchiver)

    client, app, cfg, reg = server
    lg = ResourceEventLogger(cfg)
    mid = client.post("/v2/models", json={"name": "re", "model_ref": "tiny"}).json()["id"]
    data = {"id": 991, "state": "scheduled", "model_id": mid,
            "model_name": "re", "worker_id": 1, "gpu_indexes": [0],
            "computed_resource_claim": {"vram": {"0": 1 << 30}, "ram": 2 << 30}}
    lg._record(data)
    lg._record(data)  # same state: deduped
    data2 = dict(data, state="running")
    lg._record(data2)
    with get_session() as s:
        rows = s.query(ResourceEvent).all()
        assert [r.event_type for r in rows] == ["scheduled", "running"]
        assert rows[0].vram_bytes == 1 << 30 and rows[0].ram_bytes == 2 << 30
        # age a row and archive it
        rows[0].timestamp = _time.time() - 90 * 86400
        s.commit()
    arch = UsageArchiver(cfg, keep_days=30)
    moved = arch.archive_once()
    assert moved >= 1
    with get_session() as s:
        assert s.query(ResourceEventArchive).count() == 1
        assert s.query(ResourceEvent).count() == 1


def test_registration_token_persisted_to_data_dir(server):
    """The server writes <data_dir>/token at startup (reference behavior;
    the all-in-one container's embedded worker joins by reading it)."""
    import pathlib

    _client, _app, cfg, reg_token = server
    p = pathlib.Path(cfg.data_dir) / "token"
    assert p.read_text().strip() == reg_token
    assert (p.stat().st_mode & 0o777) == 0o600


def test_list_pagination_and_filters(server):
    """Reference-style list params: page/perPage envelope, substring
    search, category filter — plain lists stay unchanged for existing
    clients."""
    client, app, cfg, _tok = server
    for i in range(5):
        r = client.post("/v2/models", json={
            "name": f"pag-{i}", "source": "preset", "model_ref": "tiny"})
        assert r.status_code == 201
    r = client.get("/v2/models")
    assert "pagination" not in r.json()  # no params: legacy shape
    r = client.get("/v2/models", params={"page": 2, "perPage": 2,
                                         "search": "pag-"})
    body = r.json()
    assert [m["name"] for m in body["items"]] == ["pag-2", "pag-3"]
    assert body["pagination"]["total"] == 5
    assert body["pagination"]["totalPage"] == 3
    r = client.get("/v2/models", params={"search": "pag-4"})
    assert [m["name"] for m in r.json()["items"]] == ["pag-4"]
    r = client.get("/v2/models", params={"categories": "reranker"})
    assert r.json()["items"] == []
    r = client.get("/v2/workers", params={"page": 1, "perPage": 10})
    assert "pagination" in r.json()


def test_dashboard(server):
    """Aggregate dashboard: resource counts, load, 7-day usage summary."""
    import datetime

    client, app, cfg, _tok = server
    from gpustack_amd.db import ar_create, get_session
    from gpustack_amd.schemas import ModelUsage

    today = datetime.date.today().strftime("%Y-%m-%d")
    with get_session() as s:
        ar_create(s, ModelUsage(user_id=1, model_id=1, model_name="m1",
                                date=today, prompt_tokens=100,
                                completion_tokens=50, request_count=3))
        ar_create(s, ModelUsage(user_id=1, model_id=2, model_name="m2",
                                date=today, prompt_tokens=10,
                                completion_tokens=5, request_count=1))
    r = client.get("/v2/dashboard")
    assert r.status_code == 200, r.text
    body = r.json()
    assert body["resource_counts"]["models"] >= 0
    assert body["model_usage"]["totals"]["prompt_tokens"] == 110
    assert body["model_usage"]["top_models"][0]["model_name"] == "m1"
    assert "system_load" in body


def test_gpu_devices_and_config(server):
    client, app, cfg, reg_token = server
    r = _register_worker(client, reg_token, name="gpuw")
    assert r.status_code in (200, 201), r.text
    r = client.get("/v2/gpu_devices")
    items = r.json()["items"]
    mine = [g for g in items if g["worker_name"] == "gpuw"]
    assert len(mine) >= 1
    assert mine[0]["allocatable_vram"] > 0
    assert mine[0]["id"] == "gpuw:0"
    assert mine[0]["memory"].get("total", 0) > 0
    r = client.get("/v2/gpu_devices", params={"search": "MI355",
                                              "page": 1, "perPage": 1})
    assert len(r.json()["items"]) == 1

    r = client.get("/v2/config")
    assert r.status_code == 200
    body = r.json()
    assert body.get("bootstrap_password") == "***"
    assert "data_dir" in body


def test_catalog_refs_resolve(server):
    """Every catalog entry's preset ref must exist and its categories
    match the architecture registry — no dead catalog rows."""
    client, app, cfg, _ = server
    from gpustack_amd.engine.config import PRESETS
    from gpustack_amd.utils.model_registry import categories_for_architecture

    body = client.get("/v2/catalog").json()
    items = body.get("items") or body.get("models") or []
    assert len(items) >= 15
    for it in items:
        assert it["source"] == "preset"
        spec = PRESETS.get(it["model_ref"])
        assert spec is not None, f"dead catalog ref {it['model_ref']}"
        derived = set(categories_for_architecture(spec.architecture))
        assert derived & set(it["categories"]), (it["name"], derived)
