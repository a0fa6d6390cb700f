"""SSH GPU instances (the gpustack-operator analog, server/gpu_instances.py):
CRUD + controller reconcile lifecycle on the mock provider, the K8s pod
provider against a fake kube-apiserver (httpx MockTransport), and the pod
manifest's ROCm-device honesty (kfd/dri mounts, amd.com/gpu claim,
public-key injection)."""
import json
import tempfile

import httpx
import pytest
from starlette.testclient import TestClient

from gpustack_amd.config import Config
from gpustack_amd.server.app import create_app
from gpustack_amd.server.gpu_instances import (
    FLAVORS, GPUInstanceController, K8sPodProvider, MockInstanceProvider,
    instance_pod_manifest, instance_service_manifest,
)


@pytest.fixture()
def server():
    cfg = Config(data_dir=tempfile.mkdtemp(), bootstrap_password="pw123")
    app = create_app(cfg, start_background=False)
    client = TestClient(app)
    r = client.post("/auth/login", json={"username": "admin", "password": "pw123"})
    assert r.status_code == 200
    client.headers["Authorization"] = f"Bearer {r.json()['token']}"
    return client, cfg


def test_gpu_instance_lifecycle_mock(server):
    client, cfg = server
    MockInstanceProvider.instances.clear()
    r = client.post("/v2/gpu_instances", json={
        "name": "dev1", "flavor": "mi355x-2gpu", "provider": "mock",
        "ssh_public_key": "ssh-ed25519 AAAA test@host",
    })
    assert r.status_code == 201, r.text
    gid = r.json()["id"]
    assert r.json()["state"] == "pending"

    ctl = GPUInstanceController(cfg)
    ctl.reconcile(gid)  # pending -> creating (pod created)
    g = client.get(f"/v2/gpu_instances/{gid}").json()
    assert g["state"] == "creating" and g["external_id"].startswith("mock-")
    assert len(MockInstanceProvider.instances) == 1

    ctl.reconcile(gid)  # creating -> running (mock is up immediately)
    g = client.get(f"/v2/gpu_instances/{gid}").json()
    assert g["state"] == "running"
    assert g["ssh_host"] == "mock.local" and g["ssh_port"] == 2200

    r = client.delete(f"/v2/gpu_instances/{gid}")
    assert r.json() == {"status": "deleting"}
    ctl.reconcile(gid)  # deleting -> deprovisioned + row removed
    assert client.get(f"/v2/gpu_instances/{gid}").status_code == 404
    assert not MockInstanceProvider.instances


def test_gpu_instance_validation(server):
    client, _cfg = server
    assert client.post("/v2/gpu_instances", json={
        "name": "x", "provider": "nope"}).status_code == 400
    assert client.post("/v2/gpu_instances", json={
        "name": "x", "provider": "mock", "flavor": "h100"}).status_code == 400
    flavors = client.get("/v2/gpu_instance_flavors").json()["items"]
    assert {f["name"] for f in flavors} == set(FLAVORS)


def test_gpu_instance_provider_error_lands_in_error_state(server):
    client, cfg = server
    r = client.post("/v2/gpu_instances", json={
        "name": "dead", "provider": "k8s",
        "provider_config": {"api_server": "http://127.0.0.1:1",
                            "token": "t"}})
    gid = r.json()["id"]
    GPUInstanceController(cfg).reconcile(gid)
    g = client.get(f"/v2/gpu_instances/{gid}").json()
    assert g["state"] == "error" and g["state_message"]


def test_pod_manifest_rocm_honesty():
    inst = {"name": "dev1", "flavor": "mi355x-4gpu",
            "image": "rocm/dev-ubuntu-24.04",
            "ssh_public_key": "ssh-ed25519 KEY u@h",
            "volumes": [{"size_gb": 50, "mount_path": "/work"}],
            "labels": {"team": "ml"}}
    pod = instance_pod_manifest(inst, "ns1")
    c = pod["spec"]["containers"][0]
    assert pod["metadata"]["namespace"] == "ns1"
    assert c["resources"]["limits"]["amd.com/gpu"] == "4"
    # ROCm device nodes must be mounted for the GPU to be visible
    paths = {v.get("hostPath", {}).get("path") for v in pod["spec"]["volumes"]}
    assert {"/dev/kfd", "/dev/dri"} <= paths
    mounts = {m["mountPath"] for m in c["volumeMounts"]}
    assert {"/dev/kfd", "/dev/dri", "/work"} <= mounts
    # the public key reaches authorized_keys; sshd is the entrypoint
    assert "ssh-ed25519 KEY u@h" in c["command"][2]
    assert "sshd" in c["command"][2]
    assert pod["metadata"]["labels"]["team"] == "ml"
    svc = instance_service_manifest(inst, "ns1")
    assert svc["spec"]["selector"] == {"gpustack.amd/instance": "dev1"}


def test_k8s_pod_provider_against_fake_apiserver():
    """Full provider cycle against an httpx MockTransport playing a
    minimal kube-apiserver: create pod+service, poll status through
    Pending -> Running (hostIP + allocated nodePort), delete both."""
    state = {"pods": {}, "services": {}}

    def handler(request: httpx.Request) -> httpx.Response:
        path = request.url.path
        parts = path.strip("/").split("/")
        # parts: ["api", "v1", "namespaces", ns, kind, name?]
        kind = parts[4] if len(parts) > 4 else ""
        name = parts[5] if len(parts) > 5 else None
        store = state.get(kind, {})
        if request.method == "POST":
            obj = json.loads(request.content)
            obj.setdefault("status", {})
            if kind == "services":
                obj["spec"]["ports"][0]["nodePort"] = 30022
            store[obj["metadata"]["name"]] = obj
            return httpx.Response(201, json=obj)
        if request.method == "GET":
            if name in store:
                return httpx.Response(200, json=store[name])
            return httpx.Response(404, json={"reason": "NotFound"})
        if request.method == "DELETE":
            store.pop(name, None)
            return httpx.Response(200, json={})
        return httpx.Response(405)

    from gpustack_amd.utils.k8s_client import KubeClient

    kube = KubeClient(api_server="https://kube.test", token="tok",
                      namespace="gpustack",
                      transport=httpx.MockTransport(handler))
    prov = K8sPodProvider(client=kube)
    inst = {"name": "dev2", "flavor": "mi355x-1gpu",
            "image": "rocm/dev-ubuntu-24.04", "ssh_public_key": "k"}
    eid = prov.create(inst)
    assert eid == "gpi-dev2"
    assert "gpi-dev2" in state["pods"] and "gpi-dev2" in state["services"]

    st = prov.status(eid)
    assert st["phase"] == "creating"  # no phase yet -> Pending
    state["pods"][eid]["status"] = {"phase": "Running", "hostIP": "10.0.0.5"}
    st = prov.status(eid)
    assert st == {"phase": "running", "ssh_host": "10.0.0.5",
                  "ssh_port": 30022}

    state["pods"][eid]["status"] = {"phase": "Failed", "reason": "OOM"}
    assert prov.status(eid)["phase"] == "failed"

    prov.delete(eid)
    assert not state["pods"] and not state["services"]
    assert prov.status(eid)["phase"] == "gone"


def test_k8s_client_error_surface():
    def handler(request: httpx.Request) -> httpx.Response:
        return httpx.Response(403, text="forbidden: RBAC")

    from gpustack_amd.utils.k8s_client import KubeClient, KubeError

    kube = KubeClient(api_server="https://kube.test", token="t",
                      namespace="ns", transport=httpx.MockTransport(handler))
    with pytest.raises(KubeError, match="403"):
        kube.create_pod({"metadata": {"name": "p", "namespace": "ns"}})


def test_migration_creates_gpu_instances_table(tmp_path):
    """An older DB (pre-v11) gains the gpu_instances table on init."""
    import sqlite3

    db = tmp_path / "old.db"
    con = sqlite3.connect(db)
    con.execute("CREATE TABLE workers (id INTEGER PRIMARY KEY, name VARCHAR)")
    con.execute("CREATE TABLE models (id INTEGER PRIMARY KEY, name VARCHAR)")
    con.execute("CREATE TABLE model_instances (id INTEGER PRIMARY KEY)")
    con.commit()
    con.close()

    from gpustack_amd.db import get_engine, init_db
    from gpustack_amd.db.migrations import HEAD, current_version

    init_db(f"sqlite:///{db}")
    with get_engine().begin() as conn:
        assert current_version(conn) == HEAD >= 11
    con = sqlite3.connect(db)
    tables = [r[0] for r in con.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    assert "gpu_instances" in tables
    con.close()


def test_templates_and_ssh_keys(server):
    """Templates preset flavor/image/volumes; instances can start from a
    template and reference a stored SSH key (reference:
    gpu_instance_templates + gpu_instance_ssh_public_keys)."""
    client, cfg = server
    r = client.post("/v2/gpu_instance_templates", json={
        "name": "dev-4g", "flavor": "mi355x-4gpu",
        "image": "rocm/megatron-lm", "provider": "mock",
        "volumes": [{"size_gb": 100, "mount_path": "/work"}]})
    assert r.status_code == 201, r.text
    assert client.post("/v2/gpu_instance_templates", json={
        "name": "bad", "flavor": "h100"}).status_code == 400
    r = client.post("/v2/ssh_public_keys", json={
        "name": "laptop", "public_key": "ssh-ed25519 AAAAC3 u@h"})
    assert r.status_code == 201
    assert client.post("/v2/ssh_public_keys", json={
        "name": "junk", "public_key": "not-a-key"}).status_code == 400

    r = client.post("/v2/gpu_instances", json={
        "name": "from-tpl", "template": "dev-4g", "ssh_key_name": "laptop"})
    assert r.status_code == 201, r.text
    g = r.json()
    assert g["flavor"] == "mi355x-4gpu"
    assert g["image"] == "rocm/megatron-lm"
    assert g["provider"] == "mock"
    assert g["volumes"][0]["mount_path"] == "/work"
    assert g["ssh_public_key"] == "ssh-ed25519 AAAAC3 u@h"
    # caller overrides beat template fields
    r = client.post("/v2/gpu_instances", json={
        "name": "from-tpl-2", "template": "dev-4g",
        "flavor": "mi355x-1gpu", "provider": "mock"})
    assert r.json()["flavor"] == "mi355x-1gpu"
    # unknown refs are 400s
    assert client.post("/v2/gpu_instances", json={
        "name": "x1", "template": "nope"}).status_code == 400
    assert client.post("/v2/gpu_instances", json={
        "name": "x2", "provider": "mock",
        "ssh_key_name": "nope"}).status_code == 400
    # listing + delete
    assert len(client.get("/v2/gpu_instance_templates").json()["items"]) == 1
    kid = client.get("/v2/ssh_public_keys").json()["items"][0]["id"]
    assert client.delete(f"/v2/ssh_public_keys/{kid}").status_code == 200


def test_gpu_instance_watch_stream(server):
    """watch frames replay the snapshot then relay lifecycle events
    (generator-level, like the models watch test — TestClient buffers
    streaming bodies)."""
    import json as _json
    import threading
    import time as _time

    client, cfg = server
    client.post("/v2/gpu_instances", json={"name": "w1",
                                           "provider": "mock"})
    from gpustack_amd.server.routes_v2 import watch_ndjson

    gen = watch_ndjson("gpu_instances", [{"id": 1, "name": "w1"}], None)
    frames = []

    def reader():
        for line in gen:
            frames.append(_json.loads(line))
            if len(frames) >= 2:
                break

    t = threading.Thread(target=reader, daemon=True)
    t.start()
    _time.sleep(0.2)
    client.post("/v2/gpu_instances", json={"name": "w2",
                                           "provider": "mock"})
    t.join(timeout=10)
    assert len(frames) >= 2
    assert frames[0]["data"]["name"] == "w1"
    assert frames[1]["type"] == "CREATED"
    assert frames[1]["data"]["name"] == "w2"
