"""Worker pools + cloud provisioning (reference: gpustack/cloud_providers/,
WorkerPoolController / WorkerProvisioningController scale-out)."""
import tempfile

import pytest
from fastapi.testclient import TestClient

from gpustack_amd.config import Config
from gpustack_amd.server.app import create_app
from gpustack_amd.server.controllers import WorkerPoolController
from gpustack_amd.server.providers import (
    CommandProvider, MockProvider, bootstrap_script, get_provider,
)


@pytest.fixture()
def server():
    MockProvider.instances.clear()
    cfg = Config(data_dir=tempfile.mkdtemp(), bootstrap_password="pw123")
    app = create_app(cfg, start_background=False)
    client = TestClient(app)
    r = client.post("/auth/login", json={"username": "admin", "password": "pw123"})
    client.headers["Authorization"] = f"Bearer {r.json()['token']}"
    return client, app, cfg


def test_bootstrap_script_contents():
    s = bootstrap_script("http://10.0.0.1:8080", "tok123", {"pool": "a"})
    assert "--server-url http://10.0.0.1:8080" in s
    assert "--registration-token tok123" in s
    assert "--label 'pool=a'" in s or '--label pool=a' in s


def test_provider_registry():
    assert isinstance(get_provider("mock"), MockProvider)
    with pytest.raises(ValueError):
        get_provider("droplets")
    with pytest.raises(ValueError):
        get_provider("command", {})  # needs commands


def test_command_provider_roundtrip(tmp_path):
    log = tmp_path / "calls.log"
    p = CommandProvider({
        "create_command":
            f'echo "$GPUSTACK_INSTANCE_NAME $GPUSTACK_INSTANCE_TYPE" >> {log}; '
            'echo "node-$GPUSTACK_INSTANCE_NAME"',
        "delete_command": f'echo "del $GPUSTACK_INSTANCE_ID" >> {log}',
    })
    iid = p.create("w0", "mi355x-8gpu", "#!/bin/sh\n")
    assert iid == "node-w0"
    p.delete(iid)
    lines = log.read_text().splitlines()
    assert lines == ["w0 mi355x-8gpu", "del node-w0"]


def test_pool_scale_up_and_down(server):
    client, app, cfg = server
    r = client.post("/v2/worker_pools", json={
        "name": "p1", "provider": "mock", "replicas": 3,
        "labels": {"pool": "p1"}})
    assert r.status_code == 201
    pid = r.json()["id"]
    ctrl = WorkerPoolController(cfg)
    ctrl.reconcile_all()
    pool = client.get("/v2/worker_pools").json()["items"][0]
    assert len(pool["instances"]) == 3
    assert len(MockProvider.instances) == 3
    names = {r["name"] for r in pool["instances"]}
    assert names == {"p1-0", "p1-1", "p1-2"}
    # every instance got a bootstrap with the registration token
    ud = next(iter(MockProvider.instances.values()))["user_data"]
    assert "--server-url" in ud and "--registration-token" in ud

    # scale down to 1 removes the newest two
    r = client.put(f"/v2/worker_pools/{pid}", json={"replicas": 1})
    assert r.status_code == 200
    ctrl.reconcile_all()
    pool = client.get("/v2/worker_pools").json()["items"][0]
    assert len(pool["instances"]) == 1
    assert len(MockProvider.instances) == 1

    # delete deprovisions the rest
    assert client.delete(f"/v2/worker_pools/{pid}").status_code == 200
    assert len(MockProvider.instances) == 0


def test_pool_marks_registered_workers_ready(server):
    client, app, cfg = server
    client.post("/v2/worker_pools", json={
        "name": "p2", "provider": "mock", "replicas": 1})
    ctrl = WorkerPoolController(cfg)
    ctrl.reconcile_all()
    # simulate the provisioned node's worker registering under its name
    from fixtures.workers.fixtures import mi355x_8g

    reg = app.state.bootstrap["registration_token"]
    payload = mi355x_8g(1)
    r = client.post("/v2/workers/register", json={
        "name": "p2-0", "ip": "10.9.9.9", "port": 10150,
        "status": payload["status"],
        "system_reserved": payload["system_reserved"]},
        headers={"Authorization": f"Bearer {reg}"})
    assert r.status_code in (200, 201)
    ctrl.reconcile_all()
    pool = [p for p in client.get("/v2/worker_pools").json()["items"]
            if p["name"] == "p2"][0]
    assert pool["instances"][0]["state"] == "ready"


def test_unknown_provider_rejected(server):
    client, app, cfg = server
    r = client.post("/v2/worker_pools", json={
        "name": "bad", "provider": "droplets", "replicas": 1})
    assert r.status_code == 400


def test_k8s_pool_provider_creates_worker_pods():
    """K8sProvider drives worker pods through the kube client: the pod
    carries ROCm devices, the amd.com/gpu claim and the bootstrap script
    as its command."""
    import json

    import httpx

    from gpustack_amd.server.providers import K8sProvider, bootstrap_script
    from gpustack_amd.utils.k8s_client import KubeClient

    pods = {}

    def handler(request: httpx.Request) -> httpx.Response:
        parts = request.url.path.strip("/").split("/")
        name = parts[5] if len(parts) > 5 else None
        if request.method == "POST":
            obj = json.loads(request.content)
            pods[obj["metadata"]["name"]] = obj
            return httpx.Response(201, json=obj)
        if request.method == "DELETE":
            pods.pop(name, None)
            return httpx.Response(200, json={})
        return httpx.Response(404, json={})

    kube = KubeClient(api_server="https://kube.test", token="t",
                      namespace="gpustack",
                      transport=httpx.MockTransport(handler))
    prov = K8sProvider({"image": "gpustack-amd:v1", "gpus_per_worker": 4},
                       client=kube)
    ud = bootstrap_script("http://server", "tok_x", {"pool": "a"})
    iid = prov.create("pool-a-0", "mi355x-4gpu", ud)
    assert iid == "pool-a-0" and "pool-a-0" in pods
    c = pods["pool-a-0"]["spec"]["containers"][0]
    assert c["image"] == "gpustack-amd:v1"
    assert c["resources"]["limits"]["amd.com/gpu"] == "4"
    assert "--server-url http://server" in c["command"][2]
    paths = {v["hostPath"]["path"]
             for v in pods["pool-a-0"]["spec"]["volumes"]}
    assert paths == {"/dev/kfd", "/dev/dri"}
    prov.delete("pool-a-0")
    assert not pods
