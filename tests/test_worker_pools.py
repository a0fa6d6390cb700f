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
