"""Scaling schedules (cron windows), model evaluations, admin reset."""
import tempfile
import time

from fastapi.testclient import TestClient

from gpustack_amd.config import Config
from gpustack_amd.server.app import create_app
from gpustack_amd.utils.cron import cron_matches, window_active


def test_cron_matching():
    # 2026-09-12 is a Saturday
    t = time.mktime((2026, 9, 12, 9, 30, 0, 0, 0, -1))
    assert cron_matches("30 9 * * *", t)
    assert cron_matches("* * * * 6", t)          # Saturday
    assert not cron_matches("* * * * 1-5", t)    # weekdays only
    assert cron_matches("*/15 * * * *", t)
    assert not cron_matches("0 9 * * *", t)
    assert window_active("0 9 * * *", 60, t)     # 9:00 + 60min covers 9:30
    assert not window_active("0 8 * * *", 30, t)


def _server():
    cfg = Config(data_dir=tempfile.mkdtemp(), bootstrap_password="pw")
    app = create_app(cfg, start_background=False)
    c = TestClient(app)
    tok = c.post("/auth/login", json={"username": "admin", "password": "pw"}).json()["token"]
    c.headers["Authorization"] = f"Bearer {tok}"
    return c, app, cfg


def test_scaling_scheduler_desired_replicas():
    from gpustack_amd.db import get_session
    from gpustack_amd.schemas import Model, ModelInstance
    from gpustack_amd.server.controllers import ScalingScheduler

    c, app, cfg = _server()
    c.post("/v2/models", json={
        "name": "sched-m", "model_ref": "tiny", "replicas": 1,
        "scaling_schedule": {"rules": [{"cron": "* * * * *",
                                        "duration_minutes": 1, "replicas": 3}]},
    })
    ss = ScalingScheduler(cfg)
    with get_session() as s:
        m = s.query(Model).filter_by(name="sched-m").first()
        assert ss.desired_replicas(m) == 3
        # rule outside any window -> baseline
        m.scaling_schedule = {"rules": [{"cron": "0 0 31 2 *", "replicas": 9}]}
        assert ss.desired_replicas(m) == 1


def test_model_evaluation_endpoint():
    import sys
    from pathlib import Path

    sys.path.insert(0, str(Path(__file__).resolve().parent))
    from fixtures.workers.fixtures import mi355x_8g

    c, app, cfg = _server()
    reg = app.state.bootstrap["registration_token"]
    r = c.post("/v2/model-evaluations", json={"name": "e", "model_ref": "llama-3-8b"})
    assert r.status_code == 200 and not r.json()["compatible"]
    w = mi355x_8g(1)
    c.post("/v2/workers/register", json={
        "name": "w1", "ip": "10.0.0.1", "status": w["status"],
        "system_reserved": w["system_reserved"]},
        headers={"Authorization": f"Bearer {reg}"})
    r = c.post("/v2/model-evaluations", json={"name": "e", "model_ref": "llama-3-8b"})
    assert r.json()["compatible"]
    assert r.json()["candidate"]["worker"] == "w1"
    r = c.post("/v2/model-evaluations", json={"name": "e", "model_ref": "not-a-model"})
    assert not r.json()["compatible"]


def test_reset_admin_password_cli():
    import tempfile as tf

    from gpustack_amd.main import main

    d = tf.mkdtemp()
    # bootstrap a db first
    cfg = Config(data_dir=d, bootstrap_password="old")
    create_app(cfg, start_background=False)
    assert main(["reset-admin-password", "--data-dir", d, "--password", "newpw"]) == 0
    app2 = create_app(Config(data_dir=d), start_background=False)
    c = TestClient(app2)
    assert c.post("/auth/login", json={"username": "admin", "password": "newpw"}).status_code == 200
    assert c.post("/auth/login", json={"username": "admin", "password": "old"}).status_code == 401


def test_usage_archiver_moves_old_rows():
    from gpustack_amd.db import get_session
    from gpustack_amd.schemas import ModelUsage
    from gpustack_amd.schemas.tables import ModelUsageArchive
    from gpustack_amd.server.controllers import UsageArchiver

    c, app, cfg = _server()
    with get_session() as s:
        s.add(ModelUsage(user_id=1, model_id=1, model_name="m", date="2020-01-01",
                         prompt_tokens=10, completion_tokens=5, request_count=1))
        s.add(ModelUsage(user_id=1, model_id=1, model_name="m", date="2999-01-01",
                         prompt_tokens=1, completion_tokens=1, request_count=1))
        s.commit()
    moved = UsageArchiver(cfg, keep_days=30).archive_once()
    assert moved == 1
    with get_session() as s:
        assert s.query(ModelUsage).count() == 1
        arch = s.query(ModelUsageArchive).all()
        assert len(arch) == 1 and arch[0].prompt_tokens == 10


def test_k8s_manifests_render():
    """K8s install manifests (reference: gpustack/k8s/ daemonset.jinja):
    server Deployment/Service/Secret + ROCm worker DaemonSet."""
    import yaml

    from gpustack_amd.utils.k8s_manifests import render_all

    docs = list(yaml.safe_load_all(render_all(gpus_per_node=4)))
    kinds = [d["kind"] for d in docs]
    assert kinds == ["Namespace", "Secret", "Deployment", "Service",
                     "DaemonSet"]
    ds = docs[-1]["spec"]["template"]["spec"]
    c = ds["containers"][0]
    assert c["resources"]["limits"]["amd.com/gpu"] == 4
    paths = {v["hostPath"]["path"] for v in ds["volumes"] if "hostPath" in v}
    assert "/dev/kfd" in paths and "/dev/dri" in paths
    env = {e["name"]: e for e in c["env"]}
    assert env["HSA_ENABLE_IPC_MODE_LEGACY"]["value"] == "0"


def test_k8s_manifests_cli(capsys):
    from gpustack_amd.main import main

    assert main(["manifests", "--namespace", "prod"]) == 0
    out = capsys.readouterr().out
    assert "kind: DaemonSet" in out and "namespace: prod" in out


def test_grafana_dashboard_metric_names_exist():
    """The shipped dashboard only references metric names our exporters
    actually emit (keeps deploy/grafana honest as metrics evolve)."""
    import json as _json
    import re
    from pathlib import Path

    dash = _json.loads(Path("deploy/grafana/dashboards/gpustack-amd.json")
                       .read_text())
    exprs = [t["expr"] for p in dash["panels"] for t in p["targets"]]
    used = set()
    for e in exprs:
        used.update(re.findall(r"gpustack_[a-z_]+", e))
    emitted = set()
    for src in ["gpustack_amd/server/exporter.py",
                "gpustack_amd/worker/agent.py",
                "gpustack_amd/worker/engine_server.py"]:
        emitted.update(re.findall(r"gpustack_[a-z_]+", Path(src).read_text()))
    missing = used - emitted
    assert not missing, f"dashboard references unknown metrics: {missing}"


def test_prometheus_sd_config_points_at_targets_endpoint():
    import yaml
    from pathlib import Path

    cfg = yaml.safe_load(Path("deploy/prometheus.yml").read_text())
    sd = cfg["scrape_configs"][0]["http_sd_configs"][0]
    assert sd["url"].endswith("/metrics/targets")
