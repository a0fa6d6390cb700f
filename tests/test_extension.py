"""Extension plugin hooks (reference: gpustack/extension.py entry points)."""
import tempfile

from fastapi.testclient import TestClient

from gpustack_amd import extension
from gpustack_amd.config import Config
from gpustack_amd.server.app import create_app


class _Plugin:
    loaded = []

    @staticmethod
    def routers():
        from fastapi import APIRouter

        r = APIRouter()

        @r.get("/v2/plugin-ping")
        def ping():
            return {"pong": True}

        return [r]

    @staticmethod
    def coordinator(cfg):
        return None  # fall through to default

    @staticmethod
    def on_server_start(app, cfg):
        _Plugin.loaded.append(cfg.data_dir)


def test_plugin_router_and_hooks(monkeypatch):
    monkeypatch.setattr(extension, "load_plugins", lambda: [_Plugin])
    cfg = Config(data_dir=tempfile.mkdtemp(), bootstrap_password="pw")
    app = create_app(cfg, start_background=False)
    c = TestClient(app)
    assert c.get("/v2/plugin-ping").json() == {"pong": True}
    assert _Plugin.loaded and _Plugin.loaded[-1] == cfg.data_dir


def test_metrics_targets_http_sd():
    cfg = Config(data_dir=tempfile.mkdtemp(), bootstrap_password="pw")
    app = create_app(cfg, start_background=False)
    c = TestClient(app)
    tok = c.post("/auth/login", json={"username": "admin",
                                      "password": "pw"}).json()["token"]
    c.headers["Authorization"] = f"Bearer {tok}"
    assert c.get("/metrics/targets").json() == []
