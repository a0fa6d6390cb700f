"""Configuration (reference parity: gpustack/config/config.py:109).

Precedence CLI > YAML config file > GPUSTACK_AMD_* env vars > defaults,
merged into one Config object (reference: cmd/start.py:763-781).
"""
from __future__ import annotations

import os
import secrets
from pathlib import Path

import yaml
from pydantic import BaseModel, Field

ENV_PREFIX = "GPUSTACK_AMD_"


class Config(BaseModel):
    # role
    server_url: str | None = None          # set => worker role
    # paths
    data_dir: str = Field(default_factory=lambda: os.path.expanduser("~/.gpustack-amd"))
    cache_dir: str | None = None
    # database
    database_url: str | None = None        # default: sqlite under data_dir
    # server
    host: str = "0.0.0.0"
    port: int = 8080
    metrics_port: int = 10151
    # auth
    ha_leases: bool = False                # force lease-based leader election
    # external auth: OIDC authorization-code flow (reference:
    # routes/auth.py:805,834 — OIDC login/callback + group sync)
    oidc_issuer: str | None = None         # e.g. https://idp/realms/x
    oidc_client_id: str | None = None
    oidc_client_secret: str | None = None
    oidc_username_claim: str = "preferred_username"
    oidc_admin_group: str | None = None    # groups claim granting is_admin
    # CAS external auth (reference: routes/auth.py:1019-1140)
    cas_server_url: str | None = None      # e.g. https://cas.corp/cas
    bootstrap_password: str | None = None
    jwt_secret: str | None = None
    disable_auth: bool = False
    # worker
    worker_ip: str | None = None
    worker_port: int = 10150
    worker_metrics_port: int = 10152
    token: str | None = None               # worker registration token
    worker_name: str | None = None
    labels: dict[str, str] = Field(default_factory=dict)
    gpu_devices: list[dict] | None = None  # static override (air-gapped)
    proxy_mode: str = "direct"             # direct | tunnel (NAT workers)
    system_reserved: dict = Field(default_factory=lambda: {"ram": 2 << 30, "vram": 1 << 30})
    # engine defaults
    gpu_memory_utilization: float = 0.9
    port_range: str = "40000-41000"        # engine instance ports
    heartbeat_interval: float = 15.0
    worker_status_interval: float = 30.0

    @property
    def server_role(self) -> str:
        return "worker" if self.server_url else "server"

    def resolved_database_url(self) -> str:
        if self.database_url:
            return self.database_url
        return f"sqlite:///{Path(self.data_dir) / 'gpustack.db'}"

    def ensure_dirs(self) -> None:
        Path(self.data_dir).mkdir(parents=True, exist_ok=True)
        (Path(self.data_dir) / "log").mkdir(exist_ok=True)

    def get_jwt_secret(self) -> str:
        if self.jwt_secret:
            return self.jwt_secret
        path = Path(self.data_dir) / "jwt_secret"
        if path.exists():
            self.jwt_secret = path.read_text().strip()
        else:
            self.ensure_dirs()
            self.jwt_secret = secrets.token_hex(32)
            path.write_text(self.jwt_secret)
            path.chmod(0o600)
        return self.jwt_secret

    def engine_port_range(self) -> tuple[int, int]:
        lo, hi = self.port_range.split("-")
        return int(lo), int(hi)


def load_config(config_file: str | None = None, cli_overrides: dict | None = None) -> Config:
    """Merge defaults < env < yaml < cli (reference precedence semantics)."""
    data: dict = {}
    for name in Config.model_fields:
        env = os.environ.get(ENV_PREFIX + name.upper())
        if env is not None:
            field = Config.model_fields[name]
            ann = str(field.annotation)
            if "dict" in ann or "list" in ann:
                data[name] = yaml.safe_load(env)
            elif "bool" in ann:
                data[name] = env.lower() in ("1", "true", "yes")
            else:
                data[name] = env
    if config_file:
        with open(config_file) as f:
            data.update(yaml.safe_load(f) or {})
    for k, v in (cli_overrides or {}).items():
        if v is not None:
            data[k] = v
    return Config(**data)
