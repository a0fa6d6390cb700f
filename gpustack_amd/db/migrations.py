"""Versioned schema migrations (reference: alembic revisions under
gpustack/migrations/ + the `migrate` CLI, cmd/db_migration.py).

Re-designed without alembic: an ordered list of (version, description,
fn) steps and a `schema_version` table. `create_all` brings a FRESH
database straight to head (SQLAlchemy models are the source of truth);
migrations exist for databases created by OLDER builds, where
`create_all` does not alter existing tables. Each step must therefore be
written additively (ADD COLUMN / CREATE TABLE IF NOT EXISTS) and be a
no-op on a database that already has the change.
"""
from __future__ import annotations

import logging

from sqlalchemy import inspect, text

logger = logging.getLogger(__name__)


def _add_column(conn, table: str, column: str, ddl: str) -> None:
    cols = [c["name"] for c in inspect(conn).get_columns(table)]
    if column not in cols:
        conn.execute(text(f"ALTER TABLE {table} ADD COLUMN {column} {ddl}"))


# ---- migration steps -------------------------------------------------------
# Append-only: never renumber or edit a shipped step.

def _m001_worker_proxy_mode(conn):
    _add_column(conn, "workers", "proxy_mode", "VARCHAR(16) DEFAULT 'direct'")


def _m002_model_kv_features(conn):
    _add_column(conn, "models", "extended_kv_cache", "JSON")
    _add_column(conn, "models", "speculative_config", "JSON")
    _add_column(conn, "models", "scaling_schedule", "JSON")


def _m003_instance_distributed_servers(conn):
    _add_column(conn, "model_instances", "distributed_servers", "JSON")


def _m004_instance_spec_hash(conn):
    _add_column(conn, "model_instances", "spec_hash", "VARCHAR(64) DEFAULT ''")


def _m005_model_lora_adapters(conn):
    _add_column(conn, "models", "lora_adapters", "JSON")


def _m006_worker_pools(conn):
    from ..schemas.tables import WorkerPool
    WorkerPool.__table__.create(conn, checkfirst=True)


def _m007_clusters(conn):
    from ..schemas.tables import Cluster
    Cluster.__table__.create(conn, checkfirst=True)
    _add_column(conn, "workers", "cluster_id", "INTEGER")
    _add_column(conn, "models", "cluster_id", "INTEGER")
    _add_column(conn, "registration_tokens", "cluster_id", "INTEGER")


def _m008_orgs(conn):
    from ..schemas.tables import Org
    Org.__table__.create(conn, checkfirst=True)
    _add_column(conn, "users", "org_id", "INTEGER")
    _add_column(conn, "models", "org_id", "INTEGER")


def _m009_gpu_type_selector(conn):
    _add_column(conn, "models", "gpu_type_selector", "JSON")


def _m010_resource_events(conn):
    from ..schemas.tables import ResourceEvent, ResourceEventArchive
    ResourceEvent.__table__.create(conn, checkfirst=True)
    ResourceEventArchive.__table__.create(conn, checkfirst=True)


def _m011_gpu_instances(conn):
    from ..schemas.tables import GPUInstance
    GPUInstance.__table__.create(conn, checkfirst=True)


def _m012_gpu_instance_templates(conn):
    from ..schemas.tables import GPUInstanceTemplate, SSHPublicKey
    GPUInstanceTemplate.__table__.create(conn, checkfirst=True)
    SSHPublicKey.__table__.create(conn, checkfirst=True)


MIGRATIONS: list[tuple[int, str, object]] = [
    (1, "worker.proxy_mode for tunnel workers", _m001_worker_proxy_mode),
    (2, "model KV/speculative/scaling columns", _m002_model_kv_features),
    (3, "instance cross-worker rank layout", _m003_instance_distributed_servers),
    (4, "instance spec_hash for update-triggered redeploy", _m004_instance_spec_hash),
    (5, "model.lora_adapters for dynamic multi-LoRA", _m005_model_lora_adapters),
    (6, "worker_pools table for auto-provisioned capacity", _m006_worker_pools),
    (7, "multi-cluster: clusters table + cluster_id columns", _m007_clusters),
    (8, "orgs table + user/model org scoping", _m008_orgs),
    (9, "model.gpu_type_selector for device-class placement", _m009_gpu_type_selector),
    (10, "resource-event metering pair (hot + archive)", _m010_resource_events),
    (11, "gpu_instances table (operator-analog SSH GPU pods)", _m011_gpu_instances),
    (12, "gpu-instance templates + ssh public keys", _m012_gpu_instance_templates),
]

HEAD = MIGRATIONS[-1][0] if MIGRATIONS else 0


def current_version(conn) -> int:
    conn.execute(text(
        "CREATE TABLE IF NOT EXISTS schema_version (version INTEGER NOT NULL)"))
    row = conn.execute(text("SELECT version FROM schema_version")).fetchone()
    if row is None:
        conn.execute(text("INSERT INTO schema_version (version) VALUES (0)"))
        return 0
    return int(row[0])


def _set_version(conn, v: int) -> None:
    conn.execute(text("UPDATE schema_version SET version = :v"), {"v": v})


def stamp_head(engine) -> None:
    """Mark a freshly-created (already at head) schema as up to date."""
    with engine.begin() as conn:
        current_version(conn)
        _set_version(conn, HEAD)


def migrate(engine) -> list[int]:
    """Apply pending steps; returns the versions applied."""
    applied: list[int] = []
    with engine.begin() as conn:
        v = current_version(conn)
        for ver, desc, fn in MIGRATIONS:
            if ver <= v:
                continue
            logger.info("migrating schema to v%d: %s", ver, desc)
            fn(conn)
            _set_version(conn, ver)
            applied.append(ver)
    return applied
