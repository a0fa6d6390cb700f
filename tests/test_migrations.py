"""Versioned schema migrations (reference: alembic + `migrate` CLI)."""
import sqlite3


def test_fresh_db_is_stamped_head(tmp_path):
    from gpustack_amd.db import get_engine, init_db
    from gpustack_amd.db.migrations import HEAD, current_version

    init_db(f"sqlite:///{tmp_path}/m.db")
    with get_engine().begin() as conn:
        assert current_version(conn) == HEAD


def test_old_db_upgrades(tmp_path):
    # simulate a DB from an older build: workers table without proxy_mode,
    # models without the KV feature columns
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
        assert current_version(conn) == HEAD
    con = sqlite3.connect(db)
    cols = [r[1] for r in con.execute("PRAGMA table_info(workers)")]
    assert "proxy_mode" in cols
    cols = [r[1] for r in con.execute("PRAGMA table_info(models)")]
    assert "speculative_config" in cols and "scaling_schedule" in cols
    cols = [r[1] for r in con.execute("PRAGMA table_info(model_instances)")]
    assert "distributed_servers" in cols
    con.close()


def test_migrate_idempotent(tmp_path):
    from gpustack_amd.db import get_engine, init_db
    from gpustack_amd.db.migrations import migrate

    init_db(f"sqlite:///{tmp_path}/i.db")
    assert migrate(get_engine()) == []  # nothing pending at head


def test_migrate_cli(tmp_path):
    from gpustack_amd.main import main

    rc = main(["migrate", "--database-url", f"sqlite:///{tmp_path}/c.db"])
    assert rc == 0
