"""CLI entry (reference: gpustack/main.py + cmd/start.py).

  python -m gpustack_amd start [--config FILE] [flags]           # server
  python -m gpustack_amd start --server-url http://... --token T # worker
  python -m gpustack_amd chat --model NAME --prompt "..."        # client
  python -m gpustack_amd version
"""
from __future__ import annotations

import argparse
import logging
import sys


def _add_start_flags(p: argparse.ArgumentParser) -> None:
    p.add_argument("--config", dest="config_file", default=None)
    p.add_argument("--server-url", dest="server_url", default=None)
    p.add_argument("--host", default=None)
    p.add_argument("--port", type=int, default=None)
    p.add_argument("--data-dir", dest="data_dir", default=None)
    p.add_argument("--database-url", dest="database_url", default=None)
    p.add_argument("--token", default=None)
    p.add_argument("--worker-name", dest="worker_name", default=None)
    p.add_argument("--worker-ip", dest="worker_ip", default=None)
    p.add_argument("--worker-port", dest="worker_port", type=int, default=None)
    p.add_argument("--bootstrap-password", dest="bootstrap_password", default=None)
    p.add_argument("--disable-auth", dest="disable_auth", action="store_const", const=True, default=None)
    p.add_argument("--debug", action="store_true")


def main(argv: list[str] | None = None) -> int:
    ap = argparse.ArgumentParser(prog="gpustack-amd")
    sub = ap.add_subparsers(dest="cmd")
    sp = sub.add_parser("start", help="run server (default) or worker (--server-url)")
    _add_start_flags(sp)
    cp = sub.add_parser("chat", help="quick chat against a served model")
    cp.add_argument("--url", default="http://127.0.0.1:8080")
    cp.add_argument("--api-key", default=None)
    cp.add_argument("--model", required=True)
    cp.add_argument("--prompt", required=True)
    rp = sub.add_parser("reset-admin-password")
    rp.add_argument("--data-dir", dest="data_dir", default=None)
    rp.add_argument("--database-url", dest="database_url", default=None)
    rp.add_argument("--password", required=True)
    mp = sub.add_parser("migrate", help="apply pending DB schema migrations")
    mp.add_argument("--database-url", default=None)
    mp.add_argument("--data-dir", default=None)
    kp = sub.add_parser("manifests", help="print Kubernetes install manifests")
    kp.add_argument("--namespace", default="gpustack")
    kp.add_argument("--image", default="gpustack-amd:latest")
    kp.add_argument("--server-url", default=None)
    kp.add_argument("--bootstrap-password", default="admin")
    kp.add_argument("--registration-token", default="tok_cluster")
    kp.add_argument("--gpus-per-node", type=int, default=8)
    sub.add_parser("version")
    args = ap.parse_args(argv)

    if args.cmd == "manifests":
        from .utils.k8s_manifests import render_all

        print(render_all(args.namespace, args.image, args.server_url,
                         args.bootstrap_password, args.registration_token,
                         args.gpus_per_node))
        return 0
    if args.cmd == "version":
        from . import __version__

        print(__version__)
        return 0
    if args.cmd == "reset-admin-password":
        from .config import load_config
        from .db import get_session, init_db
        from .schemas import User
        from .security import hash_password

        cfg = load_config(None, {"data_dir": args.data_dir,
                                 "database_url": args.database_url})
        init_db(cfg.resolved_database_url())
        with get_session() as s:
            admin = s.query(User).filter_by(username="admin").first()
            if admin is None:
                admin = User(username="admin", hashed_password="", is_admin=True)
                s.add(admin)
            admin.hashed_password = hash_password(args.password)
            s.commit()
        print("admin password reset")
        return 0
    if args.cmd == "migrate":
        from .config import load_config
        from .db import get_engine, init_db
        from .db.migrations import HEAD, current_version, migrate

        cfg = load_config(None, {"data_dir": args.data_dir,
                                 "database_url": args.database_url})
        init_db(cfg.resolved_database_url())  # runs pending migrations itself
        with get_engine().begin() as conn:
            v = current_version(conn)
        print(f"schema at v{v} (head v{HEAD})")
        return 0
    if args.cmd == "chat":
        import httpx

        headers = {"Authorization": f"Bearer {args.api_key}"} if args.api_key else {}
        r = httpx.post(f"{args.url}/v1/chat/completions", headers=headers, json={
            "model": args.model,
            "messages": [{"role": "user", "content": args.prompt}],
            "max_tokens": 64,
        }, timeout=120)
        print(r.json())
        return 0
    if args.cmd != "start":
        ap.print_help()
        return 1

    logging.basicConfig(
        level=logging.DEBUG if args.debug else logging.INFO,
        format="%(asctime)s %(levelname)s %(name)s: %(message)s",
    )
    from .config import load_config

    overrides = {
        k: getattr(args, k)
        for k in ("server_url", "host", "port", "data_dir", "database_url", "token",
                  "worker_name", "worker_ip", "worker_port", "bootstrap_password",
                  "disable_auth")
    }
    cfg = load_config(args.config_file, overrides)
    if cfg.server_role == "worker":
        from .worker.agent import run_worker

        run_worker(cfg)
    else:
        from .server.app import run_server

        run_server(cfg)
    return 0


if __name__ == "__main__":
    sys.exit(main())
