"""FastAPI app factory + server bootstrap (reference: gpustack/server/server.py:242).

Startup: init DB, seed admin user + default registration token, start the
scheduler + controllers, mount /v2 management, /v1 OpenAI, auth and
Prometheus /metrics routes.
"""
from __future__ import annotations

import logging
import threading
import time

from fastapi import APIRouter, Depends, FastAPI, HTTPException, Request, Response
from fastapi.responses import JSONResponse

from ..config import Config
from ..db import ar_create, get_session, init_db
from ..schemas import Cluster, LoginRequest, RegistrationToken, User
from ..security import generate_registration_token, hash_password, jwt_encode, verify_password
from . import deps
from .deps import COOKIE_NAME, get_current_user
from .routes_openai import router as openai_router
from .routes_v2 import router as v2_router

logger = logging.getLogger(__name__)

auth_router = APIRouter(prefix="/auth")


@auth_router.post("/login")
def login(body: LoginRequest, response: Response):
    cfg = deps.get_config()
    with get_session() as s:
        user = s.query(User).filter_by(username=body.username).first()
        if not user or not verify_password(body.password, user.hashed_password):
            raise HTTPException(401, "bad credentials")
    token = jwt_encode({"sub": body.username}, cfg.get_jwt_secret())
    response.set_cookie(COOKIE_NAME, token, httponly=True, samesite="lax")
    return {"token": token}


@auth_router.post("/logout")
def logout(response: Response):
    response.delete_cookie(COOKIE_NAME)
    return {"ok": True}


# -- OIDC authorization-code flow (reference: routes/auth.py:805-834 OIDC
# login/callback, :1312 auth-config discovery). Token validation is
# delegated to the IdP via the userinfo endpoint (the access token is
# verified server-side there), so no local RS256/JWKS stack is needed.

_oidc_discovery_cache: dict = {}


def _oidc_discover(issuer: str) -> dict:
    import httpx

    doc = _oidc_discovery_cache.get(issuer)
    if doc is None:
        url = issuer.rstrip("/") + "/.well-known/openid-configuration"
        doc = httpx.get(url, timeout=10).json()
        _oidc_discovery_cache[issuer] = doc
    return doc


@auth_router.get("/config")
def auth_config():
    cfg = deps.get_config()
    return {
        "password_login": True,
        "oidc": bool(cfg.oidc_issuer and cfg.oidc_client_id),
        "oidc_login_url": "/auth/oidc/login" if cfg.oidc_issuer else None,
        "cas": bool(cfg.cas_server_url),
        "cas_login_url": "/auth/cas/login" if cfg.cas_server_url else None,
    }


@auth_router.get("/oidc/login")
def oidc_login(request: Request, redirect_uri: str | None = None):
    from urllib.parse import urlencode

    cfg = deps.get_config()
    if not (cfg.oidc_issuer and cfg.oidc_client_id):
        raise HTTPException(404, "OIDC is not configured")
    doc = _oidc_discover(cfg.oidc_issuer)
    callback = redirect_uri or str(request.url_for("oidc_callback"))
    # self-validating state: signed + expiring, no server-side session
    state = jwt_encode({"cb": callback}, cfg.get_jwt_secret(), expires_in=600)
    q = urlencode({
        "response_type": "code",
        "client_id": cfg.oidc_client_id,
        "redirect_uri": callback,
        "scope": "openid profile email",
        "state": state,
    })
    from fastapi.responses import RedirectResponse

    return RedirectResponse(f"{doc['authorization_endpoint']}?{q}")


@auth_router.get("/oidc/callback")
def oidc_callback(code: str, state: str, response: Response):
    import httpx

    from ..security import jwt_decode

    cfg = deps.get_config()
    if not (cfg.oidc_issuer and cfg.oidc_client_id):
        raise HTTPException(404, "OIDC is not configured")
    st = jwt_decode(state, cfg.get_jwt_secret())
    if st is None:
        raise HTTPException(400, "bad or expired state")
    doc = _oidc_discover(cfg.oidc_issuer)
    tok = httpx.post(doc["token_endpoint"], data={
        "grant_type": "authorization_code",
        "code": code,
        "redirect_uri": st["cb"],
        "client_id": cfg.oidc_client_id,
        "client_secret": cfg.oidc_client_secret or "",
    }, timeout=10)
    if tok.status_code != 200:
        raise HTTPException(401, f"token exchange failed: {tok.text[:200]}")
    access = tok.json().get("access_token")
    if not access:
        raise HTTPException(401, "no access_token in IdP response")
    ui = httpx.get(doc["userinfo_endpoint"],
                   headers={"Authorization": f"Bearer {access}"}, timeout=10)
    if ui.status_code != 200:
        raise HTTPException(401, "userinfo rejected the access token")
    claims = ui.json()
    username = (claims.get(cfg.oidc_username_claim) or claims.get("email")
                or claims.get("sub"))
    if not username:
        raise HTTPException(401, "no usable username claim")
    groups = claims.get("groups") or []
    is_admin = bool(cfg.oidc_admin_group and cfg.oidc_admin_group in groups)
    with get_session() as s:
        user = s.query(User).filter_by(username=username).first()
        if user is None:  # JIT provisioning (reference group sync :726)
            user = User(username=username, hashed_password=hash_password(
                __import__("secrets").token_urlsafe(24)),
                is_admin=is_admin,
                full_name=claims.get("name") or "")
            ar_create(s, user)
        elif cfg.oidc_admin_group and user.is_admin != is_admin:
            user.is_admin = is_admin
            s.commit()
    token = jwt_encode({"sub": username}, cfg.get_jwt_secret())
    response.set_cookie(COOKIE_NAME, token, httponly=True, samesite="lax")
    return {"token": token, "username": username, "is_admin": is_admin}


@auth_router.get("/cas/login")
def cas_login(request: Request):
    """CAS SSO (reference: routes/auth.py:1019-1140): redirect to the CAS
    login with our callback as the service URL."""
    from urllib.parse import urlencode

    from fastapi.responses import RedirectResponse

    cfg = deps.get_config()
    if not cfg.cas_server_url:
        raise HTTPException(404, "CAS is not configured")
    service = str(request.url_for("cas_callback"))
    return RedirectResponse(
        f"{cfg.cas_server_url.rstrip('/')}/login?{urlencode({'service': service})}")


@auth_router.get("/cas/callback")
def cas_callback(ticket: str, request: Request, response: Response):
    """Validate the service ticket via CAS /serviceValidate (XML)."""
    import xml.etree.ElementTree as ET
    from urllib.parse import urlencode

    import httpx

    cfg = deps.get_config()
    if not cfg.cas_server_url:
        raise HTTPException(404, "CAS is not configured")
    service = str(request.url_for("cas_callback"))
    r = httpx.get(
        f"{cfg.cas_server_url.rstrip('/')}/serviceValidate?"
        + urlencode({"ticket": ticket, "service": service}), timeout=10)
    if r.status_code != 200:
        raise HTTPException(401, "CAS validation failed")
    ns = {"cas": "http://www.yale.edu/tp/cas"}
    try:
        root = ET.fromstring(r.text)
    except ET.ParseError:
        raise HTTPException(401, "CAS returned invalid XML")
    ok = root.find("cas:authenticationSuccess", ns)
    if ok is None:
        raise HTTPException(401, "CAS rejected the ticket")
    user_el = ok.find("cas:user", ns)
    if user_el is None or not (user_el.text or "").strip():
        raise HTTPException(401, "CAS response has no user")
    username = user_el.text.strip()
    with get_session() as s:
        user = s.query(User).filter_by(username=username).first()
        if user is None:
            user = User(username=username, hashed_password=hash_password(
                __import__("secrets").token_urlsafe(24)))
            ar_create(s, user)
    token = jwt_encode({"sub": username}, cfg.get_jwt_secret())
    response.set_cookie(COOKIE_NAME, token, httponly=True, samesite="lax")
    return {"token": token, "username": username}


@auth_router.get("/me")
def me(user: User = Depends(get_current_user)):
    return {"id": user.id, "username": user.username, "is_admin": user.is_admin}


def bootstrap_data(cfg: Config) -> dict:
    """Admin user + default registration token (reference: server.py:381,750)."""
    out = {}
    with get_session() as s:
        admin = s.query(User).filter_by(username="admin").first()
        if admin is None:
            import secrets

            password = cfg.bootstrap_password or secrets.token_urlsafe(12)
            admin = User(username="admin", hashed_password=hash_password(password),
                         is_admin=True)
            ar_create(s, admin)
            out["admin_password"] = password
            logger.info("bootstrap admin user created (password: %s)", password)
        default = s.query(Cluster).filter_by(is_default=True).first()
        if default is None:
            default = Cluster(name="default", description="default cluster",
                              is_default=True)
            ar_create(s, default)
        tok = s.query(RegistrationToken).first()
        if tok is None:
            value = cfg.token or generate_registration_token()
            tok = RegistrationToken(token=value, description="default",
                                    cluster_id=default.id)
            ar_create(s, tok)
        elif tok.cluster_id is None:
            tok.cluster_id = default.id
            s.commit()
        out["registration_token"] = tok.token
        out["default_cluster_id"] = default.id
    return out


def create_app(cfg: Config, start_background: bool = True) -> FastAPI:
    cfg.ensure_dirs()
    init_db(cfg.resolved_database_url())
    deps.init_auth(cfg)
    boot = bootstrap_data(cfg)
    if cfg.token is None:
        cfg.token = boot["registration_token"]
    # persist the registration token where operators (and the all-in-one
    # container's embedded worker) expect it — reference behavior: the
    # server writes <data_dir>/token and `gpustack start --server-url`
    # workers read it
    from pathlib import Path as _Path

    tok_path = _Path(cfg.data_dir) / "token"
    try:
        if (not tok_path.exists()
                or tok_path.read_text().strip() != cfg.token):
            tok_path.write_text(cfg.token + "\n")
            tok_path.chmod(0o600)
    except OSError as e:  # read-only data dir: log, don't fail startup
        logger.warning("could not persist registration token: %s", e)

    app = FastAPI(title="gpustack_amd", version="0.1.0")
    app.state.config = cfg
    app.state.bootstrap = boot

    app.include_router(auth_router)
    app.include_router(v2_router)
    app.include_router(openai_router)
    from ..extension import apply_routers, load_plugins, run_start_hooks

    plugins = load_plugins()
    app.state.plugins = plugins
    apply_routers(app, plugins)
    run_start_hooks(app, cfg, plugins)

    @app.get("/healthz")
    def healthz():
        return {"status": "ok"}

    @app.get("/readyz")
    def readyz():
        return {"status": "ok"}

    @app.get("/metrics")
    def metrics():
        from .exporter import render_metrics

        return Response(render_metrics(), media_type="text/plain; version=0.0.4")

    @app.get("/metrics/targets")
    def metrics_targets():
        """Prometheus http_sd discovery of every worker's /metrics endpoint
        (reference: exporter/exporter.py:272-295)."""
        from ..db import get_session as _gs
        from ..schemas import Worker, WorkerState

        targets = []
        with _gs() as s:
            for w in s.query(Worker).all():
                if w.state != WorkerState.READY.value or not w.ip:
                    continue
                port = w.metrics_port or w.port
                targets.append({
                    "targets": [f"{w.ip}:{port}"],
                    "labels": {"worker": w.name,
                               "instance": f"{w.ip}:{port}"},
                })
        return targets

    @app.exception_handler(Exception)
    async def unhandled(request, exc):
        logger.exception("unhandled error: %s", exc)
        return JSONResponse({"error": {"message": str(exc)}}, status_code=500)

    if start_background:
        start_background_tasks(cfg, app)
    return app


def start_background_tasks(cfg: Config, app: FastAPI) -> None:
    from ..scheduler.scheduler import PlacementScheduler
    from .controllers import (
        ModelController, ResourceEventLogger, ScalingScheduler,
        SystemLoadCollector, UsageArchiver, WorkerMonitor,
        WorkerPoolController,
    )
    from .coordinator import LeaseCoordinator, LocalCoordinator

    # leader election only matters with a shared external DB; a lease over
    # SQLite still works for tests / local HA pairs
    from ..extension import pick_coordinator

    coord = pick_coordinator(cfg, getattr(app.state, "plugins", []))
    if coord is None:
        if getattr(cfg, "ha_leases", False) or cfg.database_url:
            coord = LeaseCoordinator()
            coord.try_acquire()
            coord.start()
        else:
            coord = LocalCoordinator()
    app.state.coordinator = coord

    from .gpu_instances import GPUInstanceController

    sched = PlacementScheduler(cfg)
    tasks = [sched, ModelController(cfg), WorkerMonitor(cfg),
             SystemLoadCollector(cfg), ScalingScheduler(cfg), UsageArchiver(cfg),
             WorkerPoolController(cfg), ResourceEventLogger(cfg),
             GPUInstanceController(cfg)]
    for t in tasks:
        t.coordinator = coord  # leader-only gating (checked per cycle)
    app.state.scheduler = sched
    app.state.background_tasks = tasks
    threads = [
        threading.Thread(target=t.run, name=type(t).__name__, daemon=True)
        for t in tasks
    ]
    for t in threads:
        t.start()
    app.state.background_threads = threads


def stop_background_tasks(app: FastAPI) -> None:
    for t in getattr(app.state, "background_tasks", []):
        t.stop()
    coord = getattr(app.state, "coordinator", None)
    if coord is not None:
        coord.stop()


def run_server(cfg: Config) -> None:
    import uvicorn

    app = create_app(cfg)
    logger.info("server listening on %s:%d", cfg.host, cfg.port)
    uvicorn.run(app, host=cfg.host, port=cfg.port, log_level="info")
