"""FastAPI auth dependencies (reference: gpustack/api/auth.py).

Supported credentials: JWT session cookie, `Authorization: Bearer <jwt>`,
API keys (gsa_<access>_<secret>), and worker registration tokens for the
worker-facing endpoints.
"""
from __future__ import annotations

from fastapi import Depends, HTTPException, Request

from ..db import get_session
from ..schemas import ApiKey, RegistrationToken, User
from ..security import jwt_decode, parse_api_key, verify_password

COOKIE_NAME = "gpustack_session"


class AuthContext:
    def __init__(self, config):
        self.config = config


_ctx: AuthContext | None = None


def init_auth(config) -> None:
    global _ctx
    _ctx = AuthContext(config)


def get_config():
    assert _ctx is not None
    return _ctx.config


def _bearer(request: Request) -> str | None:
    auth = request.headers.get("authorization", "")
    if auth.lower().startswith("bearer "):
        return auth[7:].strip()
    return None


def resolve_user(request: Request) -> User | None:
    cfg = get_config()
    token = _bearer(request) or request.cookies.get(COOKIE_NAME)
    if not token:
        return None
    # API key?
    parsed = parse_api_key(token)
    if parsed:
        access, secret = parsed
        with get_session() as s:
            key = s.query(ApiKey).filter_by(access_key=access).first()
            if not key or not verify_password(secret, key.hashed_secret):
                return None
            import time

            if key.expires_at and key.expires_at < time.time():
                return None
            return s.get(User, key.user_id)
    payload = jwt_decode(token, cfg.get_jwt_secret())
    if not payload:
        return None
    with get_session() as s:
        return s.query(User).filter_by(username=payload.get("sub", "")).first()


def _system_principal(request: Request) -> User | None:
    """Registration-token bearers act as a system principal (reference:
    api/auth.py:262-289 worker/cluster system principals)."""
    token = _bearer(request)
    if not token or not token.startswith("tok_"):
        return None
    with get_session() as s:
        if s.query(RegistrationToken).filter_by(token=token).first():
            return User(id=0, username="system:worker", is_admin=True,
                        hashed_password="")
    return None


def get_current_user(request: Request) -> User:
    cfg = get_config()
    if cfg.disable_auth:
        with get_session() as s:
            u = s.query(User).filter_by(is_admin=True).first()
            if u:
                return u
    user = resolve_user(request) or _system_principal(request)
    if user is None:
        raise HTTPException(401, "not authenticated")
    return user


def get_admin_user(user: User = Depends(get_current_user)) -> User:
    if not user.is_admin:
        raise HTTPException(403, "admin required")
    return user


def verify_worker_token(request: Request) -> None:
    """Worker-facing endpoints authenticate with a registration token
    (reference: api/auth.py:262-289 system principals)."""
    cfg = get_config()
    if cfg.disable_auth:
        return
    token = _bearer(request)
    if not token:
        raise HTTPException(401, "worker token required")
    with get_session() as s:
        if s.query(RegistrationToken).filter_by(token=token).first():
            return
    raise HTTPException(401, "invalid worker token")


def model_allowed_for_user(user, model) -> bool:
    """Tenancy gate (reference: services.py:57 UserService
    .model_allowed_for_user): platform admins see everything; org-scoped
    models are visible to that org's members only; unscoped models are
    public to every authenticated user."""
    if getattr(user, "is_admin", False):
        return True
    org = getattr(model, "org_id", None)
    if org is None and isinstance(model, dict):
        org = model.get("org_id")
    return org is None or org == getattr(user, "org_id", None)
