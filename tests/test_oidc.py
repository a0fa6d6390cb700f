"""OIDC external auth (reference: routes/auth.py OIDC login/callback +
auth-config discovery): full authorization-code flow against a stub IdP."""
import socket
import tempfile
import threading
import time
from urllib.parse import parse_qs, urlparse

import httpx
import pytest
import uvicorn
from fastapi import FastAPI, Request
from fastapi.testclient import TestClient

from gpustack_amd.config import Config
from gpustack_amd.server.app import create_app


@pytest.fixture()
def idp():
    """Minimal OIDC IdP: discovery + token + userinfo."""
    stub = FastAPI()
    state = {"codes": {"code-1": "alice"}, "tokens": {}}

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    base = f"http://127.0.0.1:{port}"

    @stub.get("/.well-known/openid-configuration")
    def disco():
        return {
            "issuer": base,
            "authorization_endpoint": f"{base}/authorize",
            "token_endpoint": f"{base}/token",
            "userinfo_endpoint": f"{base}/userinfo",
        }

    @stub.post("/token")
    async def token(request: Request):
        form = parse_qs((await request.body()).decode())
        code = form.get("code", [""])[0]
        user = state["codes"].get(code)
        if user is None or form.get("client_id", [""])[0] != "gpustack":
            from fastapi.responses import JSONResponse

            return JSONResponse({"error": "bad code"}, status_code=401)
        at = f"at-{code}"
        state["tokens"][at] = user
        return {"access_token": at, "token_type": "bearer"}

    @stub.get("/userinfo")
    def userinfo(request: Request):
        auth = request.headers.get("authorization", "")
        user = state["tokens"].get(auth.removeprefix("Bearer "))
        if user is None:
            from fastapi.responses import JSONResponse

            return JSONResponse({"error": "bad token"}, status_code=401)
        return {"sub": "idp-1", "preferred_username": user,
                "email": f"{user}@corp", "name": "Alice A",
                "groups": ["devs", "gpustack-admins"]}

    srv = uvicorn.Server(uvicorn.Config(stub, host="127.0.0.1", port=port,
                                        log_level="warning"))
    threading.Thread(target=srv.run, daemon=True).start()
    for _ in range(100):
        try:
            httpx.get(f"{base}/.well-known/openid-configuration", timeout=1)
            break
        except httpx.HTTPError:
            time.sleep(0.1)
    yield base
    srv.should_exit = True


def _server(base, **extra):
    cfg = Config(data_dir=tempfile.mkdtemp(), bootstrap_password="pw123",
                 oidc_issuer=base, oidc_client_id="gpustack",
                 oidc_client_secret="s3", **extra)
    app = create_app(cfg, start_background=False)
    return TestClient(app), app


def test_auth_config_discovery(idp):
    client, app = _server(idp)
    doc = client.get("/auth/config").json()
    assert doc["oidc"] is True
    assert doc["oidc_login_url"] == "/auth/oidc/login"


def test_oidc_full_flow(idp):
    client, app = _server(idp, oidc_admin_group="gpustack-admins")
    # login redirects to the IdP authorize endpoint with our client_id+state
    r = client.get("/auth/oidc/login", follow_redirects=False)
    assert r.status_code in (302, 307)
    loc = urlparse(r.headers["location"])
    q = parse_qs(loc.query)
    assert q["client_id"] == ["gpustack"]
    assert q["response_type"] == ["code"]
    state = q["state"][0]

    # IdP redirects back with a code; callback exchanges + provisions user
    r = client.get("/auth/oidc/callback",
                   params={"code": "code-1", "state": state})
    assert r.status_code == 200, r.text
    body = r.json()
    assert body["username"] == "alice"
    assert body["is_admin"] is True  # gpustack-admins group mapped
    tok = body["token"]

    c2 = TestClient(app)
    c2.headers["Authorization"] = f"Bearer {tok}"
    me = c2.get("/auth/me").json()
    assert me["username"] == "alice" and me["is_admin"] is True


def test_oidc_rejects_bad_state_and_code(idp):
    client, app = _server(idp)
    r = client.get("/auth/oidc/callback",
                   params={"code": "code-1", "state": "forged"})
    assert r.status_code == 400
    r = client.get("/auth/oidc/login", follow_redirects=False)
    state = parse_qs(urlparse(r.headers["location"]).query)["state"][0]
    r = client.get("/auth/oidc/callback",
                   params={"code": "wrong", "state": state})
    assert r.status_code == 401


def test_oidc_unconfigured_404():
    cfg = Config(data_dir=tempfile.mkdtemp(), bootstrap_password="pw123")
    client = TestClient(create_app(cfg, start_background=False))
    assert client.get("/auth/oidc/login",
                      follow_redirects=False).status_code == 404
    assert client.get("/auth/config").json()["oidc"] is False


@pytest.fixture()
def cas():
    """Minimal CAS server: /login redirect target + /serviceValidate XML."""
    from fastapi.responses import PlainTextResponse

    stub = FastAPI()

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    base = f"http://127.0.0.1:{port}"

    @stub.get("/cas/serviceValidate")
    def validate(ticket: str, service: str):
        if ticket == "ST-good":
            body = ('<cas:serviceResponse xmlns:cas="http://www.yale.edu/tp/cas">'
                    "<cas:authenticationSuccess><cas:user>carol</cas:user>"
                    "</cas:authenticationSuccess></cas:serviceResponse>")
        else:
            body = ('<cas:serviceResponse xmlns:cas="http://www.yale.edu/tp/cas">'
                    '<cas:authenticationFailure code="INVALID_TICKET">bad'
                    "</cas:authenticationFailure></cas:serviceResponse>")
        return PlainTextResponse(body, media_type="application/xml")

    srv = uvicorn.Server(uvicorn.Config(stub, host="127.0.0.1", port=port,
                                        log_level="warning"))
    threading.Thread(target=srv.run, daemon=True).start()
    for _ in range(100):
        try:
            httpx.get(f"{base}/cas/serviceValidate?ticket=x&service=y",
                      timeout=1)
            break
        except httpx.HTTPError:
            time.sleep(0.1)
    yield f"{base}/cas"
    srv.should_exit = True


def test_cas_flow(cas):
    cfg = Config(data_dir=tempfile.mkdtemp(), bootstrap_password="pw123",
                 cas_server_url=cas)
    client = TestClient(create_app(cfg, start_background=False))
    doc = client.get("/auth/config").json()
    assert doc["cas"] is True
    r = client.get("/auth/cas/login", follow_redirects=False)
    assert r.status_code in (302, 307)
    assert r.headers["location"].startswith(cas + "/login?service=")
    r = client.get("/auth/cas/callback", params={"ticket": "ST-good"})
    assert r.status_code == 200, r.text
    tok = r.json()["token"]
    assert r.json()["username"] == "carol"
    c2 = TestClient(client.app)
    c2.headers["Authorization"] = f"Bearer {tok}"
    assert c2.get("/auth/me").json()["username"] == "carol"
    # bad ticket rejected
    assert client.get("/auth/cas/callback",
                      params={"ticket": "ST-bad"}).status_code == 401
