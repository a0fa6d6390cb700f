"""Auth primitives: password hashing (scrypt), HS256 JWT (stdlib hmac),
API keys in the reference's `<prefix>_<access>_<secret>` format
(reference: gpustack/security.py, gpustack/api/auth.py:385-485)."""
from __future__ import annotations

import base64
import hashlib
import hmac
import json
import os
import secrets
import time

API_KEY_PREFIX = "gsa"


# -- passwords -------------------------------------------------------------

def hash_password(password: str) -> str:
    salt = os.urandom(16)
    digest = hashlib.scrypt(password.encode(), salt=salt, n=2**14, r=8, p=1)
    return f"scrypt${salt.hex()}${digest.hex()}"


def verify_password(password: str, hashed: str) -> bool:
    try:
        _, salt_hex, digest_hex = hashed.split("$")
        digest = hashlib.scrypt(password.encode(), salt=bytes.fromhex(salt_hex), n=2**14, r=8, p=1)
        return hmac.compare_digest(digest.hex(), digest_hex)
    except Exception:  # noqa: BLE001
        return False


# -- JWT (HS256) -----------------------------------------------------------

def _b64(data: bytes) -> str:
    return base64.urlsafe_b64encode(data).rstrip(b"=").decode()


def _unb64(s: str) -> bytes:
    return base64.urlsafe_b64decode(s + "=" * (-len(s) % 4))


def jwt_encode(payload: dict, secret: str, expires_in: float = 86400 * 7) -> str:
    header = {"alg": "HS256", "typ": "JWT"}
    body = dict(payload, exp=time.time() + expires_in)
    signing = f"{_b64(json.dumps(header).encode())}.{_b64(json.dumps(body).encode())}"
    sig = hmac.new(secret.encode(), signing.encode(), hashlib.sha256).digest()
    return f"{signing}.{_b64(sig)}"


def jwt_decode(token: str, secret: str) -> dict | None:
    try:
        h, b, s = token.split(".")
        sig = hmac.new(secret.encode(), f"{h}.{b}".encode(), hashlib.sha256).digest()
        if not hmac.compare_digest(_b64(sig), s):
            return None
        payload = json.loads(_unb64(b))
        if payload.get("exp", 0) < time.time():
            return None
        return payload
    except Exception:  # noqa: BLE001
        return None


# -- API keys --------------------------------------------------------------

def generate_api_key() -> tuple[str, str, str]:
    """Returns (full_key, access_key, hashed_secret)."""
    access = secrets.token_hex(8)
    secret = secrets.token_hex(24)
    return f"{API_KEY_PREFIX}_{access}_{secret}", access, hash_password(secret)


def parse_api_key(key: str) -> tuple[str, str] | None:
    parts = key.split("_")
    if len(parts) != 3 or parts[0] != API_KEY_PREFIX:
        return None
    return parts[1], parts[2]


def generate_registration_token() -> str:
    return f"tok_{secrets.token_hex(16)}"
