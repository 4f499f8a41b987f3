"""HS256 JWT auth — dependency-free implementation.

Capability parity with /root/reference/pkg/handlers/auth.go:25-73 (login with
HS256, 24 h expiry) and pkg/middleware/jwt.go:18-64 (Bearer parse+validate).
Credentials default to the reference's admin/novastar (server.go:28-32) but
are configurable via config `jwt.*` / env.
"""

from __future__ import annotations

import base64
import hashlib
import hmac
import json
import time
from typing import Optional


def _b64url(data: bytes) -> str:
    return base64.urlsafe_b64encode(data).rstrip(b"=").decode()


def _b64url_decode(s: str) -> bytes:
    pad = -len(s) % 4
    return base64.urlsafe_b64decode(s + "=" * pad)


def create_token(username: str, key: str, expire_hours: int = 24) -> str:
    header = {"alg": "HS256", "typ": "JWT"}
    now = int(time.time())
    claims = {"username": username, "iat": now, "exp": now + expire_hours * 3600}
    signing_input = _b64url(json.dumps(header, separators=(",", ":")).encode()) + "." + _b64url(
        json.dumps(claims, separators=(",", ":")).encode()
    )
    sig = hmac.new(key.encode(), signing_input.encode(), hashlib.sha256).digest()
    return signing_input + "." + _b64url(sig)


def verify_token(token: str, key: str) -> Optional[dict]:
    """Return claims if valid and unexpired, else None."""
    try:
        signing_input, sig_part = token.rsplit(".", 1)
        header_part, claims_part = signing_input.split(".", 1)
        header = json.loads(_b64url_decode(header_part))
        if header.get("alg") != "HS256":
            return None
        expected = hmac.new(key.encode(), signing_input.encode(), hashlib.sha256).digest()
        if not hmac.compare_digest(expected, _b64url_decode(sig_part)):
            return None
        claims = json.loads(_b64url_decode(claims_part))
        if claims.get("exp", 0) < time.time():
            return None
        return claims
    except Exception:  # noqa: BLE001 — any malformed token is simply invalid
        return None
