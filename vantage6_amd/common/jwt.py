"""Minimal HS256 JWT implementation (no ``pyjwt`` wheel in this image).

Used by the server to mint user / node / container tokens, mirroring
``vantage6-server``'s flask-jwt-extended usage (token
resources). Standard JWT: ``b64url(header).b64url(payload).b64url(sig)``
with HMAC-SHA256.
"""
from __future__ import annotations

import base64
import hashlib
import hmac
import json
import time
from typing import Any


class JWTError(Exception):
    pass


def _b64url(data: bytes) -> str:
    return base64.urlsafe_b64encode(data).rstrip(b"=").decode("ascii")


def _unb64url(data: str) -> bytes:
    pad = "=" * (-len(data) % 4)
    return base64.urlsafe_b64decode(data + pad)


def encode(payload: dict[str, Any], secret: str, expires_in: float | None = 21600) -> str:
    header = {"alg": "HS256", "typ": "JWT"}
    payload = dict(payload)
    payload.setdefault("iat", int(time.time()))
    if expires_in is not None:
        payload.setdefault("exp", int(time.time() + expires_in))
    signing = _b64url(json.dumps(header, separators=(",", ":")).encode()) + "." + _b64url(
        json.dumps(payload, separators=(",", ":")).encode()
    )
    sig = hmac.new(secret.encode(), signing.encode(), hashlib.sha256).digest()
    return signing + "." + _b64url(sig)


def decode(token: str, secret: str, verify_exp: bool = True) -> dict[str, Any]:
    # every parse failure maps to JWTError: callers catch exactly that to
    # answer 401 — a stray binascii/json error from garbage input must not
    # surface as a 500
    try:
        h, p, s = token.split(".")
        signing = f"{h}.{p}"
        expect = hmac.new(secret.encode(), signing.encode(),
                          hashlib.sha256).digest()
        if not hmac.compare_digest(expect, _unb64url(s)):
            raise JWTError("bad signature")
        payload = json.loads(_unb64url(p))
    except JWTError:
        raise
    except Exception as exc:
        raise JWTError("malformed token") from exc
    if not isinstance(payload, dict):
        raise JWTError("malformed token")
    if verify_exp and "exp" in payload and payload["exp"] < time.time():
        raise JWTError("token expired")
    return payload
