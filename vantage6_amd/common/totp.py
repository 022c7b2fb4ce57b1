"""TOTP (RFC 6238) two-factor authentication — no ``pyotp`` wheel in image.

Reference parity: vantage6 4.x user MFA (``resource/token.py`` + pyotp,
SURVEY.md §3.2): users with MFA enabled must supply a 6-digit TOTP code at
login; the secret is provisioned once (otpauth:// URI for authenticator
apps).
"""
from __future__ import annotations

import base64
import hashlib
import hmac
import secrets
import struct
import time


def generate_secret() -> str:
    return base64.b32encode(secrets.token_bytes(20)).decode("ascii")


def totp_code(secret: str, at: float | None = None, period: int = 30,
              digits: int = 6) -> str:
    key = base64.b32decode(secret)
    counter = int((at if at is not None else time.time()) // period)
    msg = struct.pack(">Q", counter)
    mac = hmac.new(key, msg, hashlib.sha1).digest()
    offset = mac[-1] & 0x0F
    code = (struct.unpack(">I", mac[offset : offset + 4])[0] & 0x7FFFFFFF) % 10**digits
    return str(code).zfill(digits)


def verify_code(secret: str, code: str, at: float | None = None,
                window: int = 1) -> bool:
    """Accept codes from ±window periods (clock skew). A corrupted
    stored secret (invalid base32) verifies as False instead of raising —
    login must answer 401, never 500."""
    now = at if at is not None else time.time()
    try:
        for w in range(-window, window + 1):
            if hmac.compare_digest(totp_code(secret, now + w * 30), str(code)):
                return True
    except Exception:  # noqa: BLE001 — undecodable secret == no valid code
        return False
    return False


def provisioning_uri(secret: str, username: str, issuer: str = "vantage6_amd") -> str:
    from urllib.parse import quote

    u, i = quote(username, safe=""), quote(issuer, safe="")
    return (f"otpauth://totp/{i}:{u}?secret={secret}"
            f"&issuer={i}&algorithm=SHA1&digits=6&period=30")
