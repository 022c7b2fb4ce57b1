"""Mail service + password recovery.

Reference parity: ``vantage6-server/.../mail_service.py`` + the ``recover``
resource. No SMTP egress exists in this deployment, so the
transport is a pluggable callable (default: structured log — operators wire
a relay); the recovery-token flow matches the reference: request a
time-limited signed token by email/username, then reset with it.
"""
from __future__ import annotations

import logging
from typing import Callable, Optional

from ..common import jwt as jwtlib

log = logging.getLogger(__name__)


class MailService:
    def __init__(self, send_fn: Optional[Callable[[str, str, str], None]] = None):
        self.send_fn = send_fn or self._log_mail
        self.outbox: list[dict] = []  # kept for tests/ops inspection

    def _log_mail(self, to: str, subject: str, body: str) -> None:
        log.info("MAIL to=%s subject=%r", to, subject)

    def send(self, to: str, subject: str, body: str) -> None:
        self.outbox.append({"to": to, "subject": subject, "body": body})
        self.send_fn(to, subject, body)


def request_password_recovery(core, mail: MailService, username: str) -> None:
    """POST /api/recover/lost — always succeeds (no user enumeration)."""
    from .model import User

    user = core.session.query(User).filter_by(username=username).first()
    if user is None:
        return
    token = jwtlib.encode({"sub": user.id, "type": "recovery"},
                          core.jwt_secret, expires_in=3600)
    mail.send(user.email or user.username, "password recovery",
              f"reset token (valid 1h): {token}")


def reset_password(core, token: str, new_password: str) -> None:
    from .model import User
    from .permission import PermissionError_

    payload = jwtlib.decode(token, core.jwt_secret)
    if payload.get("type") != "recovery":
        raise PermissionError_("not a recovery token", 401)
    user = core.session.get(User, payload["sub"])
    if user is None:
        raise PermissionError_("unknown user", 401)
    core.validate_password(new_password)
    user.set_password(new_password)
    user.failed_login_attempts = 0
    core.session.commit()


def request_mfa_reset(core, mail: MailService, username: str) -> None:
    """POST /api/recover/2fa/lost — emails an MFA-reset token (reference:
    2FA reset flow). Always succeeds: no user enumeration."""
    from .model import User

    user = core.session.query(User).filter_by(username=username).first()
    if user is None:
        return
    token = jwtlib.encode({"sub": user.id, "purpose": "mfa_reset"},
                          core.jwt_secret, expires_in=3600)
    mail.send(user.email or user.username, "mfa reset",
              f"mfa reset token (valid 1h): {token}")
