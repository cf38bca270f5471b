"""Token auth with scopes and roles
(reference src/utils/auth.py:30-482 capability: access/refresh tokens,
scopes CHAT/EMBED/METRICS/DELETE/ADMIN, roles, API keys, audit log,
FastAPI dependency guards).  The reference used python-jose JWTs; that
package is not in this image, so tokens are stdlib HMAC-SHA256-signed
(same claims model, same verification semantics)."""

from __future__ import annotations

import base64
import enum
import hashlib
import hmac
import json
import secrets
import time
from dataclasses import dataclass, field
from typing import Any


class AuthScope(enum.Enum):
    CHAT = "chat"
    EMBED = "embed"
    METRICS = "metrics"
    DELETE = "delete"
    ADMIN = "admin"


class UserRole(enum.Enum):
    READER = "reader"
    WRITER = "writer"
    ADMIN = "admin"


ROLE_SCOPES = {
    UserRole.READER: {AuthScope.CHAT, AuthScope.METRICS},
    UserRole.WRITER: {AuthScope.CHAT, AuthScope.EMBED, AuthScope.METRICS},
    UserRole.ADMIN: set(AuthScope),
}


@dataclass
class TokenData:
    subject: str
    scopes: set[AuthScope]
    expires_at: float
    kind: str = "access"


class AuthError(Exception):
    pass


@dataclass
class AuthManager:
    secret: str = "sentio-dev-secret"
    token_ttl_s: int = 3600
    refresh_ttl_s: int = 86400
    api_keys: dict[str, UserRole] = field(default_factory=dict)
    audit_log: list[dict[str, Any]] = field(default_factory=list)
    sessions: dict[str, dict[str, Any]] = field(default_factory=dict)

    def _sign(self, payload: bytes) -> str:
        return base64.urlsafe_b64encode(
            hmac.new(self.secret.encode(), payload, hashlib.sha256).digest()
        ).decode().rstrip("=")

    def issue_token(self, subject: str, role: UserRole = UserRole.READER,
                    kind: str = "access") -> str:
        ttl = self.token_ttl_s if kind == "access" else self.refresh_ttl_s
        claims = {
            "sub": subject,
            "scopes": sorted(s.value for s in ROLE_SCOPES[role]),
            "exp": time.time() + ttl,
            "kind": kind,
            "jti": secrets.token_hex(8),
        }
        body = base64.urlsafe_b64encode(json.dumps(claims).encode()).decode().rstrip("=")
        sig = self._sign(body.encode())
        self._audit("issue", subject, kind)
        return f"{body}.{sig}"

    def verify_token(self, token: str) -> TokenData:
        try:
            body, sig = token.rsplit(".", 1)
        except ValueError:
            raise AuthError("malformed token")
        try:
            ok = hmac.compare_digest(self._sign(body.encode()), sig)
        except (TypeError, UnicodeError):
            # non-ASCII signature chars: invalid by construction — must be
            # an auth failure (401), never an unhandled 500
            raise AuthError("malformed token")
        if not ok:
            raise AuthError("bad signature")
        pad = "=" * (-len(body) % 4)
        try:
            claims = json.loads(base64.urlsafe_b64decode(body + pad))
        except (ValueError, UnicodeError):
            raise AuthError("malformed token body")
        if time.time() > float(claims.get("exp", 0)):
            raise AuthError("token expired")
        return TokenData(
            subject=str(claims.get("sub", "")),
            scopes={AuthScope(s) for s in claims.get("scopes", [])},
            expires_at=float(claims["exp"]),
            kind=str(claims.get("kind", "access")),
        )

    def refresh(self, refresh_token: str, role: UserRole = UserRole.READER) -> str:
        data = self.verify_token(refresh_token)
        if data.kind != "refresh":
            raise AuthError("not a refresh token")
        return self.issue_token(data.subject, role)

    # API keys
    def create_api_key(self, role: UserRole = UserRole.READER) -> str:
        key = "sk-" + secrets.token_urlsafe(24)
        self.api_keys[hashlib.sha256(key.encode()).hexdigest()] = role
        return key

    def verify_api_key(self, key: str) -> UserRole:
        role = self.api_keys.get(hashlib.sha256(key.encode()).hexdigest())
        if role is None:
            raise AuthError("unknown api key")
        return role

    # sessions (reference auth.py session-store capability): server-side
    # revocable handles, unlike the stateless HMAC tokens above
    def create_session(self, subject: str, role: UserRole = UserRole.READER,
                       ttl_s: float | None = None) -> str:
        sid = "sess-" + secrets.token_urlsafe(18)
        self.sessions[sid] = {
            "subject": subject, "role": role,
            "expires": time.time() + (ttl_s or self.token_ttl_s),
        }
        self._audit("session.create", subject, sid)
        return sid

    def validate_session(self, session_id: str) -> dict:
        s = self.sessions.get(session_id)
        if s is None:
            raise AuthError("unknown session")
        if s["expires"] < time.time():
            self.sessions.pop(session_id, None)
            raise AuthError("session expired")
        return s

    def revoke_session(self, session_id: str) -> bool:
        s = self.sessions.pop(session_id, None)
        if s is not None:
            self._audit("session.revoke", s["subject"], session_id)
        return s is not None

    def require_scopes(self, token: str, *scopes: AuthScope) -> TokenData:
        data = self.verify_token(token)
        missing = set(scopes) - data.scopes
        if missing:
            raise AuthError(f"missing scopes: {sorted(s.value for s in missing)}")
        return data

    def _audit(self, action: str, subject: str, detail: str = "") -> None:
        self.audit_log.append(
            {"t": time.time(), "action": action, "subject": subject, "detail": detail}
        )
        if len(self.audit_log) > 10000:
            del self.audit_log[:5000]


auth_manager = AuthManager()
