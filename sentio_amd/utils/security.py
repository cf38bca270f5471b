"""Input validation, security headers, log sanitization
(reference src/utils/security.py:23-254 semantics: max lengths 2000 query /
50000 document / 1000 metadata values, injection-pattern screens,
HTML-escape; SecurityHeaders CSP/HSTS set; LogSanitizer secret redaction)."""

from __future__ import annotations

import html
import logging
import re


class ValidationError(ValueError):
    pass


class InputValidator:
    MAX_QUERY_LEN = 2000
    MAX_DOCUMENT_LEN = 50000
    MAX_METADATA_VALUE_LEN = 1000
    MAX_METADATA_KEYS = 50

    _INJECTION_PATTERNS = [
        re.compile(r"(?i)\b(drop|delete|truncate|insert|update)\s+(table|from|into)\b"),
        re.compile(r"(?i)<\s*script[^>]*>"),
        re.compile(r"(?i)javascript\s*:"),
        re.compile(r"(?i)on(error|load|click)\s*="),
        re.compile(r"[;&|`$]\s*(rm|cat|wget|curl|bash|sh)\b"),
    ]

    @classmethod
    def validate_query(cls, query: str) -> str:
        if not isinstance(query, str) or not query.strip():
            raise ValidationError("query must be a non-empty string")
        if len(query) > cls.MAX_QUERY_LEN:
            raise ValidationError(f"query exceeds {cls.MAX_QUERY_LEN} characters")
        for pat in cls._INJECTION_PATTERNS:
            if pat.search(query):
                raise ValidationError("query contains disallowed pattern")
        return query.strip()

    @classmethod
    def validate_document_content(cls, content: str) -> str:
        if not isinstance(content, str) or not content.strip():
            raise ValidationError("document content must be non-empty")
        if len(content) > cls.MAX_DOCUMENT_LEN:
            raise ValidationError(f"document exceeds {cls.MAX_DOCUMENT_LEN} characters")
        return content

    @classmethod
    def validate_metadata(cls, metadata: dict | None) -> dict:
        if metadata is None:
            return {}
        if not isinstance(metadata, dict):
            raise ValidationError("metadata must be an object")
        if len(metadata) > cls.MAX_METADATA_KEYS:
            raise ValidationError("too many metadata keys")
        out = {}
        for k, v in metadata.items():
            ks = str(k)[:128]
            if isinstance(v, str):
                if len(v) > cls.MAX_METADATA_VALUE_LEN:
                    raise ValidationError(f"metadata value for '{ks}' too long")
                out[ks] = html.escape(v, quote=False) if "<" in v else v
            elif isinstance(v, (int, float, bool)) or v is None:
                out[ks] = v
            else:
                out[ks] = str(v)[: cls.MAX_METADATA_VALUE_LEN]
        return out


class SecurityHeaders:
    HEADERS = {
        "X-Content-Type-Options": "nosniff",
        "X-Frame-Options": "DENY",
        "X-XSS-Protection": "1; mode=block",
        "Referrer-Policy": "strict-origin-when-cross-origin",
        "Content-Security-Policy": "default-src 'self'",
        "Strict-Transport-Security": "max-age=31536000; includeSubDomains",
    }

    @classmethod
    def apply(cls, response) -> None:
        for k, v in cls.HEADERS.items():
            response.headers[k] = v


_SECRET_PATTERNS = [
    re.compile(r"(?i)(api[_-]?key|token|secret|password|authorization)"
               r"([\"':=\s]+)([^\s\"',;&]+)"),
    re.compile(r"(?i)bearer\s+([a-z0-9._\-]+)"),
]


class LogSanitizer:
    @staticmethod
    def sanitize(text: str) -> str:
        for pat in _SECRET_PATTERNS:
            text = pat.sub(lambda m: m.group(0).replace(m.group(m.lastindex), "***"), text)
        return text


class SanitizingFilter(logging.Filter):
    def filter(self, record: logging.LogRecord) -> bool:
        try:
            record.msg = LogSanitizer.sanitize(str(record.msg))
        except Exception:
            pass
        return True


def setup_log_sanitization() -> None:
    logging.getLogger().addFilter(SanitizingFilter())


# ---- CSRF tokens (reference security.py:331-390 region capability) ----

class CSRFProtection:
    """HMAC-signed, session-bound, expiring CSRF tokens.

    Stateless: token = "<session>.<expiry>.<hmac(secret, session|expiry)>";
    verification recomputes the MAC, so no server-side token store."""

    def __init__(self, secret: str, ttl_s: int = 3600):
        self._secret = secret.encode()
        self.ttl_s = ttl_s

    def _mac(self, session_id: str, expiry: int) -> str:
        import hashlib
        import hmac as _hmac

        msg = f"{session_id}|{expiry}".encode()
        return _hmac.new(self._secret, msg, hashlib.sha256).hexdigest()[:32]

    def generate(self, session_id: str) -> str:
        import time

        expiry = int(time.time()) + self.ttl_s
        return f"{session_id}.{expiry}.{self._mac(session_id, expiry)}"

    def verify(self, token: str, session_id: str) -> bool:
        import hmac as _hmac
        import time

        try:
            sess, expiry_s, mac = token.rsplit(".", 2)
            expiry = int(expiry_s)
        except (ValueError, AttributeError):
            return False
        if sess != session_id or expiry < time.time():
            return False
        return _hmac.compare_digest(mac, self._mac(sess, expiry))


# ---- client IP validation (reference security.py IP-validation capability) ----

def validate_client_ip(ip: str,
                       allow: list[str] | None = None,
                       block: list[str] | None = None) -> bool:
    """True iff `ip` parses, is not in any blocked CIDR, and (when an
    allowlist is given) is inside at least one allowed CIDR."""
    import ipaddress

    try:
        addr = ipaddress.ip_address(ip)
    except ValueError:
        return False
    for cidr in block or []:
        if addr in ipaddress.ip_network(cidr, strict=False):
            return False
    if allow:
        return any(addr in ipaddress.ip_network(c, strict=False)
                   for c in allow)
    return True


# ---- adaptive rate limits (reference security.py:331-560 capability) ----

class AdaptiveRateLimit:
    """Per-endpoint request budget that tightens under error pressure.

    effective = base · f(error_rate): full budget while healthy, linearly
    down to `floor_fraction` of it as the recent error rate climbs to
    `max_error_rate`.  The serving limiter polls `current_limit()`; errors
    and successes are reported by the caller (advisory, like the
    reference's — the static limiter stays the default)."""

    def __init__(self, base_per_min: int, floor_fraction: float = 0.2,
                 max_error_rate: float = 0.5, window: int = 100):
        from collections import deque

        self.base_per_min = base_per_min
        self.floor_fraction = floor_fraction
        self.max_error_rate = max_error_rate
        self._events = deque(maxlen=window)

    def record(self, ok: bool) -> None:
        self._events.append(bool(ok))

    @property
    def error_rate(self) -> float:
        if not self._events:
            return 0.0
        return 1.0 - (sum(self._events) / len(self._events))

    def current_limit(self) -> int:
        pressure = min(self.error_rate / self.max_error_rate, 1.0)
        frac = 1.0 - (1.0 - self.floor_fraction) * pressure
        return max(1, round(self.base_per_min * frac))
