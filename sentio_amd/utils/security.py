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
