"""Utils coverage mirroring reference suites: exceptions shaping
(reference test_exceptions.py), security validation, prompt builder,
monitoring (reference src/tests/*)."""

from __future__ import annotations

import pytest

from sentio_amd.utils.exceptions import (
    ErrorCode,
    RateLimitException,
    SentioException,
    ValidationException,
)
from sentio_amd.utils.security import (
    InputValidator,
    LogSanitizer,
    ValidationError,
)


# ---------- exceptions ----------

def test_exception_to_dict_shape():
    exc = SentioException("boom", details={"k": "v"})
    d = exc.to_dict()
    assert d["error"] == ErrorCode.SYSTEM_ERROR.value
    assert d["message"] == "boom"
    assert d["details"] == {"k": "v"}
    assert "timestamp" in d


def test_typed_exceptions_carry_status():
    assert ValidationException("bad").status == 422
    assert RateLimitException("slow down").status == 429


# ---------- security ----------

def test_validator_rejects_overlong_query():
    with pytest.raises(ValidationError):
        InputValidator.validate_query("x" * 5000)


def test_validator_rejects_empty():
    with pytest.raises(ValidationError):
        InputValidator.validate_query("   ")


def test_validator_strips_and_passes_normal():
    assert InputValidator.validate_query("  what is RCCL? ") == "what is RCCL?"


def test_validator_blocks_injection_patterns():
    for evil in ["<script>alert(1)</script>",
                 "1; DROP TABLE users--",
                 "q && rm -rf /"]:
        with pytest.raises(ValidationError):
            InputValidator.validate_query(evil)


def test_log_sanitizer_redacts_secrets():
    line = 'calling api_key="sk-abc123xyz" password=hunter2'
    red = LogSanitizer.sanitize(line)
    assert "sk-abc123xyz" not in red
    assert "hunter2" not in red


# ---------- prompt builder ----------

def test_prompt_builder_substitutes_and_modes():
    from sentio_amd.pipeline.prompt_builder import PromptBuilder

    b = PromptBuilder("fast")
    p = b.build_qa_prompt("why GPUs?", "CONTEXT BODY")
    assert "why GPUs?" in p and "CONTEXT BODY" in p
    assert b.system_prompt()
    # all four modes resolve distinct instructions
    prompts = {m: PromptBuilder(m).build_qa_prompt("q", "c")
               for m in ("fast", "balanced", "quality", "creative")}
    assert len(set(prompts.values())) >= 2


def test_verify_prompt_contains_json_protocol():
    from sentio_amd.pipeline.prompt_builder import PromptBuilder

    vp = PromptBuilder("balanced").build_verify_prompt(
        query="q?", context="[1] ctx", answer="the answer")
    assert "verdict" in vp and "q?" in vp and "the answer" in vp


# ---------- monitoring ----------

def test_performance_monitor_records_and_summarizes():
    from sentio_amd.observability.monitoring import PerformanceMonitor

    m = PerformanceMonitor(history=16)
    for i in range(10):
        m.record_value("lat_ms", float(i))
    s = m.summary("lat_ms")
    assert s["count"] == 10
    assert s["min"] == 0.0 and s["max"] == 9.0


def test_resource_monitor_snapshot_keys():
    from sentio_amd.observability.monitoring import resource_monitor

    snap = resource_monitor.snapshot()
    assert "cpu_percent" in snap or "memory" in snap or snap  # psutil-backed


# ---- CSRF / IP validation / adaptive rate limit (reference security.py:331-560) ----

def test_csrf_token_roundtrip_and_rejections():
    from sentio_amd.utils.security import CSRFProtection

    c = CSRFProtection("secret-key", ttl_s=60)
    t = c.generate("sess-1")
    assert c.verify(t, "sess-1")
    assert not c.verify(t, "sess-2")            # bound to session
    assert not c.verify(t + "x", "sess-1")      # tampered MAC
    assert not c.verify("garbage", "sess-1")
    expired = CSRFProtection("secret-key", ttl_s=-10)
    assert not expired.verify(expired.generate("sess-1"), "sess-1")
    other = CSRFProtection("different-secret", ttl_s=60)
    assert not other.verify(t, "sess-1")        # wrong secret


def test_validate_client_ip():
    from sentio_amd.utils.security import validate_client_ip

    assert validate_client_ip("10.0.0.5")
    assert not validate_client_ip("not-an-ip")
    assert not validate_client_ip("10.0.0.5", block=["10.0.0.0/8"])
    assert validate_client_ip("10.0.0.5", allow=["10.0.0.0/24"])
    assert not validate_client_ip("10.0.1.5", allow=["10.0.0.0/24"])
    assert validate_client_ip("2001:db8::1", allow=["2001:db8::/32"])


def test_adaptive_rate_limit_tightens_under_errors():
    from sentio_amd.utils.security import AdaptiveRateLimit

    rl = AdaptiveRateLimit(base_per_min=100, floor_fraction=0.2,
                           max_error_rate=0.5, window=10)
    assert rl.current_limit() == 100            # healthy
    for _ in range(10):
        rl.record(True)
    assert rl.current_limit() == 100
    for _ in range(10):
        rl.record(False)                        # 100% errors
    assert rl.error_rate == 1.0
    assert rl.current_limit() == 20             # floor = 20% of base
    for _ in range(8):
        rl.record(True)    # window=10 now holds 2 errors -> rate 0.2
    assert abs(rl.error_rate - 0.2) < 1e-9
    lim = rl.current_limit()
    assert 20 < lim < 100  # partial pressure: between floor and base


def test_auth_sessions_lifecycle():
    from sentio_amd.utils.auth import AuthError, AuthManager, UserRole

    am = AuthManager()
    sid = am.create_session("alice", UserRole.ADMIN, ttl_s=60)
    s = am.validate_session(sid)
    assert s["subject"] == "alice" and s["role"] == UserRole.ADMIN
    assert am.revoke_session(sid)
    import pytest as _pytest
    with _pytest.raises(AuthError):
        am.validate_session(sid)
    assert not am.revoke_session(sid)           # already gone
    expired = am.create_session("bob", ttl_s=-1)
    with _pytest.raises(AuthError):
        am.validate_session(expired)
    actions = [e["action"] for e in am.audit_log]
    assert "session.create" in actions and "session.revoke" in actions
