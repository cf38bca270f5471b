"""Caching, resilience, security, auth, chunking, tokenizer, verifier JSON."""

import time

import pytest

from sentio_amd.caching.manager import CacheManager
from sentio_amd.caching.memory import MemoryCache
from sentio_amd.engines.tokenizer import ByteTokenizer
from sentio_amd.ingest.chunker import ChunkingError, TextChunker
from sentio_amd.models.document import Document
from sentio_amd.pipeline.verifier import AnswerVerifier, extract_json_dict
from sentio_amd.resilience.breaker import CircuitBreaker, CircuitOpenError, CircuitState
from sentio_amd.resilience.retry import retry_with_backoff
from sentio_amd.utils.auth import AuthError, AuthManager, AuthScope, UserRole
from sentio_amd.utils.security import InputValidator, LogSanitizer, ValidationError


# ---- caching ----

def test_memory_cache_lru_ttl():
    c = MemoryCache(max_size=2, default_ttl=0.05)
    c.set("a", 1)
    c.set("b", 2)
    c.get("a")
    c.set("c", 3)  # evicts b (LRU)
    assert c.get("b") is None and c.get("a") == 1
    time.sleep(0.06)
    assert c.get("a") is None  # TTL expired
    assert c.stats()["evictions"] >= 1


def test_cache_manager_promotion():
    m = CacheManager("multi_tier", l1_size=2, l2_size=10)
    m.l2.set("k", "v")
    assert m.l1.get("k") is None
    assert m.get("k") == "v"      # promoted
    assert m.l1.get("k") == "v"


def test_cache_pattern_clear():
    c = MemoryCache()
    c.set("emb:1", 1)
    c.set("emb:2", 2)
    c.set("query:1", 3)
    assert c.clear_pattern("emb:*") == 2
    assert c.get("query:1") == 3


# ---- resilience ----

def test_breaker_opens_and_recovers():
    b = CircuitBreaker("t", failure_threshold=2, recovery_timeout=0.05,
                       success_threshold=1)

    def boom():
        raise ValueError("x")

    for _ in range(2):
        with pytest.raises(ValueError):
            b.call(boom)
    assert b.state == CircuitState.OPEN
    with pytest.raises(CircuitOpenError):
        b.call(lambda: 1)
    time.sleep(0.06)
    assert b.state == CircuitState.HALF_OPEN
    assert b.call(lambda: 42) == 42
    assert b.state == CircuitState.CLOSED


def test_retry_with_backoff():
    calls = {"n": 0}

    @retry_with_backoff(max_attempts=3, base_delay=0.001)
    def flaky():
        calls["n"] += 1
        if calls["n"] < 3:
            raise RuntimeError("nope")
        return "ok"

    assert flaky() == "ok"
    assert calls["n"] == 3


# ---- security ----

def test_input_validator():
    assert InputValidator.validate_query("  hello  ") == "hello"
    with pytest.raises(ValidationError):
        InputValidator.validate_query("")
    with pytest.raises(ValidationError):
        InputValidator.validate_query("x" * 2001)
    with pytest.raises(ValidationError):
        InputValidator.validate_query("<script>alert(1)</script>")
    with pytest.raises(ValidationError):
        InputValidator.validate_query("; rm -rf /")


def test_metadata_validation():
    out = InputValidator.validate_metadata({"a": 1, "b": "txt", "c": [1, 2]})
    assert out["a"] == 1 and isinstance(out["c"], str)
    with pytest.raises(ValidationError):
        InputValidator.validate_metadata({"k": "x" * 2000})


def test_log_sanitizer_redacts():
    s = LogSanitizer.sanitize("api_key=SECRET123 other=fine")
    assert "SECRET123" not in s


# ---- auth ----

def test_auth_token_roundtrip_and_scopes():
    am = AuthManager(secret="test")
    tok = am.issue_token("alice", UserRole.WRITER)
    data = am.verify_token(tok)
    assert data.subject == "alice"
    assert AuthScope.EMBED in data.scopes
    am.require_scopes(tok, AuthScope.CHAT)
    with pytest.raises(AuthError):
        am.require_scopes(tok, AuthScope.ADMIN)


def test_auth_rejects_tampered_and_expired():
    am = AuthManager(secret="test", token_ttl_s=-1)
    expired = am.issue_token("bob")
    with pytest.raises(AuthError):
        am.verify_token(expired)
    am2 = AuthManager(secret="test")
    tok = am2.issue_token("bob")
    with pytest.raises(AuthError):
        am2.verify_token(tok[:-2] + "zz")


def test_api_keys():
    am = AuthManager()
    key = am.create_api_key(UserRole.ADMIN)
    assert am.verify_api_key(key) == UserRole.ADMIN
    with pytest.raises(AuthError):
        am.verify_api_key("sk-not-real")


# ---- chunker ----

def test_chunker_respects_size_and_parent():
    ch = TextChunker(chunk_size=50, chunk_overlap=10)
    doc = Document(text=" ".join(f"word{i}" for i in range(100)), id="p1")
    chunks = ch.split([doc])
    assert len(chunks) > 1
    assert all(len(c.text) <= 50 for c in chunks)
    assert all(c.metadata["parent_id"] == "p1" for c in chunks)


def test_chunker_rejects_bad_overlap():
    with pytest.raises(ChunkingError):
        TextChunker(chunk_size=10, chunk_overlap=20)


def test_chunker_fixed_strategy():
    ch = TextChunker(chunk_size=10, chunk_overlap=2, strategy="fixed")
    chunks = ch.split_text("abcdefghijklmnopqrstuvwxyz")
    assert all(len(c) <= 10 for c in chunks)
    assert "".join(c[: 8] for c in chunks).startswith("abcdefgh")


# ---- tokenizer ----

def test_tokenizer_roundtrip():
    t = ByteTokenizer()
    for text in ("hello world", "ünïcødé ✓", ""):
        ids = t.encode(text)
        assert t.decode(ids) == text


def test_tokenizer_batch_padding():
    t = ByteTokenizer()
    padded, lens = t.encode_batch(["ab", "abcdef"], max_len=32)
    assert len(padded[0]) == len(padded[1])
    assert lens == [3, 7]  # bos + bytes


# ---- verifier JSON extraction ----

def test_extract_json_variants():
    assert extract_json_dict('{"verdict": "pass"}')["verdict"] == "pass"
    assert extract_json_dict('noise {"verdict": "warn",} more')["verdict"] == "warn"
    assert extract_json_dict('```json\n{"verdict": "fail"}\n```')["verdict"] == "fail"
    assert extract_json_dict('{"ok": True}')["ok"] is True
    assert extract_json_dict("no json here") is None


def test_verifier_normalizes_bad_output():
    class Gen:
        def generate(self, prompts, **kw):
            return ["absolutely not json"]

    v = AnswerVerifier(Gen())
    out = v.verify("q", "ctx", "ans")
    assert out["verdict"] == "warn"
    assert out["notes"] == ["invalid_json"]


def test_verifier_parses_fail_with_revision():
    class Gen:
        def generate(self, prompts, **kw):
            return ['{"verdict": "fail", "citations_ok": false, '
                    '"notes": ["bad"], "revised_answer": "fixed [1]"}']

    v = AnswerVerifier(Gen())
    out = v.verify("q", "ctx", "ans")
    assert out["verdict"] == "fail"
    assert out["revised_answer"] == "fixed [1]"


def test_rate_limit_settings_from_env(monkeypatch):
    from sentio_amd.config import Settings

    monkeypatch.setenv("RATE_LIMIT_CHAT_PER_MIN", "555")
    monkeypatch.setenv("RATE_LIMIT_EMBED_PER_MIN", "44")
    s = Settings.from_env()
    assert s.rate_limit_chat_per_min == 555
    assert s.rate_limit_embed_per_min == 44


def test_r2_serving_flags_from_env(monkeypatch):
    """Round-2 batching knobs flow through the env config layer."""
    from sentio_amd.config import Settings

    monkeypatch.setenv("CONTINUOUS_BATCHING", "false")
    monkeypatch.setenv("DYNAMIC_BATCHING", "true")
    monkeypatch.setenv("MAX_BATCH_SIZE", "48")
    s = Settings.from_env()
    assert s.continuous_batching is False
    assert s.dynamic_batching is True
    assert s.max_batch_size == 48
    monkeypatch.setenv("CONTINUOUS_BATCHING", "true")
    assert Settings.from_env().continuous_batching is True


def test_frontend_selection_by_flags(monkeypatch):
    """Container picks the continuous frontend for real engines, the wave
    batcher when continuous is off, and never wraps mocks in the slot
    loop (mock engines lack slot sessions)."""
    from sentio_amd.config import Settings
    from sentio_amd.serving.container import ServiceContainer

    s = Settings()
    s.mock_compute = True
    s.device = "cpu"
    s.dynamic_batching = True
    s.continuous_batching = True
    c = ServiceContainer(s)
    fe = c.generator_frontend()
    from sentio_amd.serving.batcher import BatchedGenerator

    assert isinstance(fe, BatchedGenerator)   # mock -> wave batcher

    s2 = Settings()
    s2.mock_compute = True
    s2.device = "cpu"
    s2.dynamic_batching = False
    c2 = ServiceContainer(s2)
    fe2 = c2.generator_frontend()
    assert not hasattr(fe2, "batcher")        # raw engine passthrough
