"""Closed-loop HTTP load generator for the /chat endpoint.

Deploy-envelope evidence (reference deploy/kubernetes/hpa.yaml targets:
50 RPS/pod, p95 <= 2 s): drives C concurrent clients against a running
server over real sockets — so uvicorn, the auth/rate-limit middleware, the
dynamic batcher and the engines are all in the measured path, unlike
TestClient-based tests.

Usage:
    # terminal 1 (mock mode on CPU, or real engines on a GPU box):
    MOCK_COMPUTE=true DISABLE_AUTH=true python -m sentio_amd.serving.app
    # terminal 2:
    python scripts/load_test.py --url http://127.0.0.1:8000 \
        --clients 16 --requests 200 --seed-docs 50
"""

from __future__ import annotations

import argparse
import json
import statistics
import sys
import time
from concurrent.futures import ThreadPoolExecutor


def _client(url: str):
    import httpx

    return httpx.Client(base_url=url, timeout=60.0)


def seed(url: str, n: int) -> None:
    with _client(url) as c:
        for i in range(n):
            r = c.post("/embed", json={
                "content": f"corpus document {i}: gpus, retrieval, topic {i % 7}",
                "metadata": {"i": i},
            })
            r.raise_for_status()


def run(url: str, clients: int, total: int, stream_frac: float = 0.0) -> dict:
    import threading

    latencies: list[float] = []
    errors = 0
    throttled = 0
    tls = threading.local()   # persistent connection per worker thread

    n_stream = int(total * stream_frac)

    def one(i: int) -> float | str | None:
        c = getattr(tls, "client", None)
        if c is None:
            c = tls.client = _client(url)
        t0 = time.perf_counter()
        if i < n_stream:
            # SSE stream: consume every delta; latency = full stream drain
            body = ""
            with c.stream("POST", "/chat/stream",
                          json={"question": f"what about topic {i % 7}?"}) as r:
                if r.status_code == 429:
                    return "throttled"
                if r.status_code != 200:
                    return None
                for line in r.iter_lines():
                    if line.startswith("data: "):
                        body += line[6:]
            dt = time.perf_counter() - t0
            return dt if body else None
        r = c.post("/chat", json={"question": f"what about topic {i % 7}?"})
        dt = time.perf_counter() - t0
        if r.status_code == 429:      # server-side rate limit, not a failure
            return "throttled"
        if r.status_code != 200 or not r.json().get("answer"):
            return None
        return dt

    import random

    order = list(range(total))
    random.Random(7).shuffle(order)   # interleave stream/non-stream arrivals
    t_start = time.perf_counter()
    with ThreadPoolExecutor(max_workers=clients) as ex:
        for dt in ex.map(one, order):
            if dt is None:
                errors += 1
            elif dt == "throttled":
                throttled += 1
            else:
                latencies.append(dt)
    wall = time.perf_counter() - t_start

    latencies.sort()
    pct = (lambda p: latencies[min(int(p * len(latencies)), len(latencies) - 1)]
           if latencies else float("nan"))
    served = len(latencies)
    return {
        "requests": total,
        "clients": clients,
        "stream_requests": n_stream,
        "served": served,
        "throttled": throttled,
        "errors": errors,
        "wall_s": round(wall, 3),
        "rps": round(served / wall, 2),
        "p50_s": round(pct(0.50), 3),
        "p95_s": round(pct(0.95), 3),
        "p99_s": round(pct(0.99), 3),
        "envelope_50rps": served / wall >= 50.0,
        "envelope_p95_2s": bool(latencies) and pct(0.95) <= 2.0,
    }


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--url", default="http://127.0.0.1:8000")
    ap.add_argument("--clients", type=int, default=16)
    ap.add_argument("--requests", type=int, default=200)
    ap.add_argument("--seed-docs", type=int, default=0)
    ap.add_argument("--stream-frac", type=float, default=0.0,
                    help="fraction of requests sent to /chat/stream (SSE)")
    args = ap.parse_args()

    with _client(args.url) as c:
        c.get("/health").raise_for_status()
    if args.seed_docs:
        seed(args.url, args.seed_docs)
    result = run(args.url, args.clients, args.requests, args.stream_frac)
    print(json.dumps(result))
    return 0 if result["errors"] == 0 else 1


if __name__ == "__main__":
    sys.exit(main())
