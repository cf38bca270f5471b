"""FastAPI serving surface (reference src/api/app.py:81-665 capability):
endpoints /chat, /chat/stream, /embed, /health, /health/detailed,
/health/ready, /health/live, /clear, /info, /metrics, /metrics/performance;
sliding-window rate limiting (100/min chat, 10/min embed), security headers
middleware, pydantic request validation, structured error handlers."""

from __future__ import annotations

import logging
import time
from collections import defaultdict, deque
from contextlib import asynccontextmanager
from fastapi import FastAPI, HTTPException, Request
from fastapi.responses import (
    HTMLResponse,
    JSONResponse,
    PlainTextResponse,
    StreamingResponse,
)
from pydantic import BaseModel, Field, field_validator

from sentio_amd.config import Settings
from sentio_amd.models.document import Document
from sentio_amd.observability.metrics import metrics_collector
from sentio_amd.observability.monitoring import performance_monitor, resource_monitor
from sentio_amd.serving.container import ServiceContainer, get_container
from sentio_amd.serving.handlers import ChatHandler, HealthHandler
from sentio_amd.utils.exceptions import SentioException
from sentio_amd.utils.security import (
    InputValidator,
    SecurityHeaders,
    ValidationError,
    setup_log_sanitization,
)

logger = logging.getLogger(__name__)


# ---------------- rate limiting (reference app.py:81-101,259-281) ----------------

class RateLimiter:
    def __init__(self, chat_per_min: int = 100, embed_per_min: int = 10):
        self.limits = {"/embed": embed_per_min, "default": chat_per_min}
        self._hits: dict[tuple[str, str], deque] = defaultdict(deque)

    def allow(self, client: str, path: str) -> bool:
        limit = self.limits.get(path, self.limits["default"])
        now = time.time()
        dq = self._hits[(client, path)]
        while dq and now - dq[0] > 60.0:
            dq.popleft()
        if len(dq) >= limit:
            return False
        dq.append(now)
        return True


# ---------------- request/response models (reference app.py:118-203) ----------------

class ChatRequest(BaseModel):
    question: str = Field(..., min_length=1, max_length=2000)
    history: list[dict[str, str]] | None = Field(default_factory=list)
    top_k: int | None = Field(default=3, ge=1, le=20)
    temperature: float | None = Field(default=0.7, ge=0.0, le=2.0)

    @field_validator("question")
    @classmethod
    def _validate_question(cls, v: str) -> str:
        try:
            return InputValidator.validate_query(v)
        except ValidationError as exc:
            raise ValueError(str(exc))


class SourceModel(BaseModel):
    text: str
    source: str
    score: float = Field(..., ge=0.0, le=1.0)
    metadata: dict | None = None


class ChatResponse(BaseModel):
    answer: str
    sources: list[SourceModel]
    metadata: dict | None = None


class EmbedRequest(BaseModel):
    id: int | str | None = None
    content: str = Field(..., min_length=1, max_length=50000)
    metadata: dict | None = None


class HealthResponse(BaseModel):
    status: str
    timestamp: float
    version: str
    services: dict[str, str]


# ---------------- app factory ----------------

def create_app(settings: Settings | None = None,
               container: ServiceContainer | None = None) -> FastAPI:
    setup_log_sanitization()
    container = container or get_container(settings)
    s = container.settings
    limiter = RateLimiter(s.rate_limit_chat_per_min, s.rate_limit_embed_per_min)
    chat_handler = ChatHandler(container)
    health_handler = HealthHandler(container)

    @asynccontextmanager
    async def lifespan(app: FastAPI):
        container.initialize_all()
        hc = container.health_checker()
        hc.run_checks()
        hc.start()
        from sentio_amd.observability import tracing as _tracing

        _tracing.start_otlp_exporter()   # no-op without OTLP_ENDPOINT
        yield
        _tracing.stop_otlp_exporter()
        hc.stop()
        # drain the batching workers so SIGTERM shutdown (k8s) is clean:
        # in-flight requests were already completed by uvicorn's drain
        fe = container._cache.get("generator_frontend")
        if fe is not None and hasattr(fe, "batcher"):
            fe.batcher.stop()
        for eng_key in ("encoder", "reranker"):
            micro = getattr(container._cache.get(eng_key), "micro", None)
            if micro is not None:
                micro.stop()

    app = FastAPI(title="sentio-amd", version=__import__("sentio_amd").__version__,
                  lifespan=lifespan)
    app.state.container = container

    # CORS open only in no-auth dev mode (reference app.py:370-377)
    if not s.auth_enabled:
        from fastapi.middleware.cors import CORSMiddleware

        app.add_middleware(CORSMiddleware, allow_origins=["*"],
                           allow_methods=["*"], allow_headers=["*"])

    # scope requirements when auth is enabled (reference auth.py:444-470
    # require_scopes dependency guards; DISABLE_AUTH=true skips, as the
    # reference's CORS-open dev mode did)
    from sentio_amd.utils.auth import AuthError, AuthScope

    _SCOPE_MAP = {
        ("POST", "/chat"): AuthScope.CHAT,
        ("POST", "/chat/stream"): AuthScope.CHAT,
        ("POST", "/embed"): AuthScope.EMBED,
        ("POST", "/clear"): AuthScope.DELETE,
        ("GET", "/metrics"): AuthScope.METRICS,
        ("GET", "/metrics/performance"): AuthScope.METRICS,
    }

    def _check_auth(request: Request) -> JSONResponse | None:
        scope = _SCOPE_MAP.get((request.method, request.url.path))
        if scope is None:
            return None                      # health/info/ui stay open
        header = request.headers.get("Authorization", "")
        mgr = container.auth_manager()
        try:
            if header.startswith("Bearer "):
                mgr.require_scopes(header[7:].strip(), scope)
                return None
            api_key = request.headers.get("X-API-Key", "")
            if api_key:
                from sentio_amd.utils.auth import ROLE_SCOPES

                role = mgr.verify_api_key(api_key)
                if scope in ROLE_SCOPES[role]:
                    return None
                raise AuthError(f"api key lacks scope {scope.value}")
            raise AuthError("missing credentials")
        except AuthError as exc:
            return JSONResponse(status_code=401,
                                content={"error": "AUTH_ERROR",
                                         "message": str(exc)})

    @app.middleware("http")
    async def security_and_rate_limit(request: Request, call_next):
        client = request.client.host if request.client else "unknown"
        path = request.url.path
        if s.auth_enabled:
            denied = _check_auth(request)
            if denied is not None:
                SecurityHeaders.apply(denied)
                return denied
        if request.method == "POST" and not limiter.allow(client, path):
            return JSONResponse(
                status_code=429,
                content={"error": "RATE_LIMITED",
                         "message": "rate limit exceeded, retry later"},
            )
        response = await call_next(request)
        SecurityHeaders.apply(response)
        return response

    @app.exception_handler(SentioException)
    async def sentio_exc_handler(request: Request, exc: SentioException):
        return JSONResponse(status_code=exc.status, content=exc.to_dict())

    # ---------------- endpoints ----------------

    @app.get("/health", response_model=HealthResponse)
    async def health():
        return health_handler.basic()

    @app.get("/health/detailed")
    async def health_detailed():
        return health_handler.detailed()

    @app.get("/health/ready")
    async def health_ready():
        if health_handler.ready():
            return {"status": "ready"}
        raise HTTPException(status_code=503, detail="not ready")

    @app.get("/health/live")
    async def health_live():
        return {"status": "alive"}

    @app.post("/chat", response_model=ChatResponse)
    async def chat(req: ChatRequest):
        with metrics_collector.track_request("/chat"):
            t0 = time.perf_counter()
            result = await _run_sync(chat_handler.process, req.question,
                                     req.top_k, req.temperature, req.history)
            performance_monitor.record_value(
                "chat_latency_ms", (time.perf_counter() - t0) * 1e3)
            return result

    @app.post("/chat/stream")
    async def chat_stream(req: ChatRequest):
        """SSE token streaming (reference streamed via OpenAI SSE pass-through;
        here tokens stream straight off the decode loop)."""
        container.initialize_all()

        def gen():
            from sentio_amd.pipeline.context import prepare_context
            from sentio_amd.pipeline.prompt_builder import PromptBuilder
            from sentio_amd.pipeline.state import create_initial_state

            state = create_initial_state(req.question,
                                         {"user_top_k": req.top_k})
            # retrieval stages only, then stream generation
            pipe = container.pipeline()
            for name, fn in pipe.stages:
                if name == "generator":
                    break
                state = fn(state)
            docs = state.get("selected_documents") or []
            builder = PromptBuilder(s.generation_mode)
            prompt = builder.system_prompt() + "\n\n" + builder.build_qa_prompt(
                req.question, prepare_context(docs))
            # generator_frontend, NOT the raw engine: the raw stream()
            # holds the engine's generation lock between SSE yields, so an
            # abandoned/slow client (e.g. read timeout) leaks the lock and
            # wedges every later generation.  The batched frontend streams
            # from a per-request token queue instead.
            for delta in container.generator_frontend().stream(
                    prompt, max_new_tokens=s.llm_max_tokens,
                    temperature=req.temperature or 0.3):
                yield f"data: {delta}\n\n"
            yield "data: [DONE]\n\n"

        return StreamingResponse(gen(), media_type="text/event-stream")

    @app.post("/embed")
    async def embed(req: EmbedRequest):
        with metrics_collector.track_request("/embed"):
            try:
                content = InputValidator.validate_document_content(req.content)
                metadata = InputValidator.validate_metadata(req.metadata)
            except ValidationError as exc:
                raise HTTPException(status_code=422, detail=str(exc))
            doc = Document(text=content, metadata=metadata,
                           id=str(req.id) if req.id is not None else None or None)
            if req.id is not None:
                doc.id = str(req.id)
            result = await _run_sync(container.ingestor().ingest_document, doc)
            result.pop("status", None)
            return {"status": "success", "document_id": doc.id, **result}

    @app.post("/clear")
    async def clear():
        await _run_sync(container.clear_indexes)
        return {"status": "cleared"}

    @app.get("/info")
    async def info():
        cfg = {k: v for k, v in vars(s).items() if not k.startswith("_")
               and k != "auth_secret"}
        return {
            "name": "sentio-amd",
            "version": __import__("sentio_amd").__version__,
            "config": cfg,
            "device": container.device,
            "index": container.dense_index().stats(),
            "resources": resource_monitor.snapshot(),
        }

    @app.get("/metrics")
    async def metrics():
        metrics_collector.record_gpu()
        return PlainTextResponse(metrics_collector.prometheus_text())

    @app.get("/metrics/performance")
    async def metrics_performance():
        from sentio_amd.observability.kernel_timer import timer_snapshot

        return {
            "metrics": metrics_collector.snapshot(),
            "monitors": performance_monitor.all_summaries(),
            "gpu_regions": timer_snapshot(),
        }

    @app.get("/ui")
    async def ui_page():
        from sentio_amd.serving.ui import UI_HTML

        return HTMLResponse(UI_HTML)

    return app


async def _run_sync(fn, *args):
    import anyio

    return await anyio.to_thread.run_sync(fn, *args)


def main() -> None:  # uvicorn entry
    import uvicorn

    from sentio_amd.config import settings

    uvicorn.run(create_app(), host=settings.api_host, port=settings.api_port)


if __name__ == "__main__":
    main()
