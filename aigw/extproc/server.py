"""The gateway data plane: one asyncio shard, optionally pinned to one GPU.

Replaces the reference's Envoy + router-extproc + upstream-extproc sandwich
(internal/extproc/server.go, processor_impl.go) with a single in-process
pipeline per request:

    parse body → model extract → route match (model header + rules)
    → rate-limit pre-check → semantic-cache lookup
    → per-try: translate request → header/body mutations → auth signing
      → upstream dispatch → (retriable failure? next backend, re-translated
        from the ORIGINAL body with original headers restored — A.8)
    → response translation (unary or streamed re-chunking)
    → usage extraction → cost programs → rate-limit charge → metrics

Invariants preserved from the reference: client-spoofable
x-ai-eg-* headers stripped at ingress; model routing happens after body
parse and before backend selection; auth signs the FINAL mutated body;
gateway-generated local replies are never re-translated; content-length is
dropped on streamed responses.
"""

from __future__ import annotations

import asyncio
import json
import logging
import time
from typing import Optional

from aiohttp import web

from aigw import internalapi
from aigw.backendauth import build_auth_handler
from aigw.backendauth.auth import CredentialMissingError
from aigw.extproc.router import RETRIABLE_STATUSES, backend_attempts, max_attempts
from aigw.extproc.upstream_client import UpstreamError
from aigw.filterapi.config import APISchemaName, Backend
from aigw.filterapi.runtime import CompiledRoute, RuntimeConfig
from aigw.llmcost import CostVars
from aigw.metrics import GenAIMetrics
from aigw.metrics.genai import provider_from_schema
from aigw.mutator import apply_body_mutation, apply_header_mutation
from aigw.ratelimit import RateLimiter
from aigw.translator import TranslationError, Usage, get_translator

try:  # C++ hot-path helpers (csrc/aigw_native.cpp); built in-tree
    import aigw_native as _native
except ImportError:  # pragma: no cover - built by setup.py everywhere
    _native = None

logger = logging.getLogger("aigw.server")

# request-id generation: uuid4() costs ~3-5 us per call (fork-safe RNG +
# string formatting) and runs once per unary response when the client
# sent no x-request-id — a random prefix + counter is unique enough for
# log correlation at a fraction of the cost
import itertools as _itertools
import os as _os

_REQ_ID_PREFIX = _os.urandom(6).hex()
_req_id_counter = _itertools.count()


def _next_request_id() -> str:
    return f"{_REQ_ID_PREFIX}-{next(_req_id_counter):x}"
access_logger = logging.getLogger("aigw.access")
# per-request JSON access lines are opt-in (AIGW_ACCESS_LOG=1 in the CLI)
if access_logger.level == logging.NOTSET:
    access_logger.setLevel(logging.WARNING)

# endpoint key -> (operation name for metrics)
JSON_ENDPOINTS = {
    "/v1/chat/completions": "chat",
    "/v1/completions": "text_completion",
    "/v1/embeddings": "embeddings",
    "/v1/images/generations": "image_generation",
    "/v1/responses": "responses",
    "/v1/responses/input_tokens": "responses_input_tokens",
    "/v1/audio/speech": "speech",
    "/tokenize": "tokenize",
    "/anthropic/v1/messages": "chat",
    "/anthropic/v1/messages/count_tokens": "count_tokens",
    "/v2/rerank": "rerank",
}
MULTIPART_ENDPOINTS = {
    "/v1/audio/transcriptions": "transcription",
    "/v1/audio/translations": "translation",
}
# endpoints whose requests may stream responses
STREAMABLE = {"/v1/chat/completions", "/v1/completions", "/v1/responses", "/anthropic/v1/messages"}


# Headers never forwarded upstream (proxy hop-by-hop semantics + the
# model-header internals, which travel via route state instead).
_HOP_BY_HOP = frozenset(
    {
        "host",
        "content-length",
        "content-type",
        "connection",
        "keep-alive",
        "proxy-authenticate",
        "proxy-authorization",
        "te",
        "trailer",
        "transfer-encoding",
        "upgrade",
        "expect",
        "accept-encoding",
        internalapi.MODEL_NAME_HEADER,
        internalapi.ORIGINAL_PATH_HEADER,
    }
)


# Endpoints that must carry a model in the body (endpointspec ParseBody);
# /tokenize and audio endpoints resolve the model elsewhere.
_MODEL_REQUIRED = frozenset(
    {
        "/v1/chat/completions",
        "/v1/completions",
        "/v1/embeddings",
        "/v1/images/generations",
        "/v1/responses",
        "/anthropic/v1/messages",
        "/anthropic/v1/messages/count_tokens",
        "/v2/rerank",
        "/v1/audio/speech",
    }
)

# Schemas whose request translation is a byte-level passthrough for OpenAI
# clients (body unchanged modulo model override) — eligible for the
# never-parse fast path.
_PASSTHROUGH_SCHEMAS = frozenset({APISchemaName.OPENAI, APISchemaName.AZURE_OPENAI})


class RequestView:
    """Transport-agnostic request: the processing core never touches the
    HTTP front's request object, so the same pipeline runs under the
    aiohttp front (full surface) and the raw lean front (hot paths)."""

    __slots__ = ("method", "path", "host", "remote", "headers", "body", "_stream_factory")

    def __init__(self, method, path, host, remote, headers, body, stream_factory):
        self.method = method
        self.path = path
        self.host = host
        self.remote = remote
        self.headers = headers  # lowercase dict, pre-spoof-strip
        self.body = body
        self._stream_factory = stream_factory

    async def start_stream(self, status: int, headers: dict[str, str]):
        """Begin a streamed response; returns a writer with
        write(bytes)/finish()/result()."""
        return await self._stream_factory(status, headers)


class _AiohttpStreamWriter:
    __slots__ = ("_resp",)

    def __init__(self, resp: web.StreamResponse):
        self._resp = resp

    async def write(self, data: bytes) -> None:
        await self._resp.write(data)

    async def finish(self):
        await self._resp.write_eof()

    def abort(self) -> None:
        """Terminate without proper EOF framing (mid-stream cut)."""
        self._resp.force_close()

    def result(self) -> web.StreamResponse:
        return self._resp


def _aiohttp_view(request: web.Request, body: bytes) -> RequestView:
    async def stream_factory(status: int, headers: dict[str, str]):
        resp = web.StreamResponse(status=status)
        for k, v in headers.items():
            resp.headers[k] = v
        await resp.prepare(request)
        return _AiohttpStreamWriter(resp)

    return RequestView(
        method=request.method,
        path=request.path,
        host=request.headers.get("host", request.host or ""),
        remote=request.remote,
        headers={k.lower(): v for k, v in request.headers.items()},
        body=body,
        stream_factory=stream_factory,
    )


async def _prepend(first, aiter):
    if first is not None:
        yield first
    async for item in aiter:
        yield item


_SAMPLING_KEYS = (
    "temperature", "top_p", "top_k", "n", "max_tokens", "max_completion_tokens",
    "stop", "presence_penalty", "frequency_penalty", "logit_bias", "seed",
    "response_format", "tools", "tool_choice", "logprobs", "top_logprobs",
    "reasoning_effort",
)


def _cache_fingerprint(model: str, route_name: str, body: dict,
                       client_headers: dict[str, str]) -> bytes:
    """Scope tag for semantic-cache values: two near-identical prompts may
    embed to the same key vector, so everything that must NOT be shared —
    model, route, generation parameters, and the client credential — is
    folded into a 16-byte digest compared on the VALUE at hit time."""
    import hashlib

    ident = (
        client_headers.get("authorization")
        or client_headers.get("x-api-key")
        or client_headers.get("api-key")
        or ""
    )
    params = {k: body[k] for k in _SAMPLING_KEYS if k in body}
    blob = json.dumps(
        [model, route_name, params,
         hashlib.sha256(ident.encode("utf-8", "replace")).hexdigest()],
        sort_keys=True, separators=(",", ":"), default=str,
    )
    return hashlib.sha256(blob.encode()).digest()[:16]


def _json_error(status: int, message: str, err_type: str = "invalid_request_error") -> web.Response:
    # gateway-generated ("local reply") error envelope, wire-compatible
    # with the reference's formatUserFacingErrorJSON
    # (processor_impl.go:190-196): {"type":"error","error":{type,code,message}}
    return web.json_response(
        {"type": "error",
         "error": {"type": err_type, "code": str(status), "message": message}},
        status=status,
    )


class GatewayServer:
    def __init__(
        self,
        runtime: RuntimeConfig,
        *,
        metrics: Optional[GenAIMetrics] = None,
        limiter: Optional[RateLimiter] = None,
        gpu_services=None,
        tracer=None,
        max_body_bytes: int = 50 * 1024 * 1024,  # reference raises Envoy's buffer to 50MiB
        endpoint_prefixes: Optional[dict[str, str]] = None,
        root_prefix: str = "",
        strict_schema: bool = False,
        # typed-union validation at ingress (aigw.apischema): reject
        # wrong-shape unions with 400 + a compact pydantic summary before
        # any translation (the reference validates by unmarshal into its
        # Go types; here it is opt-in because the scanner fast path
        # covers the structural minimum at ~30x lower cost)
        # global path prefix joined onto every endpoint (mainlib
        # --rootPrefix; main.go path.Join(flags.rootPrefix, ...))
        # extra path prefixes per API family (mainlib --endpointPrefixes,
        # main.go:143-147), e.g. {"openai": "/openai"} additionally mounts
        # /openai/v1/chat/completions etc.
    ):
        self.endpoint_prefixes = endpoint_prefixes or {}
        self.root_prefix = root_prefix.rstrip("/")
        self.runtime = runtime
        self.metrics = metrics or GenAIMetrics()
        self.limiter = limiter or RateLimiter(runtime.rate_limits)
        self.gpu = gpu_services  # aigw.gpu.services.GPUServices or None
        self.tracer = tracer  # aigw.tracing.Tracer or None
        if tracer is not None:
            from aigw.tracing import ChatSpanRecorder

            self.span_recorder = ChatSpanRecorder(tracer)
        else:
            self.span_recorder = None
        self.max_body_bytes = max_body_bytes
        self.strict_schema = strict_schema
        self._session = None  # lean upstream client (aigw.extproc.upstream_client)
        self._started_at = time.time()
        # graceful drain (the Envoy drain-sequence analogue): when
        # draining, /health fails (LBs stop routing here), new requests
        # get 503 + connection:close, and drain() waits for in-flight
        # work to finish before the process exits
        self.inflight = 0
        self.draining = False
        self._idle_event = asyncio.Event()
        # per-backend live stats for the KV-occupancy endpoint picker:
        # name -> [active_requests, estimated_active_tokens]
        self._ep_stats: dict[str, list] = {}
        # replica telemetry poller (aigw.extproc.telemetry); started by
        # the CLI / make_app when any backend configures telemetry
        self.telemetry = None
        # extra Prometheus providers appended to /metrics (native front)
        self.metrics_extra: list = []
        # access-log header->attribute mapping; unset defaults to
        # session tracking, explicitly empty clears it
        # (requestheaderattrs/resolve.go:15-21)
        import os as _osm

        raw = _osm.environ.get("AIGW_LOG_REQUEST_HEADER_ATTRIBUTES")
        if raw is None:
            raw = "agent-session-id:session.id"
        self.log_header_attrs: dict[str, str] = {}
        for pair in raw.split(","):
            if ":" in pair:
                h, a = pair.split(":", 1)
                self.log_header_attrs[h.strip().lower()] = a.strip()

    # ---- lifecycle -----------------------------------------------------------

    async def start(self) -> None:
        if self._session is None:
            from aigw.extproc.upstream_client import LeanClient

            self._session = LeanClient()
        if self.telemetry is None and any(
            b.telemetry is not None
            for cr in self.runtime.routes for tier in cr.tiers for b in tier
        ):
            from aigw.extproc.telemetry import ReplicaTelemetry

            self.telemetry = ReplicaTelemetry(self.runtime)
            await self.telemetry.scrape_once()  # first rows before traffic
            await self.telemetry.start()

    async def refresh_telemetry(self) -> None:
        """Re-register the telemetry poller against the CURRENT runtime
        (pool membership / hot reload changed the backend set)."""
        if self.telemetry is not None:
            await self.telemetry.stop()
            self.telemetry = None
        if any(b.telemetry is not None
               for cr in self.runtime.routes for tier in cr.tiers for b in tier):
            from aigw.extproc.telemetry import ReplicaTelemetry

            self.telemetry = ReplicaTelemetry(self.runtime)
            await self.telemetry.scrape_once()
            await self.telemetry.start()

    async def close(self) -> None:
        if self._session is not None:
            await self._session.close()
            self._session = None
        if self.telemetry is not None:
            await self.telemetry.stop()
            self.telemetry = None

    def swap_runtime(self, rc: RuntimeConfig) -> None:
        """Hot reload: new requests see the new config; in-flight requests
        keep the runtime they resolved (server.go:81-88 semantics)."""
        self.runtime = rc
        self.limiter.rules = {r.name: r for r in rc.rate_limits}

    async def drain(self, timeout_s: float = 30.0) -> int:
        """Stop taking new requests and wait (bounded) for in-flight ones.
        Returns the number still in flight at the deadline (0 = clean)."""
        self.draining = True
        if self.inflight == 0:
            return 0
        self._idle_event.clear()
        try:
            await asyncio.wait_for(self._idle_event.wait(), timeout=timeout_s)
        except asyncio.TimeoutError:
            pass
        return self.inflight

    @web.middleware
    async def _drain_middleware(self, request, handler):
        if self.draining and request.path not in ("/health", "/metrics"):
            return web.Response(
                status=503,
                text='{"error": {"message": "shutting down", "type": "unavailable"}}',
                content_type="application/json",
                headers={"connection": "close"},
            )
        self.inflight += 1
        try:
            return await handler(request)
        finally:
            self.inflight -= 1
            if self.inflight == 0 and self.draining:
                self._idle_event.set()

    def make_app(self) -> web.Application:
        app = web.Application(
            client_max_size=self.max_body_bytes, middlewares=[self._drain_middleware]
        )

        root = self.root_prefix

        def mount(path: str, handler) -> None:
            app.router.add_post(root + path, handler)
            for family, prefix in self.endpoint_prefixes.items():
                if family == "anthropic" and path.startswith("/anthropic/"):
                    app.router.add_post(root + prefix + path[len("/anthropic") :], handler)
                elif family == "cohere" and path.startswith("/v2/"):
                    app.router.add_post(root + prefix + path, handler)
                elif family == "openai" and not path.startswith(("/anthropic/", "/v2/")):
                    app.router.add_post(root + prefix + path, handler)

        for ep in JSON_ENDPOINTS:
            mount(ep, self._make_handler(ep))
        for ep in MULTIPART_ENDPOINTS:
            mount(ep, self._make_multipart_handler(ep))
        app.router.add_get(root + "/v1/models", self._handle_models)
        app.router.add_get(root + "/anthropic/v1/models", self._handle_anthropic_models)
        app.router.add_get("/health", self._handle_health)
        app.router.add_get("/metrics", self._handle_metrics)
        app.router.add_get("/debug/tasks", self._handle_debug_tasks)
        if self.gpu is not None:
            app.router.add_post("/v1/gateway/tokenize", self._handle_gpu_tokenize)
        mcp_cfg = self.runtime.config.mcp
        if mcp_cfg is not None and mcp_cfg.routes:
            from aigw.mcp import MCPProxy

            self._mcp_proxies = []
            for mr in mcp_cfg.routes:
                proxy = MCPProxy(mr, mcp_cfg.session_seed, metrics=self.metrics,
                                 tracer=self.tracer,
                                 fallback_seed=mcp_cfg.fallback_session_seed)
                self._mcp_proxies.append(proxy)
                for method in ("POST", "GET", "DELETE"):
                    app.router.add_route(method, mr.path, proxy.handle)
                if mr.oauth is not None:
                    # RFC 9728/8414 metadata, outside the authn gate
                    suffix = "" if mr.path == "/" else mr.path
                    app.router.add_get(
                        "/.well-known/oauth-protected-resource" + suffix,
                        proxy.handle_protected_resource_metadata)
                    app.router.add_get(
                        "/.well-known/oauth-authorization-server" + suffix,
                        proxy.handle_authorization_server_metadata)

            async def _close_mcp(_app):
                for p in self._mcp_proxies:
                    await p.close()

            app.on_cleanup.append(_close_mcp)
        app.on_startup.append(lambda _app: self.start())
        app.on_cleanup.append(lambda _app: self.close())
        return app

    # ---- admin ---------------------------------------------------------------

    async def _handle_health(self, request: web.Request) -> web.Response:
        if self.draining:  # LBs/k8s stop routing here during the drain window
            return web.json_response(
                {"status": "draining", "inflight": self.inflight}, status=503
            )
        return web.json_response({"status": "ok", "uptime_s": time.time() - self._started_at})

    async def _handle_metrics(self, request: web.Request) -> web.Response:
        body = self.metrics.render()
        for extra in self.metrics_extra:
            # native-front counters etc. (providers return Prometheus text)
            try:
                chunk = extra()
                body += chunk.encode() if isinstance(chunk, str) else chunk
            except Exception:
                logger.exception("extra metrics provider failed")
        return web.Response(body=body, content_type="text/plain")

    async def _handle_debug_tasks(self, request: web.Request) -> web.Response:
        """Live asyncio task dump — the profiling surface the reference gets
        from its always-on pprof server (internal/pprof/pprof.go:18-50).
        Loopback-only, disabled with AIGW_DISABLE_DEBUG (DISABLE_PPROF
        analogue)."""
        import os as _os

        if _os.environ.get("AIGW_DISABLE_DEBUG") or request.remote not in (
            "127.0.0.1",
            "::1",
        ):
            return web.Response(status=403)
        tasks = []
        for t in asyncio.all_tasks():
            frame = t.get_stack(limit=1)
            tasks.append(
                {
                    "name": t.get_name(),
                    "done": t.done(),
                    "where": f"{frame[0].f_code.co_filename}:{frame[0].f_lineno}"
                    if frame
                    else "",
                }
            )
        return web.json_response({"tasks": tasks, "count": len(tasks)})

    async def _handle_models(self, request: web.Request) -> web.Response:
        rt = self.runtime
        models = rt.models_for_host(request.headers.get("host", request.host or ""))
        return web.json_response(
            {
                "object": "list",
                "data": [
                    {
                        "id": m.name,
                        "object": "model",
                        "created": m.created_at,
                        "owned_by": m.owned_by,
                    }
                    for m in models
                ],
            }
        )

    async def _handle_anthropic_models(self, request: web.Request) -> web.Response:
        rt = self.runtime
        models = rt.models_for_host(request.headers.get("host", request.host or ""))
        return web.json_response(
            {
                "data": [
                    {
                        "id": m.name,
                        "type": "model",
                        "display_name": m.name,
                        "created_at": m.created_at,
                    }
                    for m in models
                ],
                "has_more": False,
            }
        )

    async def _handle_gpu_tokenize(self, request: web.Request) -> web.Response:
        body = await request.json()
        text = body.get("prompt") or body.get("text") or ""
        ids = await self.gpu.tokenize(text)
        return web.json_response({"count": len(ids), "tokens": ids})

    # ---- request processing --------------------------------------------------

    def _make_handler(self, endpoint: str):
        async def handler(request: web.Request) -> web.StreamResponse:
            body = await request.read()
            return await self._process(_aiohttp_view(request, body), endpoint)

        return handler

    def _make_multipart_handler(self, endpoint: str):
        async def handler(request: web.Request) -> web.StreamResponse:
            return await self._process_multipart(request, endpoint)

        return handler

    @staticmethod
    def _ingress_headers(view) -> dict[str, str]:
        headers = dict(view.headers)
        for h in internalapi.INTERNAL_HEADERS:
            headers.pop(h, None)  # spoof protection (server.go:439-455)
        return headers

    def _translator_kwargs(self, backend: Backend) -> dict:
        kw: dict = {"api_version": backend.schema.version}
        if backend.auth is not None:
            kw["gcp_project"] = backend.auth.gcp_project
            kw["gcp_region"] = backend.auth.gcp_region
        return kw

    async def _process(self, request: RequestView, endpoint: str):
        start = time.monotonic()
        rt = self.runtime
        raw = request.body or b"{}"
        # Fast path: one C++ pass extracts model/stream/chat-text without
        # building Python objects; full json.loads happens lazily only when
        # a translator or mutation needs the dict.
        body: Optional[dict] = None
        chat_text = b""
        if _native is not None:
            ok, model, stream_flag, chat_text = _native.scan_chat_body(raw)
            if not ok:
                return _json_error(400, "invalid request body: malformed JSON")
        if _native is None or body is not None:
            try:
                body = json.loads(raw)
                if not isinstance(body, dict):
                    raise ValueError("body must be a JSON object")
            except ValueError as e:
                return _json_error(400, f"invalid request body: {e}")
            model = str(body.get("model", ""))
            stream_flag = bool(body.get("stream"))

        if not model and endpoint in _MODEL_REQUIRED:
            # parse failure -> 400, matching the reference's ParseBody error
            # (processor_impl.go:248-252), not a 404 route miss
            return _json_error(400, "missing required field 'model'")
        if body is not None and endpoint in ("/v1/chat/completions",) and not isinstance(
            body.get("messages", []), list
        ):
            return _json_error(400, "'messages' must be an array")
        if self.strict_schema:
            from aigw.apischema import SchemaError, validate_request

            if body is None:
                body = json.loads(raw)
            try:
                validate_request(endpoint, body)
            except SchemaError as e:
                return _json_error(400, f"schema validation failed: {e}")

        headers = self._ingress_headers(request)
        headers[rt.model_header] = model
        headers[internalapi.ORIGINAL_PATH_HEADER] = request.path
        stream = stream_flag and endpoint in STREAMABLE

        route = rt.select_route(headers)
        if route is None:
            return _json_error(404, f"no route matched model {model!r}", "model_not_found")

        # usage-based rate-limit pre-admission
        decision = self.limiter.check(headers)
        if not decision.allowed:
            self.metrics.ratelimit_denials.labels(rule=decision.rule).inc()
            resp = _json_error(429, "token budget exhausted", "rate_limit_exceeded")
            resp.headers["retry-after"] = str(int(decision.retry_after_s) + 1)
            return resp

        # GPU-side input token count (admission accounting / local usage).
        gpu_input_tokens = 0
        if self.gpu is not None and endpoint in ("/v1/chat/completions", "/anthropic/v1/messages"):
            if not chat_text:
                if body is None:
                    body = json.loads(raw)
                from aigw.gpu.services import extract_chat_text

                chat_text = extract_chat_text(body)
            gpu_input_tokens = await self.gpu.count_text_tokens(chat_text)

        # semantic response cache (GPU MFMA embed + HBM index). Streamed
        # requests are cached too: the translated SSE transcript is stored
        # on miss and replayed chunk-wise on hit. Scope lives in a value
        # tag (a key prefix cannot separate scopes in embedding space —
        # mean-pooled vectors of near-identical texts coincide): one mode
        # byte (stream/unary) plus a fingerprint of (model, route,
        # sampling params, client credential), so a cached answer never
        # crosses models, generation settings, or tenants.
        cache_key_vec = None
        cache_tag = b""
        if (
            self.gpu is not None
            and self.gpu.cache_enabled
            and endpoint == "/v1/chat/completions"
        ):
            if body is None:
                body = json.loads(raw)
            cache_tag = (b"S" if stream else b"U") + _cache_fingerprint(
                model, route.route.name, body, request.headers
            )
            hit, cache_key_vec = await self.gpu.cache_lookup_text(chat_text or b" ")
            if hit is not None and hit[: len(cache_tag)] == cache_tag:
                self.metrics._child(self.metrics.cache_events, ("hit",)).inc()
                body_bytes = hit[len(cache_tag):]
                if not stream:
                    resp = web.Response(body=body_bytes, content_type="application/json")
                    resp.headers["x-aigw-cache"] = "hit"
                    return resp
                writer = await request.start_stream(
                    200,
                    {"content-type": "text/event-stream",
                     "cache-control": "no-cache", "x-aigw-cache": "hit"},
                )
                for off in range(0, len(body_bytes), 16384):
                    await writer.write(body_bytes[off : off + 16384])
                await writer.finish()
                return writer.result()
            self.metrics._child(self.metrics.cache_events, ("miss",)).inc()

        span = None
        if self.tracer is not None:
            span = self.tracer.start_span(f"{JSON_ENDPOINTS.get(endpoint, 'request')} {model}",
                                          headers)
        try:
            resp = await self._dispatch(
                request, endpoint, route, headers, body, stream, start,
                model=model, gpu_input_tokens=gpu_input_tokens, cache_key_vec=cache_key_vec,
                cache_tag=cache_tag,
                span=span, raw=raw,
            )
            if span is not None:
                self.tracer.end_span(span, error=None if resp.status < 500 else f"status {resp.status}")
            return resp
        except Exception as e:
            if span is not None:
                self.tracer.end_span(span, error=str(e))
            raise

    async def _dispatch(
        self,
        request: web.Request,
        endpoint: str,
        route: CompiledRoute,
        headers: dict[str, str],
        body: dict,
        stream: bool,
        start: float,
        *,
        model: str,
        gpu_input_tokens: int = 0,
        cache_key_vec=None,
        cache_tag: bytes = b"",
        span=None,
        raw: bytes = b"",
    ) -> web.StreamResponse:
        rt = self.runtime
        assert self._session is not None, "GatewayServer.start() not called"
        attempts_left = max_attempts(route)
        if attempts_left == 0:
            return _json_error(503, "route has no backends", "no_backend")
        last_error: Optional[str] = None
        attempt = 0

        attempts_iter = backend_attempts(route)
        if (
            route.route.endpoint_picker
            and self.gpu is not None
            and route.tiers
            and len(route.tiers[0]) > 1
        ):
            tier0 = route.tiers[0]
            stats_rows = []
            for b in tier0:
                row = None
                if self.telemetry is not None:
                    # scraped replica state (vLLM /metrics): real KV
                    # occupancy + queue depth; stale rows fall back to
                    # the gateway-local in-flight estimate
                    row = self.telemetry.fresh_row(
                        b.name,
                        max_age_s=(b.telemetry.interval_s * 3
                                   if b.telemetry else 5.0),
                    )
                if row is None:
                    st = self._ep_stats.setdefault(b.name, [0, 0.0])
                    row = [st[1], 100000.0, float(st[0]), float(st[0])]
                stats_rows.append(row)
            pick = await self.gpu.pick_endpoint(
                stats_rows, float(gpu_input_tokens or 256)
            )
            ordered = [tier0[pick]] + [b for i, b in enumerate(tier0) if i != pick]

            def _gen():
                yield from ordered
                for tier in route.tiers[1:]:
                    yield from tier

            attempts_iter = _gen()

        for backend in attempts_iter:
            if attempts_left <= 0:
                break
            attempts_left -= 1
            first = attempt == 0
            attempt += 1
            if not first:
                self.metrics.retries_total.labels(route=route.route.name).inc()

            try:
                translator = get_translator(
                    endpoint, backend.schema.name, **self._translator_kwargs(backend)
                )
            except TranslationError as e:
                last_error = str(e)
                continue

            # First attempt works on the live parsed body (translators may
            # mutate it); every RETRY re-parses the ORIGINAL raw bytes, which
            # is the A.8 "restore original body per try" semantics without a
            # defensive deep copy on the hot path. When the C++ scanner
            # already yielded model/stream and this backend is a byte-level
            # passthrough, the dict is never materialized at all.
            override = backend.model_name_override or route.route.model_name_override
            if first and body is None:
                if (
                    backend.schema.name in _PASSTHROUGH_SCHEMAS
                    and backend.body_mutation is None
                    and not override
                    and self.tracer is None
                    and not (stream and route.costs)
                ):
                    body_copy = {"model": model, "stream": stream}
                else:
                    body = body_copy = json.loads(raw)
            elif first:
                body_copy = body
            else:
                body_copy = json.loads(raw)
            try:
                tr = translator.request(
                    body_copy,
                    model_override=override,
                    stream=stream,
                    force_include_usage=bool(route.costs),
                    raw=raw if first else b"",
                )
            except TranslationError as e:
                return _json_error(422, str(e))

            out_body = tr.body
            if backend.body_mutation is not None:
                doc = json.loads(out_body)
                apply_body_mutation(doc, backend.body_mutation)
                out_body = json.dumps(doc, separators=(",", ":")).encode()

            # Headers: fresh copy of ORIGINAL client headers each try
            # (headermutator restore-on-retry), then route/backend mutations,
            # then translator-set headers, then auth over the FINAL body.
            up_headers = {k: v for k, v in headers.items() if k not in _HOP_BY_HOP}
            up_headers["content-type"] = "application/json"
            up_headers = apply_header_mutation(up_headers, route.route.header_mutation)
            up_headers = apply_header_mutation(up_headers, backend.header_mutation)
            up_headers.update(tr.headers)
            for h in tr.remove_headers:
                up_headers.pop(h, None)
            auth = build_auth_handler(backend)
            if auth is not None:
                try:
                    up_headers = auth(up_headers, out_body, "POST", tr.path)
                except CredentialMissingError as e:
                    # per-request credential source configured but absent
                    # (or incomplete) with no fallback -> 401 local reply
                    # (credential_override.go ErrCredentialMissing)
                    return _json_error(401, str(e), "authentication_error")
            else:
                # propagate client Authorization for auth-less backends
                if "authorization" in headers:
                    up_headers.setdefault("authorization", headers["authorization"])
            for h in rt.override_strip_headers:
                # overrides not consumed by THIS backend's handler never
                # travel upstream (reserved override names + all configured)
                up_headers.pop(h, None)

            if backend.upstream.hostname:
                up_headers["host"] = backend.upstream.hostname
            if span is not None:
                # span context joins provider-side traces (A.8 / §5.1)
                up_headers["traceparent"] = span.traceparent()
                self.span_recorder.record_request(
                    span, body,
                    provider=provider_from_schema(backend.schema.name.value, backend.name),
                    backend=backend.name,
                    operation=JSON_ENDPOINTS.get(endpoint, "request"),
                )
            st = self._ep_stats.setdefault(backend.name, [0, 0.0])
            if backend.max_concurrency and st[0] >= backend.max_concurrency:
                # circuit open: saturated backend = failed attempt
                last_error = f"backend {backend.name} saturated ({st[0]} in flight)"
                self.metrics._child(
                    self.metrics.requests_total,
                    (endpoint, backend.name, "circuit_open"),
                ).inc()
                continue
            st[0] += 1
            st[1] += gpu_input_tokens
            try:
                upstream = await self._session.post(
                    host=backend.upstream.host,
                    port=backend.upstream.port,
                    tls=backend.upstream.tls,
                    path=backend.upstream.path_prefix + tr.path,
                    headers=up_headers,
                    body=out_body,
                    timeout_s=backend.timeout_s,
                    server_name=backend.upstream.hostname,
                    ca_file=backend.upstream.ca_file,
                    ca_pem=backend.upstream.ca_pem,
                )
                status = upstream.status
                if status >= 400:
                    try:
                        err_body = await upstream.read()
                    except BaseException:
                        # failed mid-error-body read: drop the connection
                        # instead of leaking it into the retry loop
                        upstream.close()
                        raise
                    upstream.release()
                    if status in RETRIABLE_STATUSES and attempts_left > 0:
                        last_error = f"upstream {backend.name} returned {status}"
                        logger.warning("%s; trying next backend", last_error)
                        continue
                    out = translator.response_error(status, err_body, upstream.headers)
                    self._finish_metrics(
                        endpoint, route, backend, model, "", Usage(), start,
                        status=status, error_type=f"upstream_{status}",
                    )
                    return web.Response(
                        body=out, status=status, content_type="application/json"
                    )
                if stream:
                    return await self._stream_response(
                        request, endpoint, route, backend, translator, upstream,
                        headers, model, start, gpu_input_tokens, span=span,
                        cache_key_vec=cache_key_vec, cache_tag=cache_tag,
                    )
                return await self._unary_response(
                    endpoint, route, backend, translator, upstream, headers,
                    model, start, gpu_input_tokens, body, cache_key_vec,
                    cache_tag=cache_tag, span=span,
                )
            except (UpstreamError, OSError, asyncio.TimeoutError, asyncio.IncompleteReadError) as e:
                last_error = f"upstream {backend.name}: {type(e).__name__}: {e}"
                logger.warning("%s; %d attempts left", last_error, attempts_left)
                continue
            finally:
                st[0] -= 1
                st[1] -= gpu_input_tokens

        self._finish_metrics(
            endpoint, route, None, model, "", Usage(), start, status=503,
            error_type="no_healthy_upstream",
        )
        return _json_error(503, last_error or "no healthy upstream", "upstream_error")

    # ---- response paths ------------------------------------------------------

    def _apply_costs(
        self, route: CompiledRoute, backend: Backend, headers: dict[str, str],
        model: str, usage: Usage, gpu_input_tokens: int,
    ) -> dict[str, int]:
        if gpu_input_tokens and not usage.input_tokens:
            # upstream reported no usage; fall back to the gateway's own
            # GPU-tokenized count
            usage.input_tokens = gpu_input_tokens
            usage.total_tokens = max(usage.total_tokens, usage.input_tokens + usage.output_tokens)
        v = CostVars(
            model=model,
            backend=backend.name if backend else "",
            route_name=route.route.name,
            input_tokens=usage.input_tokens,
            cached_input_tokens=usage.cached_input_tokens,
            cache_creation_input_tokens=usage.cache_creation_input_tokens,
            output_tokens=usage.output_tokens,
            total_tokens=usage.total_tokens,
            reasoning_tokens=usage.reasoning_tokens,
        )
        costs: dict[str, int] = {
            internalapi.META_INPUT_TOKENS: usage.input_tokens,
            internalapi.META_OUTPUT_TOKENS: usage.output_tokens,
            internalapi.META_TOTAL_TOKENS: usage.total_tokens,
        }
        for cc in route.costs:
            try:
                costs[cc.metadata_key] = cc.evaluate(v)
            except Exception:
                logger.exception("cost %s failed", cc.metadata_key)
        self.limiter.charge(headers, costs)
        return costs

    def _finish_metrics(
        self, endpoint, route, backend, model, response_model, usage, start,
        *, status: int, error_type: str = "", ttft: float = -1.0,
        headers=None,
    ) -> None:
        provider = (
            provider_from_schema(backend.schema.name.value, backend.name) if backend else ""
        )
        labels = self.metrics.labels(
            operation=JSON_ENDPOINTS.get(endpoint, endpoint),
            provider=provider,
            original_model=model,
            request_model=model,
            response_model=response_model,
        )
        elapsed = time.monotonic() - start
        self.metrics.record_request(labels, elapsed, error_type)
        if usage.total_tokens or usage.input_tokens or usage.output_tokens:
            self.metrics.record_tokens(labels, usage)
        if ttft >= 0:
            self.metrics.record_stream_latency(labels, ttft, elapsed, usage.output_tokens)
        self.metrics._child(
            self.metrics.requests_total,
            (endpoint, backend.name if backend else "", str(status)),
        ).inc()
        if access_logger.isEnabledFor(logging.INFO):
            # enriched access log, carrying what the reference pushes into
            # Envoy access logs via dynamic metadata (§5.5: backend_name,
            # token counts, token_latency_ttft/itl)
            access_logger.info(
                json.dumps(
                    {
                        "endpoint": endpoint,
                        "route": route.route.name if route else "",
                        "backend": backend.name if backend else "",
                        "status": status,
                        "duration_ms": round(elapsed * 1000.0, 2),
                        "model": model,
                        "response_model": response_model,
                        "input_tokens": usage.input_tokens,
                        "output_tokens": usage.output_tokens,
                        "total_tokens": usage.total_tokens,
                        "ttft_ms": round(ttft * 1000.0, 2) if ttft >= 0 else None,
                        "error_type": error_type or None,
                        # header->attribute mapping (requestheaderattrs
                        # defaults: agent-session-id -> session.id)
                        **{attr: headers.get(h)
                           for h, attr in self.log_header_attrs.items()
                           if headers is not None and headers.get(h)},
                    },
                    separators=(",", ":"),
                )
            )

    async def _unary_response(
        self, endpoint, route, backend, translator, upstream, headers, model,
        start, gpu_input_tokens, orig_body, cache_key_vec, cache_tag=b"", span=None,
    ) -> web.Response:
        # idle-bounded read: a stalled upstream raises UpstreamError, which
        # _dispatch treats as a failed try (fallback proceeds — nothing has
        # been written to the client yet)
        data = await upstream.read(idle_timeout=backend.stream_idle_timeout_s)
        upstream.release()
        try:
            rtl = translator.response_body(upstream.status, data)
        except Exception as e:
            logger.exception("response translation failed")
            return _json_error(502, f"response translation failed: {e}", "translation_error")
        usage = rtl.usage or Usage()
        self._apply_costs(route, backend, headers, model, usage, gpu_input_tokens)
        if span is not None:
            self.span_recorder.record_response(span, usage, rtl.response_model)
        self._finish_metrics(
            endpoint, route, backend, model, rtl.response_model, usage, start,
            status=upstream.status, headers=headers,
        )
        hdrs = translator.response_headers(upstream.status, upstream.headers)
        content_type = hdrs.get("content-type") or upstream.headers.get(
            "content-type", "application/json"
        )
        resp = web.Response(body=rtl.body, status=upstream.status)
        resp.headers["content-type"] = content_type
        rid = headers.get("x-request-id")
        resp.headers["x-request-id"] = rid if rid is not None else _next_request_id()
        if (
            cache_key_vec is not None
            and upstream.status == 200
            and len(rtl.body) < (2 << 20)
            # same 2 MiB cap as streamed transcripts: 64k slots x unbounded
            # bodies would otherwise grow host memory without limit
        ):
            await self.gpu.cache_insert(cache_key_vec, cache_tag + rtl.body)
        return resp

    async def _stream_response(
        self, request, endpoint, route, backend, translator, upstream,
        headers, model, start, gpu_input_tokens, span=None, cache_key_vec=None,
        cache_tag=b"",
    ) -> web.StreamResponse:
        transcript = bytearray() if cache_key_vec is not None else None
        idle = backend.stream_idle_timeout_s
        chunks = upstream.iter_chunks(idle_timeout=idle)
        first_chunk = None
        if idle > 0:
            # wait for the first body byte BEFORE committing a response to
            # the client: an idle upstream here resets the try and fallback
            # proceeds (per_try_idle_timeout pre-first-byte semantics) —
            # UpstreamError propagates to _dispatch's retry loop
            try:
                first_chunk = await chunks.__anext__()
            except StopAsyncIteration:
                first_chunk = None
        hdrs = translator.response_headers(upstream.status, upstream.headers)
        content_type = hdrs.get("content-type") or upstream.headers.get(
            "content-type", "text/event-stream"
        )
        # content-length is dropped for streamed bodies (A.8)
        writer = await request.start_stream(
            upstream.status, {"content-type": content_type, "cache-control": "no-cache"}
        )

        usage = Usage()
        response_model = ""
        ttft = -1.0
        aborted = False
        cut = False
        try:
            async for chunk in _prepend(first_chunk, chunks):
                if not chunk:
                    continue
                rtl = translator.response_chunk(chunk)
                if rtl.body:
                    if ttft < 0:
                        ttft = time.monotonic() - start
                    await writer.write(rtl.body)
                    if transcript is not None and len(transcript) < (2 << 20):
                        transcript.extend(rtl.body)
                if rtl.usage is not None:
                    usage.merge_max(rtl.usage)
                if rtl.response_model:
                    response_model = rtl.response_model
            tail = translator.response_flush()
            if tail.body:
                await writer.write(tail.body)
                if transcript is not None:
                    transcript.extend(tail.body)
            if tail.usage is not None:
                usage.merge_max(tail.usage)
            if (
                transcript is not None
                and upstream.status == 200
                and len(transcript) < (2 << 20)
            ):
                await self.gpu.cache_insert(cache_key_vec, cache_tag + bytes(transcript))
        except (UpstreamError, ValueError) as e:
            # mid-stream failure (idle timeout, or malformed provider bytes
            # the translator/codec rejects): the response has started, so
            # fallback is no longer possible — cut the stream (the client
            # sees truncated framing; Envoy's analogue is a stream reset)
            cut = True
            logger.warning("cutting stream to %s: %s", backend.name, e)
        except (ConnectionResetError, asyncio.CancelledError):
            aborted = True
            logger.info("client disconnected mid-stream")
            raise
        finally:
            if aborted or cut:
                upstream.close()
            else:
                upstream.release()
            self._apply_costs(route, backend, headers, model, usage, gpu_input_tokens)
            if span is not None:
                self.span_recorder.record_response(span, usage, response_model)
            self._finish_metrics(
                endpoint, route, backend, model, response_model, usage, start,
                status=upstream.status, ttft=ttft if ttft >= 0 else 0.0,
                headers=headers,
            )
        if cut:
            abort = getattr(writer, "abort", None)
            if abort is not None:
                abort()
            return writer.result()
        await writer.finish()
        return writer.result()

    # ---- multipart (audio) ---------------------------------------------------

    async def _process_multipart(self, request: web.Request, endpoint: str) -> web.StreamResponse:
        from aigw.translator.misc_endpoints import _parse_multipart

        start = time.monotonic()
        rt = self.runtime
        raw = await request.read()
        content_type = request.headers.get("content-type", "")
        try:
            parts, _ = _parse_multipart(raw, content_type)
        except ValueError as e:
            return _json_error(400, str(e))
        model = ""
        for name, _h, payload in parts:
            if name == "model":
                model = payload.decode("utf-8", "replace")
        headers = {k.lower(): v for k, v in request.headers.items()}
        for h in internalapi.INTERNAL_HEADERS:
            headers.pop(h, None)
        headers[rt.model_header] = model
        route = rt.select_route(headers)
        if route is None:
            return _json_error(404, f"no route matched model {model!r}", "model_not_found")
        decision = self.limiter.check(headers)
        if not decision.allowed:
            self.metrics.ratelimit_denials.labels(rule=decision.rule).inc()
            return _json_error(429, "token budget exhausted", "rate_limit_exceeded")
        span = None
        operation = MULTIPART_ENDPOINTS.get(endpoint, "audio")
        if self.tracer is not None:
            span = self.tracer.start_span(f"{operation} {model}", headers)

        assert self._session is not None
        attempts_left = max_attempts(route)
        last_error = None
        for backend in backend_attempts(route):
            if attempts_left <= 0:
                break
            attempts_left -= 1
            translator = get_translator(
                endpoint, backend.schema.name, **self._translator_kwargs(backend)
            )
            override = backend.model_name_override or route.route.model_name_override
            tr, model = translator.request_multipart(raw, content_type, model_override=override)
            # client headers forward exactly like the JSON path (fresh copy
            # per try, hop-by-hop + internal stripped), then mutations,
            # then translator headers (multipart content-type wins), then
            # auth over the final body — A.8 ordering
            up_headers = {k: v for k, v in headers.items() if k not in _HOP_BY_HOP}
            up_headers = apply_header_mutation(up_headers, route.route.header_mutation)
            up_headers = apply_header_mutation(up_headers, backend.header_mutation)
            up_headers.update(tr.headers)
            auth = build_auth_handler(backend)
            if auth is not None:
                try:
                    up_headers = auth(up_headers, tr.body, "POST", tr.path)
                except CredentialMissingError as e:
                    return _json_error(401, str(e), "authentication_error")
            else:
                # propagate client Authorization for auth-less backends
                if "authorization" in headers:
                    up_headers.setdefault("authorization", headers["authorization"])
            for h in rt.override_strip_headers:
                up_headers.pop(h, None)
            if backend.upstream.hostname:
                up_headers["host"] = backend.upstream.hostname
            if span is not None:
                up_headers["traceparent"] = span.traceparent()
                self.span_recorder.record_request(
                    span, {"model": model},
                    provider=provider_from_schema(backend.schema.name.value, backend.name),
                    backend=backend.name, operation=operation,
                )
            try:
                upstream = await self._session.post(
                    host=backend.upstream.host,
                    port=backend.upstream.port,
                    tls=backend.upstream.tls,
                    path=backend.upstream.path_prefix + tr.path,
                    headers=up_headers,
                    body=tr.body,
                    timeout_s=backend.timeout_s,
                    server_name=backend.upstream.hostname,
                    ca_file=backend.upstream.ca_file,
                    ca_pem=backend.upstream.ca_pem,
                )
                if upstream.status >= 400:
                    try:
                        err = await upstream.read()
                    except BaseException:
                        upstream.close()
                        raise
                    upstream.release()
                    if upstream.status in RETRIABLE_STATUSES and attempts_left > 0:
                        continue
                    if span is not None:
                        self.tracer.end_span(span, error=f"status {upstream.status}")
                    return web.Response(
                        body=translator.response_error(upstream.status, err, {}),
                        status=upstream.status,
                        content_type="application/json",
                    )
                resp = await self._unary_response(
                    endpoint, route, backend, translator, upstream, headers,
                    model, start, 0, None, None, span=span,
                )
                if span is not None:
                    self.tracer.end_span(span)
                return resp
            except (UpstreamError, OSError, asyncio.TimeoutError,
                    asyncio.IncompleteReadError) as e:
                last_error = str(e)
                continue
        if span is not None:
            self.tracer.end_span(span, error=last_error or "no healthy upstream")
        return _json_error(503, last_error or "no healthy upstream", "upstream_error")


async def run_server(
    server: GatewayServer,
    host: str = "0.0.0.0",
    port: int = internalapi.DEFAULT_LISTEN_PORT,
    reuse_port: bool = False,
) -> web.AppRunner:
    app = server.make_app()
    runner = web.AppRunner(app, access_log=None)
    await runner.setup()
    site = web.TCPSite(
        runner, host, port, backlog=4096, reuse_address=True, reuse_port=reuse_port
    )
    await site.start()
    logger.info("aigw listening on http://%s:%d", host, port)
    return runner
