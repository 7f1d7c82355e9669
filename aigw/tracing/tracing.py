"""Request tracing with OpenInference / OTel-GenAI semantic conventions.

Parity with internal/tracing:

- convention selected by ``AI_GATEWAY_TRACING_SEMCONV`` ∈ {``openinference``
  (default), ``gen_ai``}; an unknown value is a startup ERROR, not a
  fallback (semconv.go:66-96);
- ``OTEL_SDK_DISABLED=true`` disables tracing entirely;
- spans cover the whole gateway request; streamed chunks are recorded as
  span events; usage lands in convention-specific attributes;
- W3C ``traceparent`` is injected into UPSTREAM request headers so
  provider-side traces join (processor_impl.go:314-320);
- exporters: OTLP/HTTP JSON (``OTEL_EXPORTER_OTLP_ENDPOINT``, batched on a
  background thread — aigw/tracing/otlp.py, no SDK dependency), JSONL file
  (``AIGW_TRACE_FILE``), console, in-memory (tests).
"""

from __future__ import annotations

import json
import os
import secrets
import time
from dataclasses import dataclass, field
from typing import Optional


@dataclass
class Span:
    name: str
    trace_id: str
    span_id: str
    parent_span_id: str = ""
    start_ns: int = 0
    end_ns: int = 0
    attributes: dict = field(default_factory=dict)
    events: list = field(default_factory=list)
    status: str = "OK"

    def set(self, key: str, value) -> None:
        if value is not None:
            self.attributes[key] = value

    def add_event(self, name: str, attributes: Optional[dict] = None) -> None:
        self.events.append({"name": name, "ts_ns": time.time_ns(),
                            "attributes": attributes or {}})

    def traceparent(self) -> str:
        return f"00-{self.trace_id}-{self.span_id}-01"


class InMemoryExporter:
    def __init__(self):
        self.spans: list[Span] = []

    def export(self, span: Span) -> None:
        self.spans.append(span)


class JSONLExporter:
    def __init__(self, path: str):
        self.path = path
        self._f = open(path, "a", encoding="utf-8")

    def export(self, span: Span) -> None:
        self._f.write(json.dumps({
            "name": span.name,
            "traceId": span.trace_id,
            "spanId": span.span_id,
            "parentSpanId": span.parent_span_id,
            "startTimeUnixNano": span.start_ns,
            "endTimeUnixNano": span.end_ns,
            "attributes": span.attributes,
            "events": span.events,
            "status": span.status,
        }) + "\n")
        self._f.flush()


class ConsoleExporter:
    def export(self, span: Span) -> None:
        dur_ms = (span.end_ns - span.start_ns) / 1e6
        print(f"[span] {span.name} {dur_ms:.1f}ms {span.attributes}")


class Tracer:
    def __init__(self, semconv: str = "openinference", exporter=None,
                 header_attributes: Optional[dict[str, str]] = None):
        if semconv not in ("openinference", "gen_ai"):
            raise ValueError(
                f"unknown tracing semantic convention {semconv!r} "
                "(expected 'openinference' or 'gen_ai')"
            )
        self.semconv = semconv
        self.exporter = exporter or InMemoryExporter()
        # request-header name -> span attribute name (mainlib flag
        # -spanRequestHeaderAttributes)
        self.header_attributes = header_attributes or {}

    def start_span(self, name: str, headers: Optional[dict[str, str]] = None) -> Span:
        parent_trace = parent_span = ""
        if headers:
            tp = headers.get("traceparent", "")
            parts = tp.split("-")
            if len(parts) == 4 and len(parts[1]) == 32:
                parent_trace, parent_span = parts[1], parts[2]
        span = Span(
            name=name,
            trace_id=parent_trace or secrets.token_hex(16),
            span_id=secrets.token_hex(8),
            parent_span_id=parent_span,
            start_ns=time.time_ns(),
        )
        if headers:
            for hdr, attr in self.header_attributes.items():
                v = headers.get(hdr.lower())
                if v is not None:
                    span.set(attr, v)
        return span

    def end_span(self, span: Span, error: Optional[str] = None) -> None:
        span.end_ns = time.time_ns()
        if error:
            span.status = "ERROR"
            span.set("error.message", error)
        self.exporter.export(span)


class ChatSpanRecorder:
    """Fills span attributes per operation and the selected convention
    (internal/tracing/openinference/{openai,anthropic,cohere} + otelgenai
    recorder families). Despite the historical name this recorder is
    operation-aware: embeddings get EMBEDDING span kind and
    embedding.* attributes, rerank gets RERANKER + reranker.*, audio
    operations carry their operation name — chat semantics never leak
    onto non-chat spans."""

    # operation -> OpenInference span kind (openinference semconv)
    _OI_KIND = {
        "chat": "LLM",
        "text_completion": "LLM",
        "responses": "LLM",
        "embeddings": "EMBEDDING",
        "rerank": "RERANKER",
        "speech": "LLM",
        "transcription": "LLM",
        "translation": "LLM",
        "image_generation": "LLM",
        "count_tokens": "LLM",
        "tokenize": "LLM",
    }

    def __init__(self, tracer: Tracer, capture_content: bool = False):
        self.tracer = tracer
        self.capture_content = capture_content

    def record_request(self, span: Span, body: dict, *, provider: str,
                       backend: str, operation: str = "chat") -> None:
        body = body or {}
        model = body.get("model", "")
        if self.tracer.semconv == "openinference":
            span.set("openinference.span.kind",
                     self._OI_KIND.get(operation, "LLM"))
            if operation == "embeddings":
                span.set("embedding.model_name", model)
                inp = body.get("input")
                n = len(inp) if isinstance(inp, list) else 1 if inp else 0
                span.set("embedding.invocation_parameters", json.dumps(
                    {k: body[k] for k in ("dimensions", "encoding_format")
                     if k in body}) or None)
                span.set("embedding.embeddings_count", n)
            elif operation == "rerank":
                span.set("reranker.model_name", model)
                span.set("reranker.top_n", body.get("top_n"))
                docs = body.get("documents")
                span.set("reranker.documents_count",
                         len(docs) if isinstance(docs, list) else None)
                if self.capture_content:
                    span.set("reranker.query", body.get("query"))
            else:
                span.set("llm.model_name", model)
                span.set("llm.provider", provider)
                span.set("llm.system", provider)
                params = {
                    k: body[k]
                    for k in ("temperature", "top_p", "max_tokens",
                              "max_completion_tokens", "stream", "voice",
                              "response_format", "size", "language")
                    if k in body
                }
                span.set("llm.invocation_parameters",
                         json.dumps(params) if params else None)
                if self.capture_content and operation == "chat":
                    for i, m in enumerate(body.get("messages") or []):
                        span.set(f"llm.input_messages.{i}.message.role", m.get("role"))
                        c = m.get("content")
                        if isinstance(c, str):
                            span.set(f"llm.input_messages.{i}.message.content", c)
        else:
            span.set("gen_ai.operation.name", operation)
            span.set("gen_ai.provider.name", provider)
            span.set("gen_ai.request.model", model)
            if operation in ("chat", "text_completion", "responses"):
                span.set("gen_ai.request.temperature", body.get("temperature"))
                span.set("gen_ai.request.max_tokens",
                         body.get("max_completion_tokens") or body.get("max_tokens"))
        span.set("aigw.operation", operation)
        span.set("aigw.backend", backend)

    def record_response(self, span: Span, usage, response_model: str,
                        finish_reason: str = "") -> None:
        if self.tracer.semconv == "openinference":
            span.set("llm.token_count.prompt", usage.input_tokens)
            span.set("llm.token_count.completion", usage.output_tokens)
            span.set("llm.token_count.total", usage.total_tokens)
        else:
            span.set("gen_ai.response.model", response_model)
            span.set("gen_ai.usage.input_tokens", usage.input_tokens)
            span.set("gen_ai.usage.output_tokens", usage.output_tokens)
        if finish_reason:
            span.set("gen_ai.response.finish_reasons", [finish_reason])


def tracing_from_env(env: Optional[dict] = None) -> Optional[Tracer]:
    """NewTracingFromEnv analogue: returns None when disabled."""
    env = env if env is not None else dict(os.environ)
    if env.get("OTEL_SDK_DISABLED", "").lower() == "true":
        return None
    semconv = env.get("AI_GATEWAY_TRACING_SEMCONV", "openinference")
    trace_file = env.get("AIGW_TRACE_FILE", "")
    if trace_file:
        exporter = JSONLExporter(trace_file)
    elif env.get("OTEL_TRACES_EXPORTER", "") == "console":
        exporter = ConsoleExporter()
    elif env.get("OTEL_EXPORTER_OTLP_ENDPOINT") or env.get(
        "OTEL_EXPORTER_OTLP_TRACES_ENDPOINT"
    ):
        from aigw.tracing.otlp import OTLPHTTPExporter, parse_otlp_headers

        endpoint = env.get("OTEL_EXPORTER_OTLP_TRACES_ENDPOINT") or env[
            "OTEL_EXPORTER_OTLP_ENDPOINT"
        ]
        exporter = OTLPHTTPExporter(
            endpoint,
            headers=parse_otlp_headers(env.get("OTEL_EXPORTER_OTLP_HEADERS", "")),
            service_name=env.get("OTEL_SERVICE_NAME", "ai-gateway"),
            flush_interval_s=float(env.get("OTEL_BSP_SCHEDULE_DELAY", "5000")) / 1000.0,
        )
    elif env.get("OTEL_TRACES_EXPORTER"):
        exporter = ConsoleExporter()
    else:
        return None
    # span header->attribute mapping. Unset defaults to session-id
    # tracking; an EXPLICITLY EMPTY value clears the default
    # (requestheaderattrs/resolve.go:15-21 semantics)
    raw = env.get("AIGW_SPAN_REQUEST_HEADER_ATTRIBUTES")
    if raw is None:
        raw = "agent-session-id:session.id"
    header_attrs = {}
    for pair in raw.split(","):
        if ":" in pair:
            h, a = pair.split(":", 1)
            header_attrs[h.strip()] = a.strip()
    return Tracer(semconv=semconv, exporter=exporter, header_attributes=header_attrs)
