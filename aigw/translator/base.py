"""Translator core types.

Mirrors internal/translator/translator.go:47-83: a Translator is created
per request (stateful across a streamed response, never shared between
requests) and exposes request-body translation, response-header mutation,
per-chunk response translation with cumulative usage extraction, and error
translation. Concrete translators live in sibling modules; the registry at
the bottom is the endpoint×schema matrix from SURVEY.md §2.2/§A.9.
"""

from __future__ import annotations

import json
from dataclasses import dataclass, field
from typing import Optional

from aigw.filterapi.config import APISchemaName


@dataclass
class Usage:
    """Cumulative token usage for one request (openai.Usage superset)."""

    input_tokens: int = 0
    output_tokens: int = 0
    total_tokens: int = 0
    cached_input_tokens: int = 0
    cache_creation_input_tokens: int = 0
    reasoning_tokens: int = 0

    def merge_max(self, other: "Usage") -> None:
        """Streaming usage chunks report cumulative counts; keep the max
        (reference: metrics_impl.go totalOutputTokens max-tracking)."""
        self.input_tokens = max(self.input_tokens, other.input_tokens)
        self.output_tokens = max(self.output_tokens, other.output_tokens)
        self.total_tokens = max(self.total_tokens, other.total_tokens)
        self.cached_input_tokens = max(self.cached_input_tokens, other.cached_input_tokens)
        self.cache_creation_input_tokens = max(
            self.cache_creation_input_tokens, other.cache_creation_input_tokens
        )
        self.reasoning_tokens = max(self.reasoning_tokens, other.reasoning_tokens)


def usage_from_openai(u: dict) -> Usage:
    pd = u.get("prompt_tokens_details") or {}
    cd = u.get("completion_tokens_details") or {}
    return Usage(
        input_tokens=u.get("prompt_tokens", 0) or 0,
        output_tokens=u.get("completion_tokens", 0) or 0,
        total_tokens=u.get("total_tokens", 0) or 0,
        cached_input_tokens=pd.get("cached_tokens", 0) or 0,
        reasoning_tokens=cd.get("reasoning_tokens", 0) or 0,
    )



@dataclass
class RequestTranslation:
    """Result of translating a client request for one backend try."""

    path: str
    body: bytes
    headers: dict[str, str] = field(default_factory=dict)  # headers to set
    remove_headers: list[str] = field(default_factory=list)


@dataclass
class ResponseTranslation:
    """Result of translating (a chunk of) an upstream response."""

    body: bytes = b""
    usage: Optional[Usage] = None
    # response model name as reported by the provider (for metrics)
    response_model: str = ""
    end_of_stream: bool = False


class TranslationError(ValueError):
    """Client-facing 4xx: the request cannot be represented upstream."""


class Translator:
    """Base class. One instance per request try; stateful for streaming."""

    # Content-Type the translated upstream response stream arrives as;
    # streaming translators that re-encode to SSE override response
    # content-type via response_headers().
    def request(self, body: dict, *, model_override: str = "", stream: bool = False,
                force_include_usage: bool = False, raw: bytes = b"") -> RequestTranslation:
        raise NotImplementedError

    def response_headers(self, status: int, headers: dict[str, str]) -> dict[str, str]:
        """Header mutations for the downstream response (e.g. content-type
        rewrite for event-stream→SSE; openai_awsbedrock.go:286-298)."""
        return {}

    def response_body(self, status: int, body: bytes) -> ResponseTranslation:
        """Unary (non-streaming) response translation."""
        raise NotImplementedError

    def response_chunk(self, chunk: bytes) -> ResponseTranslation:
        """Streaming response translation; called per upstream chunk."""
        raise NotImplementedError

    def response_flush(self) -> ResponseTranslation:
        """End of upstream stream; emit any synthesized trailer events."""
        return ResponseTranslation(end_of_stream=True)

    def response_error(self, status: int, body: bytes, headers: dict[str, str]) -> bytes:
        """Translate an upstream error into the client schema (default:
        OpenAI error envelope, translator.go ResponseError semantics)."""
        text = body.decode("utf-8", "replace")
        try:
            parsed = json.loads(text)
            if isinstance(parsed, dict) and "error" in parsed:
                return body
            msg = text
        except ValueError:
            msg = text
        return json.dumps(
            {"error": {"message": msg, "type": "upstream_error", "code": str(status)}}
        ).encode("utf-8")


def override_model(body: dict, model_override: str) -> str:
    """Apply ModelNameOverride and return the effective model name
    (processor_impl.go model override via sjson)."""
    if model_override:
        body["model"] = model_override
    return body.get("model", "")


def jdump(obj) -> bytes:
    return json.dumps(obj, separators=(",", ":"), ensure_ascii=False).encode("utf-8")


# --- registry ----------------------------------------------------------------

# (endpoint, backend schema) -> Translator factory. Client schema is implied
# by the endpoint (OpenAI-prefix endpoints take OpenAI bodies; /anthropic
# endpoints take Anthropic bodies) as in mainlib/main.go:326-354.
_REGISTRY: dict[tuple[str, APISchemaName], type] = {}


def register(endpoint: str, schema: APISchemaName):
    def deco(cls):
        _REGISTRY[(endpoint, schema)] = cls
        return cls

    return deco


def get_translator(endpoint: str, schema: APISchemaName, **kwargs) -> Translator:
    cls = _REGISTRY.get((endpoint, schema))
    if cls is None:
        raise TranslationError(
            f"unsupported schema {schema.value} for endpoint {endpoint}"
        )
    return cls(**kwargs)


def supported_matrix() -> dict[str, list[str]]:
    out: dict[str, list[str]] = {}
    for (ep, schema) in _REGISTRY:
        out.setdefault(ep, []).append(schema.value)
    return out
