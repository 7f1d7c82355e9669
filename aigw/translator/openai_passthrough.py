"""OpenAI-schema → OpenAI-schema translators (passthrough family).

Behavioral parity with internal/translator/openai_openai.go (chat),
openai_completions.go (legacy completions), openai_embeddings.go,
imagegeneration_openai_openai.go, and the Azure variants
(openai_azureopenai.go, openai_azureopenai_embeddings.go — path rewrite to
the deployments API). "Passthrough" still does real work: model override,
forced ``stream_options.include_usage`` when token costs are configured
(endpointspec.go:138-154), per-SSE-chunk usage extraction
(openai_openai.go:185-223), and null-body guards (:144-153).
"""

from __future__ import annotations

import json
from typing import Optional

from aigw.filterapi.config import APISchemaName
from aigw.translator.base import (
    RequestTranslation,
    ResponseTranslation,
    TranslationError,
    Translator,
    Usage,
    jdump,
    override_model,
    register,
    usage_from_openai,
)
from aigw.translator.sse import SSEDecoder


class _OpenAIPassthrough(Translator):
    PATH = "/v1/chat/completions"

    def __init__(self, path_prefix: str = "", api_version: str = "", **_ignored):
        self.path_prefix = path_prefix
        self.api_version = api_version
        self.stream = False
        self._sse = SSEDecoder()
        self._usage = Usage()
        self._model = ""

    # -- request --------------------------------------------------------------

    def _path(self, model: str) -> str:
        return self.path_prefix + self.PATH

    def request(self, body, *, model_override="", stream=False, force_include_usage=False, raw=b""):
        if not isinstance(body, dict):
            raise TranslationError("request body must be a JSON object")
        self.stream = stream
        needs_usage = stream and force_include_usage and not (
            (body.get("stream_options") or {}).get("include_usage")
        )
        if raw and not model_override and not needs_usage:
            # True passthrough: forward the client's bytes untouched
            # (openai_openai.go passthrough path — no re-serialization).
            self._model = body.get("model", "")
            return RequestTranslation(path=self._path(self._model), body=raw)
        model = override_model(body, model_override)
        self._model = model
        if needs_usage:
            so = body.get("stream_options") or {}
            so["include_usage"] = True
            body["stream_options"] = so
        return RequestTranslation(path=self._path(model), body=jdump(body))

    # -- response -------------------------------------------------------------

    def response_body(self, status, body):
        try:
            parsed = json.loads(body)
        except ValueError:
            return ResponseTranslation(body=body, end_of_stream=True)
        usage = usage_from_openai(parsed.get("usage") or {}) if isinstance(parsed, dict) else Usage()
        model = parsed.get("model", "") if isinstance(parsed, dict) else ""
        return ResponseTranslation(
            body=body, usage=usage, response_model=model, end_of_stream=True
        )

    def response_chunk(self, chunk):
        """Scan SSE lines for the usage chunk; bytes pass through verbatim
        (openai_openai.go:185-223 — passthrough does not re-serialize)."""
        out = bytearray(chunk)
        usage: Optional[Usage] = None
        model = ""
        for ev in self._sse.feed(chunk):
            if not ev.data or ev.data == "[DONE]":
                continue
            try:
                parsed = json.loads(ev.data)
            except ValueError:
                continue
            if isinstance(parsed, dict):
                if parsed.get("usage"):
                    u = usage_from_openai(parsed["usage"])
                    self._usage.merge_max(u)
                    usage = self._usage
                model = parsed.get("model", "") or model
        return ResponseTranslation(body=bytes(out), usage=usage, response_model=model)


@register("/v1/chat/completions", APISchemaName.OPENAI)
class OpenAIToOpenAIChat(_OpenAIPassthrough):
    PATH = "/v1/chat/completions"


@register("/v1/completions", APISchemaName.OPENAI)
class OpenAIToOpenAICompletions(_OpenAIPassthrough):
    PATH = "/v1/completions"


@register("/v1/embeddings", APISchemaName.OPENAI)
class OpenAIToOpenAIEmbeddings(_OpenAIPassthrough):
    PATH = "/v1/embeddings"


@register("/v1/images/generations", APISchemaName.OPENAI)
class OpenAIToOpenAIImages(_OpenAIPassthrough):
    PATH = "/v1/images/generations"


@register("/v1/responses", APISchemaName.OPENAI)
class OpenAIToOpenAIResponses(_OpenAIPassthrough):
    """OpenAI Responses API passthrough (openai_responses.go). Usage fields
    differ from chat (input_tokens/output_tokens instead of prompt/completion)."""

    PATH = "/v1/responses"

    @staticmethod
    def _usage_from_responses(u: dict) -> Usage:
        itd = u.get("input_tokens_details") or {}
        otd = u.get("output_tokens_details") or {}
        return Usage(
            input_tokens=u.get("input_tokens", 0) or 0,
            output_tokens=u.get("output_tokens", 0) or 0,
            total_tokens=u.get("total_tokens", 0) or 0,
            cached_input_tokens=itd.get("cached_tokens", 0) or 0,
            reasoning_tokens=otd.get("reasoning_tokens", 0) or 0,
        )

    def response_body(self, status, body):
        try:
            parsed = json.loads(body)
        except ValueError:
            return ResponseTranslation(body=body, end_of_stream=True)
        usage = self._usage_from_responses(parsed.get("usage") or {})
        return ResponseTranslation(
            body=body,
            usage=usage,
            response_model=parsed.get("model", ""),
            end_of_stream=True,
        )

    def response_chunk(self, chunk):
        usage = None
        model = ""
        for ev in self._sse.feed(chunk):
            if not ev.data or ev.data == "[DONE]":
                continue
            try:
                parsed = json.loads(ev.data)
            except ValueError:
                continue
            resp = parsed.get("response") if isinstance(parsed, dict) else None
            if isinstance(resp, dict):
                model = resp.get("model", "") or model
                if resp.get("usage"):
                    self._usage.merge_max(self._usage_from_responses(resp["usage"]))
                    usage = self._usage
        return ResponseTranslation(body=chunk, usage=usage, response_model=model)


@register("/v1/responses/input_tokens", APISchemaName.OPENAI)
class OpenAIResponsesInputTokens(_OpenAIPassthrough):
    """Responses input-token counting endpoint (endpointspec.go tokenize
    family; OpenAI + Azure variants)."""

    PATH = "/v1/responses/input_tokens"

    def response_body(self, status, body):
        try:
            parsed = json.loads(body)
            tokens = parsed.get("input_tokens", 0) or 0
        except ValueError:
            tokens = 0
        return ResponseTranslation(
            body=body,
            usage=Usage(input_tokens=tokens, total_tokens=tokens),
            end_of_stream=True,
        )


@register("/v1/responses/input_tokens", APISchemaName.AZURE_OPENAI)
class AzureResponsesInputTokens(OpenAIResponsesInputTokens):
    def _path(self, model):
        api_version = self.api_version or "2025-01-01-preview"
        return f"{self.path_prefix}/openai/responses/input_tokens?api-version={api_version}"


@register("/v1/audio/speech", APISchemaName.OPENAI)
class OpenAIToOpenAISpeech(_OpenAIPassthrough):
    PATH = "/v1/audio/speech"

    def response_body(self, status, body):
        # Binary audio; no usage envelope.
        return ResponseTranslation(body=body, end_of_stream=True)


class _AzurePathMixin:
    """Azure OpenAI deployments-API path rewrite (openai_azureopenai.go)."""

    AZURE_SUFFIX = "chat/completions"

    def _path(self, model: str) -> str:
        api_version = self.api_version or "2025-01-01-preview"
        return (
            f"{self.path_prefix}/openai/deployments/{model}/{self.AZURE_SUFFIX}"
            f"?api-version={api_version}"
        )


@register("/v1/chat/completions", APISchemaName.AZURE_OPENAI)
class OpenAIToAzureChat(_AzurePathMixin, _OpenAIPassthrough):
    AZURE_SUFFIX = "chat/completions"


@register("/v1/embeddings", APISchemaName.AZURE_OPENAI)
class OpenAIToAzureEmbeddings(_AzurePathMixin, _OpenAIPassthrough):
    AZURE_SUFFIX = "embeddings"


@register("/v1/completions", APISchemaName.AZURE_OPENAI)
class OpenAIToAzureCompletions(_AzurePathMixin, _OpenAIPassthrough):
    AZURE_SUFFIX = "completions"


@register("/v1/responses", APISchemaName.AZURE_OPENAI)
class OpenAIToAzureResponses(OpenAIToOpenAIResponses):
    def _path(self, model: str) -> str:
        api_version = self.api_version or "2025-01-01-preview"
        return f"{self.path_prefix}/openai/responses?api-version={api_version}"


@register("/tokenize", APISchemaName.OPENAI)
class OpenAIToVLLMTokenize(_OpenAIPassthrough):
    """Tokenize passthrough to a vLLM-style backend /tokenize
    (translator/tokenize.go:24-81 — the reference does NOT tokenize locally;
    our GPU tokenizer is exposed separately via /v1/gateway/tokenize)."""

    PATH = "/tokenize"

    def response_body(self, status, body):
        return ResponseTranslation(body=body, end_of_stream=True)
