"""Remaining endpoint translators: Cohere rerank, Anthropic count_tokens
family, audio transcription/translation multipart handling.

Parity targets: internal/translator/cohere_rerank_v2.go, counttokens_*.go,
openai_transcription.go / openai_translation.go + multipart_helper.go.
"""

from __future__ import annotations

import json
import re

from aigw.filterapi.config import APISchemaName
from aigw.translator.base import (
    RequestTranslation,
    ResponseTranslation,
    Translator,
    Usage,
    jdump,
    register,
)


@register("/v2/rerank", APISchemaName.COHERE)
class CohereRerank(Translator):
    """Cohere /v2/rerank passthrough with model override + billed-unit
    usage extraction (cohere_rerank_v2.go)."""

    def __init__(self, **kw):
        self._model = ""

    def request(self, body, *, model_override="", stream=False, force_include_usage=False, raw=b""):
        if model_override:
            body["model"] = model_override
        self._model = body.get("model", "")
        return RequestTranslation(path="/v2/rerank", body=jdump(body))

    def response_body(self, status, body):
        try:
            resp = json.loads(body)
            bu = ((resp.get("meta") or {}).get("billed_units") or {})
            usage = Usage(
                input_tokens=bu.get("input_tokens", 0) or 0,
                output_tokens=bu.get("output_tokens", 0) or 0,
                total_tokens=(bu.get("input_tokens", 0) or 0) + (bu.get("output_tokens", 0) or 0),
            )
        except ValueError:
            usage = Usage()
        return ResponseTranslation(
            body=body, usage=usage, response_model=self._model, end_of_stream=True
        )


class _CountTokensBase(Translator):
    """Anthropic /v1/messages/count_tokens family (counttokens_*.go)."""

    def __init__(self, api_version: str = "", gcp_project: str = "", gcp_region: str = ""):
        self.gcp_project = gcp_project
        self.gcp_region = gcp_region
        self._model = ""

    def _path(self, model):
        return "/v1/messages/count_tokens"

    def _adjust(self, body, model):
        return body

    def request(self, body, *, model_override="", stream=False, force_include_usage=False, raw=b""):
        if model_override:
            body["model"] = model_override
        self._model = body.get("model", "")
        return RequestTranslation(path=self._path(self._model), body=jdump(self._adjust(body, self._model)))

    def response_body(self, status, body):
        try:
            resp = json.loads(body)
            tokens = resp.get("input_tokens", 0) or 0
        except ValueError:
            tokens = 0
        return ResponseTranslation(
            body=body,
            usage=Usage(input_tokens=tokens, total_tokens=tokens),
            response_model=self._model,
            end_of_stream=True,
        )


@register("/anthropic/v1/messages/count_tokens", APISchemaName.ANTHROPIC)
class AnthropicCountTokens(_CountTokensBase):
    pass


@register("/anthropic/v1/messages/count_tokens", APISchemaName.GCP_ANTHROPIC)
class GCPAnthropicCountTokens(_CountTokensBase):
    def _path(self, model):
        return (
            f"/v1/projects/{self.gcp_project}/locations/{self.gcp_region}"
            f"/publishers/anthropic/models/count-tokens:rawPredict"
        )

    def _adjust(self, body, model):
        body["anthropic_version"] = "vertex-2023-10-16"
        return body


@register("/anthropic/v1/messages/count_tokens", APISchemaName.AWS_ANTHROPIC)
class AWSAnthropicCountTokens(_CountTokensBase):
    """count_tokens against Bedrock-hosted Anthropic
    (counttokens_awsanthropic.go: count-tokens invoke path)."""

    def _path(self, model):
        return f"/model/{model}/count-tokens"

    def _adjust(self, body, model):
        body.pop("model", None)
        body["anthropic_version"] = "bedrock-2023-05-31"
        return body


@register("/tokenize", APISchemaName.GCP_VERTEX_AI)
class GCPVertexTokenize(Translator):
    """/tokenize → Vertex :countTokens (tokenize_gcp.go)."""

    def __init__(self, api_version: str = "", gcp_project: str = "", gcp_region: str = ""):
        self.gcp_project = gcp_project
        self.gcp_region = gcp_region
        self._model = ""

    def request(self, body, *, model_override="", stream=False, force_include_usage=False, raw=b""):
        if model_override:
            body["model"] = model_override
        self._model = body.get("model", "")
        text = body.get("prompt") or body.get("text") or ""
        greq = {"contents": [{"role": "user", "parts": [{"text": text}]}]}
        path = (
            f"/v1/projects/{self.gcp_project}/locations/{self.gcp_region}"
            f"/publishers/google/models/{self._model}:countTokens"
        )
        return RequestTranslation(path=path, body=jdump(greq))

    def response_body(self, status, body):
        try:
            resp = json.loads(body)
            n = resp.get("totalTokens", 0) or 0
        except ValueError:
            n = 0
        return ResponseTranslation(
            body=jdump({"count": n, "tokens": []}),
            usage=Usage(input_tokens=n, total_tokens=n),
            response_model=self._model,
            end_of_stream=True,
        )


@register("/tokenize", APISchemaName.ANTHROPIC)
@register("/tokenize", APISchemaName.GCP_ANTHROPIC)
class AnthropicTokenize(Translator):
    """/tokenize → Anthropic count_tokens (tokenize path for Anthropic
    backends, tokenize.go family)."""

    def __init__(self, api_version: str = "", gcp_project: str = "", gcp_region: str = ""):
        self.gcp_project = gcp_project
        self.gcp_region = gcp_region
        self.is_gcp = bool(gcp_project)
        self._model = ""

    def request(self, body, *, model_override="", stream=False, force_include_usage=False, raw=b""):
        if model_override:
            body["model"] = model_override
        self._model = body.get("model", "")
        text = body.get("prompt") or body.get("text") or ""
        areq = {
            "model": self._model,
            "messages": [{"role": "user", "content": text}],
        }
        if self.is_gcp:
            areq.pop("model", None)
            areq["anthropic_version"] = "vertex-2023-10-16"
            path = (
                f"/v1/projects/{self.gcp_project}/locations/{self.gcp_region}"
                f"/publishers/anthropic/models/count-tokens:rawPredict"
            )
        else:
            path = "/v1/messages/count_tokens"
        return RequestTranslation(path=path, body=jdump(areq))

    def response_body(self, status, body):
        try:
            resp = json.loads(body)
            n = resp.get("input_tokens", 0) or 0
        except ValueError:
            n = 0
        return ResponseTranslation(
            body=jdump({"count": n, "tokens": []}),
            usage=Usage(input_tokens=n, total_tokens=n),
            response_model=self._model,
            end_of_stream=True,
        )


def _tokenize_chat_body(body: dict, model: str) -> dict:
    """Normalize a tokenize request to chat form: {messages} stays as-is,
    a completion-style {prompt}/{text} becomes one user message
    (tokenize_awsanthropic.go:84-101 ChatRequest synthesis)."""
    if isinstance(body.get("messages"), list):
        return {"model": model, "messages": body["messages"],
                **({"tools": body["tools"]} if body.get("tools") else {})}
    text = body.get("prompt") or body.get("text") or ""
    return {"model": model, "messages": [{"role": "user", "content": text}]}


def _aws_count_tokens_path(model: str) -> str:
    """Bedrock CountTokens path; cross-region inference prefixes (e.g.
    "us.anthropic.claude-...") are stripped because CountTokens rejects
    CRIS ids (anthropic_helper.go awsAnthropicCountTokensPath:1385-1400)."""
    bare = re.sub(r"^[a-z]{2}\.", "", model)
    return f"/model/{bare}/count-tokens"


@register("/tokenize", APISchemaName.AWS_BEDROCK)
class AWSBedrockTokenize(Translator):
    """/tokenize → Bedrock CountTokens in Converse form
    (tokenize_awsbedrock.go: {"input":{"converse":{...}}} and the
    inputTokens response field)."""

    def __init__(self, api_version: str = "", gcp_project: str = "", gcp_region: str = ""):
        self._model = ""

    def request(self, body, *, model_override="", stream=False,
                force_include_usage=False, raw=b""):
        from aigw.translator.chat_bedrock import openai_to_converse_request

        if model_override:
            body["model"] = model_override
        self._model = body.get("model", "")
        chat = _tokenize_chat_body(body, self._model)
        converse = openai_to_converse_request(chat)
        inner = {"messages": converse.get("messages", [])}
        if converse.get("system"):
            inner["system"] = converse["system"]
        if converse.get("toolConfig"):
            inner["toolConfig"] = converse["toolConfig"]
        return RequestTranslation(
            path=_aws_count_tokens_path(self._model),
            body=jdump({"input": {"converse": inner}}),
        )

    def response_body(self, status, body):
        try:
            n = json.loads(body).get("inputTokens", 0) or 0
        except ValueError:
            n = 0
        return ResponseTranslation(
            body=jdump({"count": n, "tokens": []}),
            usage=Usage(input_tokens=n, total_tokens=n),
            response_model=self._model,
            end_of_stream=True,
        )


@register("/tokenize", APISchemaName.AWS_ANTHROPIC)
class AWSAnthropicTokenize(Translator):
    """/tokenize → Bedrock CountTokens in InvokeModel form: the Anthropic
    count body rides base64-inside-JSON
    (tokenize_awsanthropic.go: {"input":{"invokeModel":{"body":"<b64>"}}};
    Bedrock validates the inner body as a real request, so it carries
    anthropic_version and a default max_tokens)."""

    def __init__(self, api_version: str = "", gcp_project: str = "", gcp_region: str = ""):
        self.api_version = api_version or "bedrock-2023-05-31"
        self._model = ""

    def request(self, body, *, model_override="", stream=False,
                force_include_usage=False, raw=b""):
        import base64

        from aigw.translator.anthropic_schema import openai_to_anthropic_request

        if model_override:
            body["model"] = model_override
        self._model = body.get("model", "")
        chat = _tokenize_chat_body(body, self._model)
        areq = openai_to_anthropic_request(chat)
        areq.pop("model", None)  # model travels in the URL path
        areq["anthropic_version"] = self.api_version
        areq["max_tokens"] = 1  # unconditional, tokenize_awsanthropic.go:69-73
        b64 = base64.b64encode(jdump(areq)).decode()
        return RequestTranslation(
            path=_aws_count_tokens_path(self._model),
            body=jdump({"input": {"invokeModel": {"body": b64}}}),
        )

    def response_body(self, status, body):
        try:
            doc = json.loads(body)
            n = doc.get("inputTokens", doc.get("input_tokens", 0)) or 0
        except ValueError:
            n = 0
        return ResponseTranslation(
            body=jdump({"count": n, "tokens": []}),
            usage=Usage(input_tokens=n, total_tokens=n),
            response_model=self._model,
            end_of_stream=True,
        )


# --- multipart audio endpoints -------------------------------------------------


def _parse_multipart(body: bytes, content_type: str) -> tuple[list[tuple[str, dict, bytes]], str]:
    """Minimal multipart/form-data parser (endpointspec.go:872-1076).
    Returns ([(name, part_headers, payload)], boundary)."""
    m = re.search(r'boundary="?([^";]+)"?', content_type)
    if not m:
        raise ValueError("missing multipart boundary")
    boundary = m.group(1)
    delim = b"--" + boundary.encode()
    parts = []
    segments = body.split(delim)
    for seg in segments[1:-1]:
        seg = seg.lstrip(b"\r\n")
        if not seg or seg.startswith(b"--"):
            continue
        header_blob, _, payload = seg.partition(b"\r\n\r\n")
        payload = payload[:-2] if payload.endswith(b"\r\n") else payload
        headers: dict[str, str] = {}
        name = ""
        for line in header_blob.split(b"\r\n"):
            k, _, v = line.partition(b":")
            headers[k.decode().lower().strip()] = v.decode().strip()
        cd = headers.get("content-disposition", "")
        nm = re.search(r'name="([^"]*)"', cd)
        if nm:
            name = nm.group(1)
        parts.append((name, headers, payload))
    return parts, boundary


def _encode_multipart(parts: list[tuple[str, dict, bytes]], boundary: str) -> bytes:
    out = bytearray()
    for name, headers, payload in parts:
        out.extend(b"--" + boundary.encode() + b"\r\n")
        for k, v in headers.items():
            out.extend(f"{k}: {v}\r\n".encode())
        out.extend(b"\r\n")
        out.extend(payload)
        out.extend(b"\r\n")
    out.extend(b"--" + boundary.encode() + b"--\r\n")
    return bytes(out)


class _MultipartAudio(Translator):
    """Audio transcription/translation: multipart body; model override
    requires re-encoding the form (multipart_helper.go)."""

    PATH = "/v1/audio/transcriptions"

    def __init__(self, **kw):
        self._model = ""
        self.content_type = ""

    def request_multipart(self, body: bytes, content_type: str, *, model_override=""):
        parts, boundary = _parse_multipart(body, content_type)
        model = ""
        for i, (name, headers, payload) in enumerate(parts):
            if name == "model":
                model = payload.decode("utf-8", "replace")
                if model_override:
                    parts[i] = (name, headers, model_override.encode())
                    model = model_override
        self._model = model
        if model_override:
            new_boundary = boundary
            body = _encode_multipart(parts, new_boundary)
        return RequestTranslation(
            path=self.PATH,
            body=body,
            headers={"content-type": f"multipart/form-data; boundary={boundary}"},
        ), model

    def request(self, body, *, model_override="", stream=False, force_include_usage=False, raw=b""):
        raise NotImplementedError("multipart endpoints use request_multipart")

    def response_body(self, status, body):
        try:
            resp = json.loads(body)
            u = resp.get("usage") or {}
            usage = Usage(
                input_tokens=u.get("input_tokens", 0) or 0,
                output_tokens=u.get("output_tokens", 0) or 0,
                total_tokens=u.get("total_tokens", 0) or 0,
            )
        except ValueError:
            usage = Usage()
        return ResponseTranslation(
            body=body, usage=usage, response_model=self._model, end_of_stream=True
        )


@register("/v1/audio/transcriptions", APISchemaName.OPENAI)
class OpenAITranscription(_MultipartAudio):
    PATH = "/v1/audio/transcriptions"


@register("/v1/audio/translations", APISchemaName.OPENAI)
class OpenAITranslation(_MultipartAudio):
    PATH = "/v1/audio/translations"

