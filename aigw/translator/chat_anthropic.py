"""OpenAI chat → Anthropic-family backends.

Parity targets: internal/translator/openai_gcpanthropic.go (Vertex
rawPredict/streamRawPredict), openai_awsanthropic.go (Bedrock invoke /
invoke-with-response-stream, event-stream wrapped), and the native Anthropic
API. All three share the schema conversion in anthropic_schema.py and
differ in path construction, version stamping, and the stream framing.
"""

from __future__ import annotations

import base64
import json

from aigw.filterapi.config import APISchemaName
from aigw.translator.anthropic_schema import (
    AnthropicToOpenAIStream,
    anthropic_to_openai_response,
    openai_to_anthropic_request,
    usage_from_anthropic,
)
from aigw.translator.base import (
    RequestTranslation,
    ResponseTranslation,
    Translator,
    Usage,
    jdump,
    register,
)
from aigw.translator.eventstream import EventStreamDecoder
from aigw.translator.sse import DONE_EVENT, SSEDecoder, encode_data


class _AnthropicBackedChat(Translator):
    """Common response-side machinery: Anthropic events → OpenAI chunks."""

    def __init__(self, api_version: str = "", gcp_project: str = "", gcp_region: str = ""):
        self.api_version = api_version
        self.gcp_project = gcp_project
        self.gcp_region = gcp_region
        self.stream = False
        self._sse = SSEDecoder()
        self._machine = AnthropicToOpenAIStream()
        self._model = ""

    # request path is vendor-specific
    def _path(self, model: str, stream: bool) -> str:
        raise NotImplementedError

    def _adjust_request(self, areq: dict, model: str) -> dict:
        return areq

    def request(self, body, *, model_override="", stream=False, force_include_usage=False, raw=b""):
        if model_override:
            body = dict(body)
            body["model"] = model_override
        self._model = body.get("model", "")
        self.stream = stream
        areq = openai_to_anthropic_request(body)
        areq = self._adjust_request(areq, self._model)
        return RequestTranslation(path=self._path(self._model, stream), body=jdump(areq))

    def response_headers(self, status, headers):
        if self.stream:
            return {"content-type": "text/event-stream"}
        return {}

    def response_body(self, status, body):
        resp = json.loads(body)
        out, usage = anthropic_to_openai_response(resp)
        out["model"] = self._model or out.get("model", "")
        return ResponseTranslation(
            body=jdump(out),
            usage=usage,
            response_model=resp.get("model", ""),
            end_of_stream=True,
        )

    def _emit_openai_chunks(self, events) -> ResponseTranslation:
        out = bytearray()
        usage = None
        done = False
        for event_type, data in events:
            for chunk in self._machine.feed_event(event_type, data):
                chunk["model"] = self._model or chunk.get("model", "")
                out.extend(encode_data(json.dumps(chunk, separators=(",", ":"))))
            if event_type == "message_stop":
                usage = self._machine.usage
                out.extend(DONE_EVENT)
                done = True
        return ResponseTranslation(
            body=bytes(out),
            usage=usage,
            response_model=self._machine.model,
            end_of_stream=done,
        )

    def response_chunk(self, chunk):
        events = []
        for ev in self._sse.feed(chunk):
            if not ev.data:
                continue
            try:
                data = json.loads(ev.data)
            except ValueError:
                continue
            events.append((ev.event or data.get("type", ""), data))
        return self._emit_openai_chunks(events)


@register("/v1/chat/completions", APISchemaName.ANTHROPIC)
class OpenAIToAnthropicChat(_AnthropicBackedChat):
    """OpenAI chat → native Anthropic /v1/messages."""

    def _path(self, model, stream):
        return "/v1/messages"


@register("/v1/chat/completions", APISchemaName.GCP_ANTHROPIC)
class OpenAIToGCPAnthropicChat(_AnthropicBackedChat):
    """OpenAI chat → Anthropic on Vertex (openai_gcpanthropic.go):
    rawPredict/streamRawPredict path, anthropic_version stamped in body,
    model dropped from the body (it lives in the URL)."""

    ANTHROPIC_VERSION = "vertex-2023-10-16"

    def _path(self, model, stream):
        verb = "streamRawPredict" if stream else "rawPredict"
        return (
            f"/v1/projects/{self.gcp_project}/locations/{self.gcp_region}"
            f"/publishers/anthropic/models/{model}:{verb}"
        )

    def _adjust_request(self, areq, model):
        areq.pop("model", None)
        areq["anthropic_version"] = self.ANTHROPIC_VERSION
        return areq


@register("/v1/chat/completions", APISchemaName.AWS_ANTHROPIC)
class OpenAIToAWSAnthropicChat(_AnthropicBackedChat):
    """OpenAI chat → Anthropic on Bedrock (openai_awsanthropic.go):
    /model/{id}/invoke[-with-response-stream]; anthropic_version
    "bedrock-2023-05-31"; stream flag dropped (framing is event-stream);
    response stream is AWS event-stream `chunk` events whose payload wraps
    the Anthropic event JSON in {"bytes": base64}."""

    ANTHROPIC_VERSION = "bedrock-2023-05-31"

    def __init__(self, **kw):
        super().__init__(**kw)
        self._es = EventStreamDecoder()

    def _path(self, model, stream):
        verb = "invoke-with-response-stream" if stream else "invoke"
        return f"/model/{model}/{verb}"

    def _adjust_request(self, areq, model):
        areq.pop("model", None)
        areq.pop("stream", None)
        areq["anthropic_version"] = self.ANTHROPIC_VERSION
        return areq

    def response_chunk(self, chunk):
        events = []
        for msg in self._es.feed(chunk):
            if msg.message_type == "exception":
                raise ValueError(f"bedrock exception: {msg.exception_type}")
            try:
                payload = json.loads(msg.payload)
                data = json.loads(base64.b64decode(payload.get("bytes", "")))
            except (ValueError, KeyError):
                continue
            events.append((data.get("type", ""), data))
        return self._emit_openai_chunks(events)


# ---- Anthropic-native endpoint (/anthropic/v1/messages) ----------------------


class _AnthropicPassthrough(Translator):
    """Anthropic client → Anthropic-family backend: body passthrough with
    model override + vendor path/version adjustments
    (anthropic_anthropic.go, anthropic_gcpanthropic.go,
    anthropic_awsanthropic.go)."""

    def __init__(self, api_version: str = "", gcp_project: str = "", gcp_region: str = ""):
        self.api_version = api_version
        self.gcp_project = gcp_project
        self.gcp_region = gcp_region
        self.stream = False
        self._sse = SSEDecoder()
        self._usage = Usage()
        self._model = ""

    def _path(self, model, stream):
        return "/v1/messages"

    def _adjust_request(self, body, model):
        return body

    def request(self, body, *, model_override="", stream=False, force_include_usage=False, raw=b""):
        if model_override:
            body["model"] = model_override
        self._model = body.get("model", "")
        self.stream = stream
        body = self._adjust_request(body, self._model)
        return RequestTranslation(path=self._path(self._model, stream), body=jdump(body))

    def response_body(self, status, body):
        try:
            resp = json.loads(body)
        except ValueError:
            return ResponseTranslation(body=body, end_of_stream=True)
        return ResponseTranslation(
            body=body,
            usage=usage_from_anthropic(resp.get("usage") or {}),
            response_model=resp.get("model", ""),
            end_of_stream=True,
        )

    def response_chunk(self, chunk):
        usage = None
        for ev in self._sse.feed(chunk):
            if not ev.data:
                continue
            try:
                data = json.loads(ev.data)
            except ValueError:
                continue
            t = ev.event or data.get("type", "")
            if t == "message_start":
                self._usage.merge_max(
                    usage_from_anthropic((data.get("message") or {}).get("usage") or {})
                )
            elif t == "message_delta":
                u = data.get("usage") or {}
                self._usage.merge_max(
                    Usage(
                        input_tokens=self._usage.input_tokens,
                        output_tokens=u.get("output_tokens", 0) or 0,
                    )
                )
            elif t == "message_stop":
                self._usage.total_tokens = (
                    self._usage.input_tokens + self._usage.output_tokens
                )
                usage = self._usage
        return ResponseTranslation(body=chunk, usage=usage, response_model=self._model)


@register("/anthropic/v1/messages", APISchemaName.ANTHROPIC)
class AnthropicToAnthropic(_AnthropicPassthrough):
    pass


@register("/anthropic/v1/messages", APISchemaName.GCP_ANTHROPIC)
class AnthropicToGCPAnthropic(_AnthropicPassthrough):
    def _path(self, model, stream):
        verb = "streamRawPredict" if stream else "rawPredict"
        return (
            f"/v1/projects/{self.gcp_project}/locations/{self.gcp_region}"
            f"/publishers/anthropic/models/{model}:{verb}"
        )

    def _adjust_request(self, body, model):
        body.pop("model", None)
        body["anthropic_version"] = "vertex-2023-10-16"
        return body


@register("/anthropic/v1/messages", APISchemaName.AWS_ANTHROPIC)
class AnthropicToAWSAnthropic(_AnthropicPassthrough):
    def __init__(self, **kw):
        super().__init__(**kw)
        self._es = EventStreamDecoder()

    def _path(self, model, stream):
        verb = "invoke-with-response-stream" if stream else "invoke"
        return f"/model/{model}/{verb}"

    def _adjust_request(self, body, model):
        body.pop("model", None)
        body.pop("stream", None)
        body["anthropic_version"] = "bedrock-2023-05-31"
        return body

    def response_headers(self, status, headers):
        if self.stream:
            return {"content-type": "text/event-stream"}
        return {}

    def response_chunk(self, chunk):
        """Re-encode Bedrock event-stream frames as Anthropic SSE."""
        out = bytearray()
        usage = None
        done = False
        for msg in self._es.feed(chunk):
            try:
                payload = json.loads(msg.payload)
                data = json.loads(base64.b64decode(payload.get("bytes", "")))
            except (ValueError, KeyError):
                continue
            t = data.get("type", "")
            out.extend(b"event: " + t.encode() + b"\n")
            out.extend(b"data: " + jdump(data) + b"\n\n")
            if t == "message_start":
                self._usage.merge_max(
                    usage_from_anthropic((data.get("message") or {}).get("usage") or {})
                )
            elif t == "message_delta":
                u = data.get("usage") or {}
                self._usage.merge_max(
                    Usage(
                        input_tokens=self._usage.input_tokens,
                        output_tokens=u.get("output_tokens", 0) or 0,
                    )
                )
            elif t == "message_stop":
                self._usage.total_tokens = (
                    self._usage.input_tokens + self._usage.output_tokens
                )
                usage = self._usage
                done = True
        return ResponseTranslation(body=bytes(out), usage=usage, end_of_stream=done)


@register("/anthropic/v1/messages", APISchemaName.OPENAI)
class AnthropicToOpenAIChat(Translator):
    """Anthropic client → OpenAI backend (anthropic_openai.go): request is
    converted Anthropic→OpenAI... which is the REVERSE of
    openai_to_anthropic_request; responses and streams are synthesized back
    into Anthropic events via OpenAIToAnthropicStream
    (openai_helper.go:516-698)."""

    def __init__(self, **kw):
        self.stream = False
        self._sse = SSEDecoder()
        from aigw.translator.anthropic_schema import OpenAIToAnthropicStream

        self._machine = OpenAIToAnthropicStream()
        self._model = ""

    def request(self, body, *, model_override="", stream=False, force_include_usage=False, raw=b""):
        if model_override:
            body["model"] = model_override
        self._model = body.get("model", "")
        self.stream = stream
        oreq = anthropic_to_openai_request(body)
        if stream:
            oreq["stream"] = True
            oreq["stream_options"] = {"include_usage": True}
        return RequestTranslation(path="/v1/chat/completions", body=jdump(oreq))

    def response_body(self, status, body):
        resp = json.loads(body)
        out, usage = openai_to_anthropic_response(resp)
        return ResponseTranslation(
            body=jdump(out),
            usage=usage,
            response_model=resp.get("model", ""),
            end_of_stream=True,
        )

    def response_chunk(self, chunk):
        from aigw.translator.anthropic_schema import encode_anthropic_events

        events = []
        done = False
        for ev in self._sse.feed(chunk):
            if not ev.data:
                continue
            if ev.data == "[DONE]":
                events.extend(self._machine.finish())
                done = True
                continue
            try:
                data = json.loads(ev.data)
            except ValueError:
                continue
            events.extend(self._machine.feed_chunk(data))
        usage = self._machine.usage if done else None
        return ResponseTranslation(
            body=encode_anthropic_events(events),
            usage=usage,
            response_model=self._machine.model,
            end_of_stream=done,
        )


@register("/anthropic/v1/messages", APISchemaName.AWS_BEDROCK)
class AnthropicToBedrockConverse(Translator):
    """Anthropic client → Bedrock Converse (anthropic_awsbedrock.go):
    composed from the shared conversions — Anthropic→OpenAI request,
    OpenAI→Converse request; Converse→OpenAI response→Anthropic response;
    for streams, the Bedrock event-stream→OpenAI-chunk machine feeds the
    OpenAI→Anthropic SSE synthesizer."""

    def __init__(self, **kw):
        from aigw.translator.anthropic_schema import OpenAIToAnthropicStream
        from aigw.translator.chat_bedrock import OpenAIToBedrockChat

        self._bedrock = OpenAIToBedrockChat()
        self._machine = OpenAIToAnthropicStream()
        self._sse = SSEDecoder()
        self._model = ""
        self.stream = False

    def request(self, body, *, model_override="", stream=False, force_include_usage=False, raw=b""):
        if model_override:
            body["model"] = model_override
        self._model = body.get("model", "")
        self.stream = stream
        oreq = anthropic_to_openai_request(body)
        if stream:
            oreq["stream"] = True
        return self._bedrock.request(oreq, stream=stream)

    def response_headers(self, status, headers):
        if self.stream:
            return {"content-type": "text/event-stream"}
        return {}

    def response_body(self, status, body):
        from aigw.translator.chat_bedrock import converse_to_openai_response

        resp = json.loads(body)
        oresp, _usage = converse_to_openai_response(resp, self._model)
        aresp, usage = openai_to_anthropic_response(oresp)
        return ResponseTranslation(
            body=jdump(aresp), usage=usage, response_model=self._model, end_of_stream=True
        )

    def response_chunk(self, chunk):
        from aigw.translator.anthropic_schema import encode_anthropic_events

        # Bedrock event-stream -> OpenAI SSE chunks (existing machine) ...
        inner = self._bedrock.response_chunk(chunk)
        events = []
        done = False
        for ev in self._sse.feed(inner.body):
            if not ev.data:
                continue
            if ev.data == "[DONE]":
                events.extend(self._machine.finish())
                done = True
                continue
            try:
                data = json.loads(ev.data)
            except ValueError:
                continue
            events.extend(self._machine.feed_chunk(data))
        usage = self._machine.usage if done else None
        return ResponseTranslation(
            body=encode_anthropic_events(events),
            usage=usage,
            response_model=self._model,
            end_of_stream=done,
        )


# --- Anthropic request -> OpenAI request (reverse direction) ------------------


def anthropic_to_openai_request(body: dict) -> dict:
    """Translate an Anthropic MessagesRequest to an OpenAI
    ChatCompletionRequest (anthropic_openai.go request direction)."""
    out: dict = {"model": body.get("model", "")}
    if body.get("max_tokens"):
        out["max_completion_tokens"] = body["max_tokens"]
    messages: list[dict] = []
    system = body.get("system")
    if system:
        if isinstance(system, list):
            text = "\n".join(b.get("text", "") for b in system if b.get("type") == "text")
        else:
            text = system
        messages.append({"role": "system", "content": text})
    for msg in body.get("messages", []):
        role = msg.get("role")
        content = msg.get("content")
        if isinstance(content, str):
            messages.append({"role": role, "content": content})
            continue
        parts: list[dict] = []
        tool_calls: list[dict] = []
        tool_results: list[dict] = []
        for block in content or []:
            t = block.get("type")
            if t == "text":
                parts.append({"type": "text", "text": block.get("text", "")})
            elif t == "image":
                src = block.get("source") or {}
                if src.get("type") == "base64":
                    url = f"data:{src.get('media_type','image/png')};base64,{src.get('data','')}"
                else:
                    url = src.get("url", "")
                parts.append({"type": "image_url", "image_url": {"url": url}})
            elif t == "tool_use":
                tool_calls.append(
                    {
                        "id": block.get("id", ""),
                        "type": "function",
                        "function": {
                            "name": block.get("name", ""),
                            "arguments": json.dumps(block.get("input") or {}),
                        },
                    }
                )
            elif t == "tool_result":
                inner = block.get("content")
                if isinstance(inner, list):
                    inner = "".join(b.get("text", "") for b in inner if b.get("type") == "text")
                tool_results.append(
                    {
                        "role": "tool",
                        "tool_call_id": block.get("tool_use_id", ""),
                        "content": inner or "",
                    }
                )
        if tool_results:
            messages.extend(tool_results)
        if role == "assistant":
            m: dict = {"role": "assistant"}
            text = "".join(p["text"] for p in parts if p.get("type") == "text")
            m["content"] = text or None
            if tool_calls:
                m["tool_calls"] = tool_calls
            messages.append(m)
        elif parts:
            messages.append({"role": "user", "content": parts})
    out["messages"] = messages
    for src, dst in (("temperature", "temperature"), ("top_p", "top_p")):
        if body.get(src) is not None:
            out[dst] = body[src]
    if body.get("stop_sequences"):
        out["stop"] = body["stop_sequences"]
    tools = body.get("tools")
    if tools:
        out["tools"] = [
            {
                "type": "function",
                "function": {
                    "name": t.get("name", ""),
                    "description": t.get("description", ""),
                    "parameters": t.get("input_schema") or {"type": "object"},
                },
            }
            for t in tools
        ]
    tc = body.get("tool_choice")
    if isinstance(tc, dict):
        kind = tc.get("type")
        if kind == "auto":
            out["tool_choice"] = "auto"
        elif kind == "any":
            out["tool_choice"] = "required"
        elif kind == "tool":
            out["tool_choice"] = {"type": "function", "function": {"name": tc.get("name", "")}}
    return out


def openai_to_anthropic_response(resp: dict) -> tuple[dict, Usage]:
    """OpenAI ChatCompletionResponse → Anthropic MessagesResponse."""
    from aigw.translator.anthropic_schema import OPENAI_TO_ANTHROPIC_STOP
    from aigw.translator.base import usage_from_openai

    choice = (resp.get("choices") or [{}])[0]
    msg = choice.get("message") or {}
    content: list[dict] = []
    if msg.get("content"):
        content.append({"type": "text", "text": msg["content"]})
    for tc in msg.get("tool_calls") or []:
        fn = tc.get("function") or {}
        try:
            args = json.loads(fn.get("arguments") or "{}")
        except ValueError:
            args = {}
        content.append(
            {"type": "tool_use", "id": tc.get("id", ""), "name": fn.get("name", ""), "input": args}
        )
    usage = usage_from_openai(resp.get("usage") or {})
    out = {
        "id": resp.get("id", ""),
        "type": "message",
        "role": "assistant",
        "model": resp.get("model", ""),
        "content": content,
        "stop_reason": OPENAI_TO_ANTHROPIC_STOP.get(
            choice.get("finish_reason") or "stop", "end_turn"
        ),
        "stop_sequence": None,
        "usage": {
            "input_tokens": usage.input_tokens - usage.cached_input_tokens,
            "output_tokens": usage.output_tokens,
            "cache_read_input_tokens": usage.cached_input_tokens,
            "cache_creation_input_tokens": 0,
        },
    }
    return out, usage
