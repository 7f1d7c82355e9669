"""OpenAI chat → AWS Bedrock Converse API.

Parity target: internal/translator/openai_awsbedrock.go (770 LoC):
message/tool/system mapping to the Converse schema (:103-284), the binary
event-stream → OpenAI SSE re-encode (:515-545), content-type rewrite to
text/event-stream (:286-298), and usage extraction from the `metadata`
stream event. Also the Bedrock embeddings variant
(openai_awsbedrock_embeddings.go → Titan invoke).
"""

from __future__ import annotations

import json
import time
from typing import Optional

from aigw.filterapi.config import APISchemaName
from aigw.translator.base import (
    RequestTranslation,
    ResponseTranslation,
    TranslationError,
    Translator,
    Usage,
    jdump,
    register,
)
from aigw.translator.eventstream import EventStreamDecoder
from aigw.translator.sse import DONE_EVENT, encode_data

BEDROCK_TO_OPENAI_STOP = {
    "end_turn": "stop",
    "stop_sequence": "stop",
    "max_tokens": "length",
    "tool_use": "tool_calls",
    "content_filtered": "content_filter",
    "guardrail_intervened": "content_filter",
}


def _usage_from_converse(u: dict) -> Usage:
    return Usage(
        input_tokens=u.get("inputTokens", 0) or 0,
        output_tokens=u.get("outputTokens", 0) or 0,
        total_tokens=u.get("totalTokens", 0) or 0,
        cached_input_tokens=u.get("cacheReadInputTokens", 0) or 0,
        cache_creation_input_tokens=u.get("cacheWriteInputTokens", 0) or 0,
    )


def _content_to_converse(content) -> list:
    if content is None:
        return []
    if isinstance(content, str):
        return [{"text": content}] if content else []
    blocks = []
    for part in content:
        t = part.get("type")
        if t == "text":
            blocks.append({"text": part.get("text", "")})
            # unified prompt caching (prompt-caching.md): a cache_control
            # breakpoint becomes a Converse cachePoint block AFTER the
            # cached content
            if part.get("cache_control"):
                blocks.append({"cachePoint": {"type": "default"}})
        elif t == "image_url":
            url = (part.get("image_url") or {}).get("url", "")
            if not url.startswith("data:"):
                raise TranslationError("Bedrock Converse requires base64 data-URI images")
            meta, _, b64 = url.partition(",")
            media = meta[5:].split(";")[0]
            fmt = media.split("/")[-1] if "/" in media else "png"
            blocks.append({"image": {"format": fmt, "source": {"bytes": b64}}})
        else:
            raise TranslationError(f"unsupported content part type {t!r} for Bedrock")
    return blocks


def openai_to_converse_request(body: dict) -> dict:
    """OpenAI ChatCompletionRequest → Bedrock ConverseRequest
    (openai_awsbedrock.go:103-284)."""
    out: dict = {}
    # unified `thinking` extension field (vendor-specific-fields.md):
    # Bedrock carries model-native fields in additionalModelRequestFields
    if isinstance(body.get("thinking"), dict):
        out["additionalModelRequestFields"] = {"thinking": body["thinking"]}
    system: list[dict] = []
    messages: list[dict] = []
    for msg in body.get("messages", []):
        role = msg.get("role")
        if role in ("system", "developer"):
            c = msg.get("content")
            if isinstance(c, list):
                system.extend({"text": p.get("text", "")} for p in c if p.get("type") == "text")
            elif c:
                system.append({"text": c})
            continue
        if role == "tool":
            inner = msg.get("content")
            if isinstance(inner, list):
                inner_blocks = _content_to_converse(inner)
            else:
                inner_blocks = [{"text": inner or ""}]
            block = {
                "toolResult": {
                    "toolUseId": msg.get("tool_call_id", ""),
                    "content": [
                        {"json": b} if "json" in b else b for b in inner_blocks
                    ],
                }
            }
            # Converse requires toolResult inside a user message; merge with
            # a preceding user message when adjacent.
            if messages and messages[-1]["role"] == "user" and any(
                "toolResult" in b for b in messages[-1]["content"]
            ):
                messages[-1]["content"].append(block)
            else:
                messages.append({"role": "user", "content": [block]})
            continue
        if role == "assistant":
            blocks = _content_to_converse(msg.get("content"))
            for tc in msg.get("tool_calls") or []:
                fn = tc.get("function") or {}
                try:
                    args = json.loads(fn.get("arguments") or "{}")
                except ValueError:
                    args = {}
                blocks.append(
                    {
                        "toolUse": {
                            "toolUseId": tc.get("id", ""),
                            "name": fn.get("name", ""),
                            "input": args,
                        }
                    }
                )
            messages.append({"role": "assistant", "content": blocks})
            continue
        if role == "user":
            messages.append({"role": "user", "content": _content_to_converse(msg.get("content"))})
            continue
        raise TranslationError(f"unsupported message role {role!r}")
    if system:
        out["system"] = system
    out["messages"] = messages

    inf: dict = {}
    max_tokens = body.get("max_completion_tokens") or body.get("max_tokens")
    if max_tokens:
        inf["maxTokens"] = max_tokens
    if body.get("temperature") is not None:
        inf["temperature"] = body["temperature"]
    if body.get("top_p") is not None:
        inf["topP"] = body["top_p"]
    stop = body.get("stop")
    if stop:
        inf["stopSequences"] = [stop] if isinstance(stop, str) else list(stop)
    if inf:
        out["inferenceConfig"] = inf

    tools = body.get("tools")
    if tools:
        specs = []
        for t in tools:
            if t.get("type") != "function":
                continue
            fn = t.get("function") or {}
            specs.append(
                {
                    "toolSpec": {
                        "name": fn.get("name", ""),
                        "description": fn.get("description", ""),
                        "inputSchema": {"json": fn.get("parameters") or {"type": "object"}},
                    }
                }
            )
        tc: dict = {"tools": specs}
        choice = body.get("tool_choice")
        if choice == "required":
            tc["toolChoice"] = {"any": {}}
        elif choice == "auto":
            tc["toolChoice"] = {"auto": {}}
        elif isinstance(choice, dict):
            tc["toolChoice"] = {"tool": {"name": (choice.get("function") or {}).get("name", "")}}
        if choice != "none":
            out["toolConfig"] = tc
    return out


def converse_to_openai_response(resp: dict, model: str) -> tuple[dict, Usage]:
    msg = (resp.get("output") or {}).get("message") or {}
    text_parts: list[str] = []
    reasoning: list[str] = []
    tool_calls: list[dict] = []
    for block in msg.get("content") or []:
        if "text" in block:
            text_parts.append(block["text"])
        elif "toolUse" in block:
            tu = block["toolUse"]
            tool_calls.append(
                {
                    "id": tu.get("toolUseId", ""),
                    "type": "function",
                    "function": {
                        "name": tu.get("name", ""),
                        "arguments": json.dumps(tu.get("input") or {}),
                    },
                }
            )
        elif "reasoningContent" in block:
            rt = (block["reasoningContent"].get("reasoningText") or {})
            reasoning.append(rt.get("text", ""))
    message: dict = {"role": "assistant", "content": "".join(text_parts) or None}
    if tool_calls:
        message["tool_calls"] = tool_calls
    if reasoning:
        message["reasoning_content"] = "".join(reasoning)
    usage = _usage_from_converse(resp.get("usage") or {})
    out = {
        "id": f"chatcmpl-bedrock-{int(time.time()*1000)}",
        "object": "chat.completion",
        "created": int(time.time()),
        "model": model,
        "choices": [
            {
                "index": 0,
                "message": message,
                "finish_reason": BEDROCK_TO_OPENAI_STOP.get(
                    resp.get("stopReason") or "end_turn", "stop"
                ),
            }
        ],
        "usage": {
            "prompt_tokens": usage.input_tokens,
            "completion_tokens": usage.output_tokens,
            "total_tokens": usage.total_tokens,
        },
    }
    return out, usage


@register("/v1/chat/completions", APISchemaName.AWS_BEDROCK)
class OpenAIToBedrockChat(Translator):
    def __init__(self, **kw):
        self.stream = False
        self._es = EventStreamDecoder()
        self._model = ""
        self._usage = Usage()
        self._finish: Optional[str] = None
        self._msg_id = f"chatcmpl-bedrock-{int(time.time()*1000)}"
        self._tool_index: dict[int, int] = {}
        self._next_tool = 0
        self._done = False

    def request(self, body, *, model_override="", stream=False, force_include_usage=False, raw=b""):
        if model_override:
            body["model"] = model_override
        self._model = body.get("model", "")
        self.stream = stream
        creq = openai_to_converse_request(body)
        verb = "converse-stream" if stream else "converse"
        return RequestTranslation(path=f"/model/{self._model}/{verb}", body=jdump(creq))

    def response_headers(self, status, headers):
        if self.stream:
            # event-stream arrives as application/vnd.amazon.eventstream;
            # the client gets OpenAI SSE (openai_awsbedrock.go:286-298).
            return {"content-type": "text/event-stream"}
        return {}

    def response_body(self, status, body):
        resp = json.loads(body)
        out, usage = converse_to_openai_response(resp, self._model)
        return ResponseTranslation(
            body=jdump(out), usage=usage, response_model=self._model, end_of_stream=True
        )

    def _chunk(self, delta: dict, finish=None, usage=None) -> bytes:
        c = {
            "id": self._msg_id,
            "object": "chat.completion.chunk",
            "created": int(time.time()),
            "model": self._model,
            "choices": [{"index": 0, "delta": delta, "finish_reason": finish}],
        }
        if usage is not None:
            c["usage"] = usage
        return encode_data(json.dumps(c, separators=(",", ":")))

    def response_chunk(self, chunk):
        """Bedrock ConverseStream event-stream → OpenAI SSE
        (openai_awsbedrock.go:515-545)."""
        out = bytearray()
        usage = None
        for msg in self._es.feed(chunk):
            if msg.message_type == "exception":
                raise ValueError(f"bedrock exception: {msg.exception_type}")
            try:
                data = json.loads(msg.payload) if msg.payload else {}
            except ValueError:
                continue
            et = msg.event_type
            if et == "messageStart":
                out.extend(self._chunk({"role": "assistant", "content": ""}))
            elif et == "contentBlockStart":
                start = (data.get("start") or {}).get("toolUse")
                if start:
                    ti = self._next_tool
                    self._next_tool += 1
                    self._tool_index[data.get("contentBlockIndex", 0)] = ti
                    out.extend(
                        self._chunk(
                            {
                                "tool_calls": [
                                    {
                                        "index": ti,
                                        "id": start.get("toolUseId", ""),
                                        "type": "function",
                                        "function": {
                                            "name": start.get("name", ""),
                                            "arguments": "",
                                        },
                                    }
                                ]
                            }
                        )
                    )
            elif et == "contentBlockDelta":
                delta = data.get("delta") or {}
                if "text" in delta:
                    out.extend(self._chunk({"content": delta["text"]}))
                elif "toolUse" in delta:
                    ti = self._tool_index.get(data.get("contentBlockIndex", 0), 0)
                    out.extend(
                        self._chunk(
                            {
                                "tool_calls": [
                                    {
                                        "index": ti,
                                        "function": {
                                            "arguments": delta["toolUse"].get("input", "")
                                        },
                                    }
                                ]
                            }
                        )
                    )
                elif "reasoningContent" in delta:
                    rc = delta["reasoningContent"]
                    if "text" in rc:
                        out.extend(self._chunk({"reasoning_content": rc["text"]}))
            elif et == "messageStop":
                self._finish = BEDROCK_TO_OPENAI_STOP.get(
                    data.get("stopReason") or "end_turn", "stop"
                )
            elif et == "metadata":
                self._usage.merge_max(_usage_from_converse(data.get("usage") or {}))
                usage = self._usage
                out.extend(
                    self._chunk(
                        {},
                        finish=self._finish or "stop",
                        usage={
                            "prompt_tokens": self._usage.input_tokens,
                            "completion_tokens": self._usage.output_tokens,
                            "total_tokens": self._usage.total_tokens,
                        },
                    )
                )
                out.extend(DONE_EVENT)
                self._done = True
        return ResponseTranslation(
            body=bytes(out), usage=usage, response_model=self._model, end_of_stream=self._done
        )


@register("/v1/embeddings", APISchemaName.AWS_BEDROCK)
class OpenAIToBedrockEmbeddings(Translator):
    """OpenAI embeddings → Bedrock Titan invoke
    (openai_awsbedrock_embeddings.go)."""

    def __init__(self, **kw):
        self._model = ""

    def request(self, body, *, model_override="", stream=False, force_include_usage=False, raw=b""):
        if model_override:
            body["model"] = model_override
        self._model = body.get("model", "")
        inp = body.get("input", "")
        if isinstance(inp, list):
            if len(inp) != 1 or not isinstance(inp[0], str):
                raise TranslationError("Bedrock Titan embeddings accept a single string input")
            inp = inp[0]
        breq = {"inputText": inp}
        if body.get("dimensions"):
            breq["dimensions"] = body["dimensions"]
        return RequestTranslation(path=f"/model/{self._model}/invoke", body=jdump(breq))

    def response_body(self, status, body):
        resp = json.loads(body)
        tokens = resp.get("inputTextTokenCount", 0) or 0
        usage = Usage(input_tokens=tokens, total_tokens=tokens)
        out = {
            "object": "list",
            "data": [
                {"object": "embedding", "index": 0, "embedding": resp.get("embedding") or []}
            ],
            "model": self._model,
            "usage": {"prompt_tokens": tokens, "total_tokens": tokens},
        }
        return ResponseTranslation(
            body=jdump(out), usage=usage, response_model=self._model, end_of_stream=True
        )
