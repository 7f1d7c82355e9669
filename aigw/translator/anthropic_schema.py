"""OpenAI ⇄ Anthropic schema conversion.

Behavioral parity with internal/translator/anthropic_helper.go (1,462 LoC)
and openai_helper.go (1,099 LoC): message/content/tool mapping in both
directions, stop-reason maps, usage maps, and the two stateful SSE machines
(Anthropic events → OpenAI chunks, and OpenAI chunks → Anthropic events,
openai_helper.go:516-698). Re-designed, not transliterated: conversions
operate on parsed dicts and the stream machines are explicit classes fed by
the shared SSE decoder.
"""

from __future__ import annotations

import json
import time
from dataclasses import dataclass, field
from typing import Optional

from aigw.translator.base import TranslationError, Usage, jdump

DEFAULT_MAX_TOKENS = 4096

# stop_reason maps (anthropic_helper.go stop-reason translation)
ANTHROPIC_TO_OPENAI_STOP = {
    "end_turn": "stop",
    "stop_sequence": "stop",
    "max_tokens": "length",
    "tool_use": "tool_calls",
    "refusal": "content_filter",
    "pause_turn": "stop",
}
OPENAI_TO_ANTHROPIC_STOP = {
    "stop": "end_turn",
    "length": "max_tokens",
    "tool_calls": "tool_use",
    "function_call": "tool_use",
    "content_filter": "refusal",
}


def usage_from_anthropic(u: dict) -> Usage:
    inp = u.get("input_tokens", 0) or 0
    out = u.get("output_tokens", 0) or 0
    cached = u.get("cache_read_input_tokens", 0) or 0
    creation = u.get("cache_creation_input_tokens", 0) or 0
    # Anthropic input_tokens excludes cache reads/creations; OpenAI
    # prompt_tokens includes them (anthropic_usage.go).
    total_in = inp + cached + creation
    return Usage(
        input_tokens=total_in,
        output_tokens=out,
        total_tokens=total_in + out,
        cached_input_tokens=cached,
        cache_creation_input_tokens=creation,
    )


# --- request: OpenAI chat -> Anthropic messages ------------------------------


def _content_to_anthropic(content) -> list:
    """OpenAI message content (str | [parts]) -> Anthropic content blocks."""
    if content is None:
        return []
    if isinstance(content, str):
        return [{"type": "text", "text": content}] if content else []
    blocks = []
    for part in content:
        t = part.get("type")
        if t == "text":
            block = {"type": "text", "text": part.get("text", "")}
            # provider-agnostic prompt caching: cache_control rides through
            # to Anthropic-family backends untouched (the reference's
            # unified cache_control API, capabilities/prompt-caching.md)
            if part.get("cache_control"):
                block["cache_control"] = part["cache_control"]
            blocks.append(block)
        elif t == "image_url":
            url = (part.get("image_url") or {}).get("url", "")
            if url.startswith("data:"):
                meta, _, b64 = url.partition(",")
                media = meta[5:].split(";")[0] or "image/png"
                blocks.append(
                    {
                        "type": "image",
                        "source": {"type": "base64", "media_type": media, "data": b64},
                    }
                )
            else:
                blocks.append({"type": "image", "source": {"type": "url", "url": url}})
        elif t == "input_audio":
            raise TranslationError("audio content is not supported by Anthropic backends")
        else:
            raise TranslationError(f"unsupported content part type {t!r}")
    return blocks


def openai_to_anthropic_request(body: dict) -> dict:
    """Translate an OpenAI ChatCompletionRequest to an Anthropic
    MessagesRequest (anthropic_helper.go request mapping)."""
    out: dict = {
        "model": body.get("model", ""),
        "max_tokens": body.get("max_completion_tokens")
        or body.get("max_tokens")
        or DEFAULT_MAX_TOKENS,
    }
    system_parts: list[str] = []
    messages: list[dict] = []
    for msg in body.get("messages", []):
        role = msg.get("role")
        if role in ("system", "developer"):
            c = msg.get("content")
            if isinstance(c, list):
                system_parts.extend(p.get("text", "") for p in c if p.get("type") == "text")
            elif c:
                system_parts.append(c)
            continue
        if role == "tool":
            content = msg.get("content")
            if isinstance(content, list):
                inner = _content_to_anthropic(content)
            else:
                inner = content or ""
            messages.append(
                {
                    "role": "user",
                    "content": [
                        {
                            "type": "tool_result",
                            "tool_use_id": msg.get("tool_call_id", ""),
                            "content": inner,
                        }
                    ],
                }
            )
            continue
        if role == "assistant":
            blocks = _content_to_anthropic(msg.get("content"))
            for tc in msg.get("tool_calls") or []:
                fn = tc.get("function") or {}
                try:
                    args = json.loads(fn.get("arguments") or "{}")
                except ValueError:
                    args = {}
                blocks.append(
                    {
                        "type": "tool_use",
                        "id": tc.get("id", ""),
                        "name": fn.get("name", ""),
                        "input": args,
                    }
                )
            messages.append({"role": "assistant", "content": blocks})
            continue
        if role == "user":
            messages.append({"role": "user", "content": _content_to_anthropic(msg.get("content"))})
            continue
        raise TranslationError(f"unsupported message role {role!r}")
    if system_parts:
        out["system"] = "\n".join(system_parts)
    out["messages"] = messages

    for src, dst in (("temperature", "temperature"), ("top_p", "top_p")):
        if body.get(src) is not None:
            out[dst] = body[src]
    stop = body.get("stop")
    if stop:
        out["stop_sequences"] = [stop] if isinstance(stop, str) else list(stop)
    if body.get("stream"):
        out["stream"] = True
    if body.get("metadata") and isinstance(body["metadata"], dict):
        user = body["metadata"].get("user_id") or body.get("user")
        if user:
            out["metadata"] = {"user_id": user}
    elif body.get("user"):
        out["metadata"] = {"user_id": body["user"]}

    tools = body.get("tools")
    if tools:
        out["tools"] = [
            {
                "name": (t.get("function") or {}).get("name", ""),
                "description": (t.get("function") or {}).get("description", ""),
                "input_schema": (t.get("function") or {}).get("parameters")
                or {"type": "object"},
            }
            for t in tools
            if t.get("type") == "function"
        ]
    choice = body.get("tool_choice")
    if choice:
        if choice == "auto":
            out["tool_choice"] = {"type": "auto"}
        elif choice == "required":
            out["tool_choice"] = {"type": "any"}
        elif choice == "none":
            out.pop("tools", None)
        elif isinstance(choice, dict):
            out["tool_choice"] = {
                "type": "tool",
                "name": (choice.get("function") or {}).get("name", ""),
            }
    # unified `thinking` extension field (vendor-specific-fields.md):
    # passed to Anthropic-family backends verbatim, overriding the
    # reasoning_effort mapping below
    if isinstance(body.get("thinking"), dict):
        out["thinking"] = body["thinking"]
    # reasoning_effort -> thinking budget (gcp/aws anthropic thinking map)
    effort = None if "thinking" in out else body.get("reasoning_effort")
    if effort:
        budgets = {"minimal": 1024, "low": 1024, "medium": 8192, "high": 24576}
        if effort in budgets:
            out["thinking"] = {"type": "enabled", "budget_tokens": budgets[effort]}
    return out


# --- response: Anthropic messages -> OpenAI chat -----------------------------


def anthropic_to_openai_response(resp: dict, created: Optional[int] = None) -> tuple[dict, Usage]:
    """Translate a unary Anthropic MessagesResponse to an OpenAI
    ChatCompletionResponse."""
    text_parts: list[str] = []
    reasoning_parts: list[str] = []
    tool_calls: list[dict] = []
    for block in resp.get("content") or []:
        t = block.get("type")
        if t == "text":
            text_parts.append(block.get("text", ""))
        elif t == "thinking":
            reasoning_parts.append(block.get("thinking", ""))
        elif t == "redacted_thinking":
            continue
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
    message: dict = {"role": "assistant", "content": "".join(text_parts) or None}
    if tool_calls:
        message["tool_calls"] = tool_calls
    if reasoning_parts:
        message["reasoning_content"] = "".join(reasoning_parts)
    usage = usage_from_anthropic(resp.get("usage") or {})
    out = {
        "id": resp.get("id", ""),
        "object": "chat.completion",
        "created": created or int(time.time()),
        "model": resp.get("model", ""),
        "choices": [
            {
                "index": 0,
                "message": message,
                "finish_reason": ANTHROPIC_TO_OPENAI_STOP.get(
                    resp.get("stop_reason") or "end_turn", "stop"
                ),
            }
        ],
        "usage": {
            "prompt_tokens": usage.input_tokens,
            "completion_tokens": usage.output_tokens,
            "total_tokens": usage.total_tokens,
            "prompt_tokens_details": {"cached_tokens": usage.cached_input_tokens},
        },
    }
    return out, usage


# --- streaming machine: Anthropic SSE events -> OpenAI chunks ----------------


@dataclass
class AnthropicToOpenAIStream:
    """Feeds on decoded Anthropic SSE events, emits OpenAI chat chunks.

    Mirrors the event mapping in anthropic→openai streaming translation;
    state (message id/model, per-index block kinds, cumulative usage) lives
    on the instance, one per request.
    """

    msg_id: str = ""
    model: str = ""
    created: int = 0
    usage: Usage = field(default_factory=Usage)
    finish_reason: str = ""
    _block_kinds: dict[int, str] = field(default_factory=dict)
    _tool_index: dict[int, int] = field(default_factory=dict)
    _next_tool: int = 0

    def _chunk(self, delta: dict, finish: Optional[str] = None, usage: Optional[dict] = None) -> dict:
        out = {
            "id": self.msg_id,
            "object": "chat.completion.chunk",
            "created": self.created or int(time.time()),
            "model": self.model,
            "choices": [{"index": 0, "delta": delta, "finish_reason": finish}],
        }
        if usage is not None:
            out["usage"] = usage
        return out

    def feed_event(self, event_type: str, data: dict) -> list[dict]:
        """Returns OpenAI chunk dicts to emit for one Anthropic event."""
        out: list[dict] = []
        if event_type == "message_start":
            msg = data.get("message") or {}
            self.msg_id = msg.get("id", "")
            self.model = msg.get("model", "")
            self.created = int(time.time())
            self.usage.merge_max(usage_from_anthropic(msg.get("usage") or {}))
            out.append(self._chunk({"role": "assistant", "content": ""}))
        elif event_type == "content_block_start":
            idx = data.get("index", 0)
            block = data.get("content_block") or {}
            kind = block.get("type", "text")
            self._block_kinds[idx] = kind
            if kind == "tool_use":
                ti = self._next_tool
                self._next_tool += 1
                self._tool_index[idx] = ti
                out.append(
                    self._chunk(
                        {
                            "tool_calls": [
                                {
                                    "index": ti,
                                    "id": block.get("id", ""),
                                    "type": "function",
                                    "function": {"name": block.get("name", ""), "arguments": ""},
                                }
                            ]
                        }
                    )
                )
        elif event_type == "content_block_delta":
            idx = data.get("index", 0)
            delta = data.get("delta") or {}
            dt = delta.get("type")
            if dt == "text_delta":
                out.append(self._chunk({"content": delta.get("text", "")}))
            elif dt == "thinking_delta":
                out.append(self._chunk({"reasoning_content": delta.get("thinking", "")}))
            elif dt == "input_json_delta":
                ti = self._tool_index.get(idx, 0)
                out.append(
                    self._chunk(
                        {
                            "tool_calls": [
                                {
                                    "index": ti,
                                    "function": {"arguments": delta.get("partial_json", "")},
                                }
                            ]
                        }
                    )
                )
        elif event_type == "message_delta":
            delta = data.get("delta") or {}
            stop = delta.get("stop_reason")
            if stop:
                self.finish_reason = ANTHROPIC_TO_OPENAI_STOP.get(stop, "stop")
            u = data.get("usage") or {}
            # message_delta carries CUMULATIVE output tokens (stream_fold.go).
            self.usage.merge_max(
                Usage(
                    output_tokens=u.get("output_tokens", 0) or 0,
                    input_tokens=self.usage.input_tokens,
                )
            )
        elif event_type == "message_stop":
            self.usage.total_tokens = self.usage.input_tokens + self.usage.output_tokens
            out.append(
                self._chunk(
                    {},
                    finish=self.finish_reason or "stop",
                    usage={
                        "prompt_tokens": self.usage.input_tokens,
                        "completion_tokens": self.usage.output_tokens,
                        "total_tokens": self.usage.total_tokens,
                        "prompt_tokens_details": {
                            "cached_tokens": self.usage.cached_input_tokens
                        },
                    },
                )
            )
        elif event_type == "error":
            raise TranslationError(json.dumps(data))
        return out


# --- streaming machine: OpenAI chunks -> Anthropic SSE events ----------------


@dataclass
class OpenAIToAnthropicStream:
    """Feeds on OpenAI chat chunks, emits Anthropic SSE events.

    Parity with openai_helper.go:516-698 (message_start /
    content_block_start / content_block_delta / content_block_stop /
    message_delta / message_stop synthesis), used when an Anthropic-native
    client is routed to an OpenAI backend.
    """

    started: bool = False
    text_block_open: bool = False
    tool_block_open: bool = False
    cur_tool_index: int = -1
    next_block_index: int = 0
    usage: Usage = field(default_factory=Usage)
    finish_reason: str = ""
    msg_id: str = ""
    model: str = ""

    def _ev(self, event_type: str, data: dict) -> tuple[str, dict]:
        data = dict(data)
        data["type"] = event_type
        return (event_type, data)

    def _close_block(self) -> list[tuple[str, dict]]:
        out = []
        if self.text_block_open or self.tool_block_open:
            out.append(self._ev("content_block_stop", {"index": self.next_block_index}))
            self.next_block_index += 1
            self.text_block_open = False
            self.tool_block_open = False
        return out

    def feed_chunk(self, chunk: dict) -> list[tuple[str, dict]]:
        out: list[tuple[str, dict]] = []
        if not self.started:
            self.started = True
            self.msg_id = chunk.get("id", "")
            self.model = chunk.get("model", "")
            out.append(
                self._ev(
                    "message_start",
                    {
                        "message": {
                            "id": self.msg_id,
                            "type": "message",
                            "role": "assistant",
                            "model": self.model,
                            "content": [],
                            "stop_reason": None,
                            "usage": {"input_tokens": 0, "output_tokens": 0},
                        }
                    },
                )
            )
        if chunk.get("usage"):
            from aigw.translator.base import usage_from_openai

            self.usage.merge_max(usage_from_openai(chunk["usage"]))
        for choice in chunk.get("choices") or []:
            delta = choice.get("delta") or {}
            content = delta.get("content")
            if content:
                if self.tool_block_open:
                    out.extend(self._close_block())
                if not self.text_block_open:
                    self.text_block_open = True
                    out.append(
                        self._ev(
                            "content_block_start",
                            {
                                "index": self.next_block_index,
                                "content_block": {"type": "text", "text": ""},
                            },
                        )
                    )
                out.append(
                    self._ev(
                        "content_block_delta",
                        {
                            "index": self.next_block_index,
                            "delta": {"type": "text_delta", "text": content},
                        },
                    )
                )
            for tc in delta.get("tool_calls") or []:
                ti = tc.get("index", 0)
                if tc.get("id"):
                    out.extend(self._close_block())
                    self.tool_block_open = True
                    self.cur_tool_index = ti
                    out.append(
                        self._ev(
                            "content_block_start",
                            {
                                "index": self.next_block_index,
                                "content_block": {
                                    "type": "tool_use",
                                    "id": tc.get("id", ""),
                                    "name": (tc.get("function") or {}).get("name", ""),
                                    "input": {},
                                },
                            },
                        )
                    )
                args = (tc.get("function") or {}).get("arguments")
                if args:
                    out.append(
                        self._ev(
                            "content_block_delta",
                            {
                                "index": self.next_block_index,
                                "delta": {"type": "input_json_delta", "partial_json": args},
                            },
                        )
                    )
            fr = choice.get("finish_reason")
            if fr:
                self.finish_reason = fr
        return out

    def finish(self) -> list[tuple[str, dict]]:
        out = self._close_block()
        out.append(
            self._ev(
                "message_delta",
                {
                    "delta": {
                        "stop_reason": OPENAI_TO_ANTHROPIC_STOP.get(
                            self.finish_reason or "stop", "end_turn"
                        ),
                        "stop_sequence": None,
                    },
                    "usage": {"output_tokens": self.usage.output_tokens},
                },
            )
        )
        out.append(self._ev("message_stop", {}))
        return out


def encode_anthropic_events(events: list[tuple[str, dict]]) -> bytes:
    """Anthropic SSE wire format: `event: <type>` + `data: <json>`."""
    out = bytearray()
    for event_type, data in events:
        out.extend(b"event: " + event_type.encode() + b"\n")
        out.extend(b"data: " + jdump(data) + b"\n\n")
    return bytes(out)


# --- stream folding (apischema/anthropic/stream_fold.go:16-119) --------------


def fold_anthropic_stream(events: list[tuple[str, dict]]) -> dict:
    """Fold a full Anthropic SSE event sequence back into a unary
    MessagesResponse (used for span recording and unary-from-stream needs).
    Hostile-index guard at 1000 blocks per the reference (:46-50)."""
    resp: dict = {}
    contents: dict[int, dict] = {}
    tool_json: dict[int, list[str]] = {}
    for event_type, data in events:
        if event_type == "message_start":
            resp = dict(data.get("message") or {})
            resp["content"] = []
        elif event_type == "content_block_start":
            idx = data.get("index", 0)
            if idx > 1000:
                raise TranslationError("content block index exceeds limit")
            contents[idx] = dict(data.get("content_block") or {})
        elif event_type == "content_block_delta":
            idx = data.get("index", 0)
            if idx > 1000:
                raise TranslationError("content block index exceeds limit")
            delta = data.get("delta") or {}
            block = contents.setdefault(idx, {"type": "text", "text": ""})
            dt = delta.get("type")
            if dt == "text_delta":
                block["text"] = block.get("text", "") + delta.get("text", "")
            elif dt == "thinking_delta":
                block["thinking"] = block.get("thinking", "") + delta.get("thinking", "")
            elif dt == "input_json_delta":
                tool_json.setdefault(idx, []).append(delta.get("partial_json", ""))
        elif event_type == "message_delta":
            delta = data.get("delta") or {}
            if delta.get("stop_reason"):
                resp["stop_reason"] = delta["stop_reason"]
            u = data.get("usage") or {}
            if u:
                ru = resp.setdefault("usage", {})
                for k, v in u.items():
                    if isinstance(v, int):
                        ru[k] = max(ru.get(k, 0) or 0, v)
    for idx in sorted(contents):
        block = contents[idx]
        if idx in tool_json:
            try:
                block["input"] = json.loads("".join(tool_json[idx]) or "{}")
            except ValueError:
                block["input"] = {}
        resp.setdefault("content", []).append(block)
    return resp
