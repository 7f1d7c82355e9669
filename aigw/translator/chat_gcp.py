"""OpenAI chat/embeddings → GCP Vertex AI (Gemini).

Parity target: internal/translator/openai_gcpvertexai.go + gemini_helper.go
(1,049 LoC): generateContent / streamGenerateContent mapping, role mapping
(assistant→model), parts mapping (text / inlineData / functionCall /
functionResponse), generationConfig, finishReason map, usageMetadata
extraction; plus openai_gcpvertexai_embeddings.go (:predict text-embedding).
"""

from __future__ import annotations

import json
import time

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
from aigw.translator.sse import DONE_EVENT, SSEDecoder, encode_data

GEMINI_TO_OPENAI_FINISH = {
    "STOP": "stop",
    "MAX_TOKENS": "length",
    "SAFETY": "content_filter",
    "RECITATION": "content_filter",
    "BLOCKLIST": "content_filter",
    "PROHIBITED_CONTENT": "content_filter",
    "SPII": "content_filter",
    "MALFORMED_FUNCTION_CALL": "tool_calls",
    "OTHER": "stop",
}


def _usage_from_gemini(u: dict) -> Usage:
    inp = u.get("promptTokenCount", 0) or 0
    out = u.get("candidatesTokenCount", 0) or 0
    reasoning = u.get("thoughtsTokenCount", 0) or 0
    return Usage(
        input_tokens=inp,
        output_tokens=out + reasoning,
        total_tokens=u.get("totalTokenCount", 0) or (inp + out + reasoning),
        cached_input_tokens=u.get("cachedContentTokenCount", 0) or 0,
        reasoning_tokens=reasoning,
    )


def _content_to_parts(content) -> list:
    if content is None:
        return []
    if isinstance(content, str):
        return [{"text": content}] if content else []
    parts = []
    for p in content:
        t = p.get("type")
        if t == "text":
            parts.append({"text": p.get("text", "")})
        elif t == "image_url":
            url = (p.get("image_url") or {}).get("url", "")
            if url.startswith("data:"):
                meta, _, b64 = url.partition(",")
                media = meta[5:].split(";")[0] or "image/png"
                parts.append({"inlineData": {"mimeType": media, "data": b64}})
            else:
                parts.append({"fileData": {"mimeType": "image/*", "fileUri": url}})
        else:
            raise TranslationError(f"unsupported content part type {t!r} for Gemini")
    return parts


def openai_to_gemini_request(body: dict) -> dict:
    out: dict = {}
    sys_parts: list[dict] = []
    contents: list[dict] = []
    tool_name_by_id: dict[str, str] = {}
    for msg in body.get("messages", []):
        role = msg.get("role")
        if role in ("system", "developer"):
            c = msg.get("content")
            if isinstance(c, list):
                sys_parts.extend({"text": p.get("text", "")} for p in c if p.get("type") == "text")
            elif c:
                sys_parts.append({"text": c})
            continue
        if role == "assistant":
            parts = _content_to_parts(msg.get("content"))
            for tc in msg.get("tool_calls") or []:
                fn = tc.get("function") or {}
                try:
                    args = json.loads(fn.get("arguments") or "{}")
                except ValueError:
                    args = {}
                tool_name_by_id[tc.get("id", "")] = fn.get("name", "")
                parts.append({"functionCall": {"name": fn.get("name", ""), "args": args}})
            contents.append({"role": "model", "parts": parts})
            continue
        if role == "tool":
            inner = msg.get("content")
            if isinstance(inner, list):
                inner = "".join(p.get("text", "") for p in inner if p.get("type") == "text")
            try:
                response_obj = json.loads(inner) if inner else {}
                if not isinstance(response_obj, dict):
                    response_obj = {"result": response_obj}
            except ValueError:
                response_obj = {"result": inner}
            name = tool_name_by_id.get(msg.get("tool_call_id", ""), msg.get("name", "tool"))
            contents.append(
                {
                    "role": "user",
                    "parts": [{"functionResponse": {"name": name, "response": response_obj}}],
                }
            )
            continue
        if role == "user":
            contents.append({"role": "user", "parts": _content_to_parts(msg.get("content"))})
            continue
        raise TranslationError(f"unsupported message role {role!r}")
    if sys_parts:
        out["systemInstruction"] = {"parts": sys_parts}
    out["contents"] = contents

    gen: dict = {}
    max_tokens = body.get("max_completion_tokens") or body.get("max_tokens")
    if max_tokens:
        gen["maxOutputTokens"] = max_tokens
    if body.get("temperature") is not None:
        gen["temperature"] = body["temperature"]
    if body.get("top_p") is not None:
        gen["topP"] = body["top_p"]
    stop = body.get("stop")
    if stop:
        gen["stopSequences"] = [stop] if isinstance(stop, str) else list(stop)
    if body.get("n"):
        gen["candidateCount"] = body["n"]
    rf = body.get("response_format")
    if isinstance(rf, dict) and rf.get("type") == "json_object":
        gen["responseMimeType"] = "application/json"
    # extension fields (vendor-specific-fields.md): unified `thinking` ->
    # generationConfig.thinkingConfig; `safetySettings` passed through
    th = body.get("thinking")
    if isinstance(th, dict):
        tc = {}
        if th.get("budget_tokens") is not None:
            tc["thinkingBudget"] = th["budget_tokens"]
        if th.get("type") == "enabled":
            tc.setdefault("includeThoughts", True)
        if tc:
            gen["thinkingConfig"] = tc
    if gen:
        out["generationConfig"] = gen
    if body.get("safetySettings"):
        out["safetySettings"] = body["safetySettings"]

    tools = body.get("tools")
    if tools:
        decls = []
        extra_tools = []
        for t in tools:
            if t.get("type") == "google_search":
                # Google Search grounding (vendor-specific-fields.md:
                # openai.go ToolTypeGoogleSearch) — config passes through
                extra_tools.append({"googleSearch": t.get("google_search") or {}})
                continue
            if t.get("type") != "function":
                continue
            fn = t.get("function") or {}
            decls.append(
                {
                    "name": fn.get("name", ""),
                    "description": fn.get("description", ""),
                    "parameters": fn.get("parameters") or {"type": "object"},
                }
            )
        out["tools"] = ([{"functionDeclarations": decls}] if decls else []) + extra_tools
        choice = body.get("tool_choice")
        mode = None
        allowed = None
        if choice == "none":
            mode = "NONE"
        elif choice == "required":
            mode = "ANY"
        elif choice == "auto":
            mode = "AUTO"
        elif isinstance(choice, dict):
            mode = "ANY"
            allowed = [(choice.get("function") or {}).get("name", "")]
        if mode:
            fcc: dict = {"mode": mode}
            if allowed:
                fcc["allowedFunctionNames"] = allowed
            out["toolConfig"] = {"functionCallingConfig": fcc}
    return out


def gemini_candidate_to_message(cand: dict) -> tuple[dict, str]:
    parts = (cand.get("content") or {}).get("parts") or []
    text: list[str] = []
    reasoning: list[str] = []
    tool_calls: list[dict] = []
    for i, p in enumerate(parts):
        if "text" in p:
            (reasoning if p.get("thought") else text).append(p["text"])
        elif "functionCall" in p:
            fc = p["functionCall"]
            tool_calls.append(
                {
                    "id": f"call_{i}_{fc.get('name','')}",
                    "type": "function",
                    "function": {
                        "name": fc.get("name", ""),
                        "arguments": json.dumps(fc.get("args") or {}),
                    },
                }
            )
    msg: dict = {"role": "assistant", "content": "".join(text) or None}
    if tool_calls:
        msg["tool_calls"] = tool_calls
    if reasoning:
        msg["reasoning_content"] = "".join(reasoning)
    finish = GEMINI_TO_OPENAI_FINISH.get(cand.get("finishReason") or "STOP", "stop")
    if tool_calls and finish == "stop":
        finish = "tool_calls"
    return msg, finish


@register("/v1/chat/completions", APISchemaName.GCP_VERTEX_AI)
class OpenAIToGeminiChat(Translator):
    def __init__(self, api_version: str = "", gcp_project: str = "", gcp_region: str = ""):
        self.gcp_project = gcp_project
        self.gcp_region = gcp_region
        self.stream = False
        self._sse = SSEDecoder()
        self._model = ""
        self._usage = Usage()
        self._msg_id = f"chatcmpl-gemini-{int(time.time()*1000)}"
        self._started = False
        self._finish = None
        self._done = False

    def request(self, body, *, model_override="", stream=False, force_include_usage=False, raw=b""):
        if model_override:
            body["model"] = model_override
        self._model = body.get("model", "")
        self.stream = stream
        greq = openai_to_gemini_request(body)
        verb = "streamGenerateContent?alt=sse" if stream else "generateContent"
        path = (
            f"/v1/projects/{self.gcp_project}/locations/{self.gcp_region}"
            f"/publishers/google/models/{self._model}:{verb}"
        )
        return RequestTranslation(path=path, body=jdump(greq))

    def response_body(self, status, body):
        resp = json.loads(body)
        cands = resp.get("candidates") or [{}]
        choices = []
        for i, cand in enumerate(cands):
            msg, finish = gemini_candidate_to_message(cand)
            choices.append({"index": i, "message": msg, "finish_reason": finish})
        usage = _usage_from_gemini(resp.get("usageMetadata") or {})
        out = {
            "id": self._msg_id,
            "object": "chat.completion",
            "created": int(time.time()),
            "model": resp.get("modelVersion", self._model),
            "choices": choices,
            "usage": {
                "prompt_tokens": usage.input_tokens,
                "completion_tokens": usage.output_tokens,
                "total_tokens": usage.total_tokens,
                "completion_tokens_details": {"reasoning_tokens": usage.reasoning_tokens},
            },
        }
        return ResponseTranslation(
            body=jdump(out),
            usage=usage,
            response_model=resp.get("modelVersion", self._model),
            end_of_stream=True,
        )

    def _chunk(self, delta, finish=None, usage=None) -> bytes:
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
        out = bytearray()
        usage = None
        for ev in self._sse.feed(chunk):
            if not ev.data:
                continue
            try:
                data = json.loads(ev.data)
            except ValueError:
                continue
            if not self._started:
                self._started = True
                out.extend(self._chunk({"role": "assistant", "content": ""}))
            for cand in data.get("candidates") or []:
                msg, finish = gemini_candidate_to_message(cand)
                delta: dict = {}
                if msg.get("content"):
                    delta["content"] = msg["content"]
                if msg.get("reasoning_content"):
                    delta["reasoning_content"] = msg["reasoning_content"]
                if msg.get("tool_calls"):
                    delta["tool_calls"] = [
                        {**tc, "index": i} for i, tc in enumerate(msg["tool_calls"])
                    ]
                if delta:
                    out.extend(self._chunk(delta))
                if cand.get("finishReason"):
                    self._finish = finish
            um = data.get("usageMetadata")
            if um:
                self._usage.merge_max(_usage_from_gemini(um))
            if self._finish and um and um.get("candidatesTokenCount") is not None:
                usage = self._usage
                out.extend(
                    self._chunk(
                        {},
                        finish=self._finish,
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
            body=bytes(out),
            usage=usage,
            response_model=self._model,
            end_of_stream=self._done,
        )


@register("/v1/embeddings", APISchemaName.GCP_VERTEX_AI)
class OpenAIToGCPEmbeddings(Translator):
    """OpenAI embeddings → Vertex text-embedding :predict
    (openai_gcpvertexai_embeddings.go)."""

    def __init__(self, api_version: str = "", gcp_project: str = "", gcp_region: str = ""):
        self.gcp_project = gcp_project
        self.gcp_region = gcp_region
        self._model = ""

    def request(self, body, *, model_override="", stream=False, force_include_usage=False, raw=b""):
        if model_override:
            body["model"] = model_override
        self._model = body.get("model", "")
        inp = body.get("input", "")
        texts = [inp] if isinstance(inp, str) else list(inp)
        greq = {"instances": [{"content": t} for t in texts]}
        if body.get("dimensions"):
            greq["parameters"] = {"outputDimensionality": body["dimensions"]}
        path = (
            f"/v1/projects/{self.gcp_project}/locations/{self.gcp_region}"
            f"/publishers/google/models/{self._model}:predict"
        )
        return RequestTranslation(path=path, body=jdump(greq))

    def response_body(self, status, body):
        resp = json.loads(body)
        data = []
        total_tokens = 0
        for i, pred in enumerate(resp.get("predictions") or []):
            emb = (pred.get("embeddings") or {})
            data.append(
                {"object": "embedding", "index": i, "embedding": emb.get("values") or []}
            )
            stats = emb.get("statistics") or {}
            total_tokens += stats.get("token_count", 0) or 0
        usage = Usage(input_tokens=total_tokens, total_tokens=total_tokens)
        out = {
            "object": "list",
            "data": data,
            "model": self._model,
            "usage": {"prompt_tokens": total_tokens, "total_tokens": total_tokens},
        }
        return ResponseTranslation(
            body=jdump(out), usage=usage, response_model=self._model, end_of_stream=True
        )
