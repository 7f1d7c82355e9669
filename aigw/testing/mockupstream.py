"""Synthetic provider upstream for tests and benchmarks.

Role of the reference's testupstream
(tests/internal/testupstreamlib/testupstream.go:8-54): a fake provider whose
behavior is driven by request headers, letting data-plane behavior be
asserted end-to-end without a real provider. Speaks enough of each provider
schema to exercise every translator:

- OpenAI:   /v1/chat/completions (unary + SSE), /v1/completions,
            /v1/embeddings, /v1/models
- Anthropic:/v1/messages (unary + SSE)
- Bedrock:  /model/{id}/converse, /model/{id}/converse-stream (event-stream),
            /model/{id}/invoke[-with-response-stream]
- Gemini:   …:generateContent / :streamGenerateContent

Control headers:
- ``x-mock-status``: force an HTTP status for this request
- ``x-mock-fail-times``: fail (503) the first N requests per connection key
- ``x-mock-response-tokens``: number of words in the generated completion
- ``x-mock-delay-ms``: artificial upstream latency
- ``x-expected-auth``: assert the Authorization/x-api-key header, else 401
"""

from __future__ import annotations

import asyncio
import json
import time
from collections import defaultdict

from aiohttp import web

from aigw.translator.eventstream import encode_event

WORDS = ("the", "quick", "brown", "fox", "jumps", "over", "lazy", "dog")


def _gen_text(n_tokens: int) -> str:
    return " ".join(WORDS[i % len(WORDS)] for i in range(n_tokens))


class MockUpstream:
    def __init__(self):
        self.fail_counts: dict[str, int] = defaultdict(int)
        self.requests: list[dict] = []  # recorded for assertions
        self.record = False

    async def _common(self, request: web.Request):
        if self.record:
            self.requests.append(
                {
                    "path": request.path_qs,
                    "headers": dict(request.headers),
                    "body": await request.read(),
                }
            )
        delay = request.headers.get("x-mock-delay-ms")
        if delay:
            await asyncio.sleep(int(delay) / 1000.0)
        expected_auth = request.headers.get("x-expected-auth")
        if expected_auth:
            got = request.headers.get("authorization") or request.headers.get("x-api-key") or ""
            if got != expected_auth:
                return web.json_response(
                    {"error": {"message": f"bad auth {got!r}"}}, status=401
                )
        fail_times = int(request.headers.get("x-mock-fail-times", 0))
        if fail_times:
            key = request.headers.get("x-mock-fail-key", request.path)
            if self.fail_counts[key] < fail_times:
                self.fail_counts[key] += 1
                return web.json_response({"error": {"message": "injected failure"}}, status=503)
        status = request.headers.get("x-mock-status")
        if status:
            return web.json_response(
                {"error": {"message": "forced status", "type": "mock"}}, status=int(status)
            )
        return None

    @staticmethod
    def _ntokens(request, body: dict) -> int:
        h = request.headers.get("x-mock-response-tokens")
        return int(h) if h else 16

    # ---- OpenAI ---------------------------------------------------------------

    async def openai_chat(self, request: web.Request) -> web.StreamResponse:
        early = await self._common(request)
        if early is not None:
            return early
        body = json.loads(await request.read() or b"{}")
        n = self._ntokens(request, body)
        model = body.get("model", "mock-model")
        prompt_tokens = sum(
            len(str(m.get("content", "")).split()) for m in body.get("messages", [])
        )
        if body.get("stream"):
            resp = web.StreamResponse()
            resp.content_type = "text/event-stream"
            await resp.prepare(request)
            for i in range(n):
                chunk = {
                    "id": "chatcmpl-mock",
                    "object": "chat.completion.chunk",
                    "created": int(time.time()),
                    "model": model,
                    "choices": [
                        {
                            "index": 0,
                            "delta": {"content": WORDS[i % len(WORDS)] + " "}
                            if i
                            else {"role": "assistant", "content": ""},
                            "finish_reason": None,
                        }
                    ],
                }
                await resp.write(b"data: " + json.dumps(chunk).encode() + b"\n\n")
            final = {
                "id": "chatcmpl-mock",
                "object": "chat.completion.chunk",
                "created": int(time.time()),
                "model": model,
                "choices": [{"index": 0, "delta": {}, "finish_reason": "stop"}],
            }
            await resp.write(b"data: " + json.dumps(final).encode() + b"\n\n")
            if (body.get("stream_options") or {}).get("include_usage"):
                usage_chunk = {
                    "id": "chatcmpl-mock",
                    "object": "chat.completion.chunk",
                    "created": int(time.time()),
                    "model": model,
                    "choices": [],
                    "usage": {
                        "prompt_tokens": prompt_tokens,
                        "completion_tokens": n,
                        "total_tokens": prompt_tokens + n,
                    },
                }
                await resp.write(b"data: " + json.dumps(usage_chunk).encode() + b"\n\n")
            await resp.write(b"data: [DONE]\n\n")
            await resp.write_eof()
            return resp
        out = {
            "id": "chatcmpl-mock",
            "object": "chat.completion",
            "created": int(time.time()),
            "model": model,
            "choices": [
                {
                    "index": 0,
                    "message": {"role": "assistant", "content": _gen_text(n)},
                    "finish_reason": "stop",
                }
            ],
            "usage": {
                "prompt_tokens": prompt_tokens,
                "completion_tokens": n,
                "total_tokens": prompt_tokens + n,
            },
        }
        return web.json_response(out)

    async def openai_embeddings(self, request: web.Request) -> web.Response:
        early = await self._common(request)
        if early is not None:
            return early
        body = json.loads(await request.read() or b"{}")
        inp = body.get("input", "")
        texts = [inp] if isinstance(inp, str) else inp
        return web.json_response(
            {
                "object": "list",
                "data": [
                    {"object": "embedding", "index": i, "embedding": [0.1] * 8}
                    for i in range(len(texts))
                ],
                "model": body.get("model", "mock-embed"),
                "usage": {"prompt_tokens": 4 * len(texts), "total_tokens": 4 * len(texts)},
            }
        )

    # ---- Anthropic ------------------------------------------------------------

    async def anthropic_messages(self, request: web.Request) -> web.StreamResponse:
        early = await self._common(request)
        if early is not None:
            return early
        body = json.loads(await request.read() or b"{}")
        n = self._ntokens(request, body)
        model = body.get("model", "claude-mock")
        if body.get("stream"):
            resp = web.StreamResponse()
            resp.content_type = "text/event-stream"
            await resp.prepare(request)

            async def ev(t, d):
                d = dict(d)
                d["type"] = t
                await resp.write(
                    f"event: {t}\n".encode() + b"data: " + json.dumps(d).encode() + b"\n\n"
                )

            await ev(
                "message_start",
                {
                    "message": {
                        "id": "msg_mock",
                        "model": model,
                        "role": "assistant",
                        "usage": {"input_tokens": 9},
                    }
                },
            )
            await ev(
                "content_block_start",
                {"index": 0, "content_block": {"type": "text", "text": ""}},
            )
            for i in range(n):
                await ev(
                    "content_block_delta",
                    {"index": 0, "delta": {"type": "text_delta", "text": WORDS[i % 8] + " "}},
                )
            await ev("content_block_stop", {"index": 0})
            await ev(
                "message_delta",
                {"delta": {"stop_reason": "end_turn"}, "usage": {"output_tokens": n}},
            )
            await ev("message_stop", {})
            await resp.write_eof()
            return resp
        return web.json_response(
            {
                "id": "msg_mock",
                "type": "message",
                "role": "assistant",
                "model": model,
                "content": [{"type": "text", "text": _gen_text(n)}],
                "stop_reason": "end_turn",
                "usage": {"input_tokens": 9, "output_tokens": n},
            }
        )

    # ---- Bedrock --------------------------------------------------------------

    async def bedrock_converse(self, request: web.Request) -> web.Response:
        early = await self._common(request)
        if early is not None:
            return early
        body = json.loads(await request.read() or b"{}")
        n = self._ntokens(request, body)
        return web.json_response(
            {
                "output": {
                    "message": {"role": "assistant", "content": [{"text": _gen_text(n)}]}
                },
                "stopReason": "end_turn",
                "usage": {"inputTokens": 9, "outputTokens": n, "totalTokens": 9 + n},
            }
        )

    async def bedrock_converse_stream(self, request: web.Request) -> web.StreamResponse:
        early = await self._common(request)
        if early is not None:
            return early
        body = json.loads(await request.read() or b"{}")
        n = self._ntokens(request, body)
        resp = web.StreamResponse()
        resp.content_type = "application/vnd.amazon.eventstream"
        await resp.prepare(request)
        await resp.write(encode_event("messageStart", b'{"role":"assistant"}'))
        for i in range(n):
            await resp.write(
                encode_event(
                    "contentBlockDelta",
                    json.dumps(
                        {"contentBlockIndex": 0, "delta": {"text": WORDS[i % 8] + " "}}
                    ).encode(),
                )
            )
        await resp.write(encode_event("messageStop", json.dumps({"stopReason": "end_turn"}).encode()))
        await resp.write(
            encode_event(
                "metadata",
                json.dumps(
                    {"usage": {"inputTokens": 9, "outputTokens": n, "totalTokens": 9 + n}}
                ).encode(),
            )
        )
        await resp.write_eof()
        return resp

    # ---- Gemini ---------------------------------------------------------------

    async def gemini_generate(self, request: web.Request) -> web.Response:
        early = await self._common(request)
        if early is not None:
            return early
        body = json.loads(await request.read() or b"{}")
        n = self._ntokens(request, body)
        return web.json_response(
            {
                "candidates": [
                    {
                        "content": {"role": "model", "parts": [{"text": _gen_text(n)}]},
                        "finishReason": "STOP",
                    }
                ],
                "usageMetadata": {
                    "promptTokenCount": 9,
                    "candidatesTokenCount": n,
                    "totalTokenCount": 9 + n,
                },
                "modelVersion": "gemini-mock-001",
            }
        )

    def make_app(self) -> web.Application:
        app = web.Application(client_max_size=64 * 1024 * 1024)
        app.router.add_post("/v1/chat/completions", self.openai_chat)
        app.router.add_post("/v1/completions", self.openai_chat)
        app.router.add_post("/v1/embeddings", self.openai_embeddings)
        app.router.add_post("/v1/messages", self.anthropic_messages)
        app.router.add_post("/model/{model_id}/converse", self.bedrock_converse)
        app.router.add_post("/model/{model_id}/converse-stream", self.bedrock_converse_stream)
        app.router.add_post("/model/{model_id}/invoke", self.anthropic_messages)
        # Azure deployments-API shape
        app.router.add_post(
            "/openai/deployments/{deployment}/chat/completions", self.openai_chat
        )
        # Gemini (path contains ':'; aiohttp treats it literally)
        app.router.add_route(
            "POST",
            "/v1/projects/{proj}/locations/{loc}/publishers/google/models/{model_verb}",
            self.gemini_generate,
        )
        app.router.add_route(
            "POST",
            "/v1/projects/{proj}/locations/{loc}/publishers/anthropic/models/{model_verb}",
            self.anthropic_messages,
        )
        return app


async def start_mock_upstream(host="127.0.0.1", port=0):
    mock = MockUpstream()
    runner = web.AppRunner(mock.make_app(), access_log=None)
    await runner.setup()
    site = web.TCPSite(runner, host, port, reuse_address=True)
    await site.start()
    actual_port = runner.addresses[0][1]
    return mock, runner, actual_port
