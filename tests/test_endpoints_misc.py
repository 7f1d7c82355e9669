"""Endpoint coverage: images, rerank, tokenize passthrough, embeddings,
responses streaming, audio transcription multipart, debug tasks."""

import asyncio
import json

import aiohttp
from aiohttp import web

from aigw.extproc.server import GatewayServer, run_server
from aigw.filterapi import RuntimeConfig, load_config


class ExtraUpstream:
    """Fake provider pieces not in the shared mockupstream."""

    async def images(self, request):
        body = json.loads(await request.read())
        return web.json_response(
            {"created": 1, "data": [{"url": "https://img/1.png"}],
             "model": body.get("model", "")}
        )

    async def rerank(self, request):
        body = json.loads(await request.read())
        return web.json_response(
            {
                "results": [{"index": 1, "relevance_score": 0.9}],
                "meta": {"billed_units": {"input_tokens": 7}},
                "model": body.get("model", ""),
            }
        )

    async def tokenize(self, request):
        body = json.loads(await request.read())
        n = len(str(body.get("prompt", "")).split())
        return web.json_response({"count": n, "tokens": list(range(n))})

    async def responses(self, request):
        body = json.loads(await request.read())
        if body.get("stream"):
            resp = web.StreamResponse()
            resp.content_type = "text/event-stream"
            await resp.prepare(request)
            events = [
                {"type": "response.created", "response": {"id": "r1", "model": "m"}},
                {"type": "response.output_text.delta", "delta": "hi"},
                {
                    "type": "response.completed",
                    "response": {
                        "id": "r1",
                        "model": "m",
                        "usage": {"input_tokens": 4, "output_tokens": 2, "total_tokens": 6},
                    },
                },
            ]
            for e in events:
                await resp.write(b"data: " + json.dumps(e).encode() + b"\n\n")
            await resp.write_eof()
            return resp
        return web.json_response(
            {
                "id": "r1",
                "object": "response",
                "model": body.get("model", ""),
                "output": [{"type": "message", "content": [{"type": "output_text", "text": "ok"}]}],
                "usage": {"input_tokens": 4, "output_tokens": 2, "total_tokens": 6},
            }
        )

    async def transcriptions(self, request):
        raw = await request.read()
        assert b"whisper-large" in raw  # model override reached the form
        return web.json_response({"text": "transcribed", "usage": {"input_tokens": 5}})

    def make_app(self):
        app = web.Application()
        app.router.add_post("/v1/images/generations", self.images)
        app.router.add_post("/v2/rerank", self.rerank)
        app.router.add_post("/tokenize", self.tokenize)
        app.router.add_post("/v1/responses", self.responses)
        app.router.add_post("/v1/audio/transcriptions", self.transcriptions)
        return app


def _cfg(port):
    return load_config(
        {
            "version": "v1",
            "routes": [
                {
                    "name": "rerank",
                    "headers": [{"name": "x-ai-eg-model", "value": "rerank-v3"}],
                    "backends": [
                        {"name": "cohere", "schema": "Cohere",
                         "upstream": {"host": "127.0.0.1", "port": port}}
                    ],
                },
                {
                    "name": "audio",
                    "headers": [{"name": "x-ai-eg-model", "value": "whisper-1"}],
                    "backends": [
                        {"name": "oai", "schema": "OpenAI",
                         "upstream": {"host": "127.0.0.1", "port": port},
                         "modelNameOverride": "whisper-large"}
                    ],
                },
                {
                    "name": "r",
                    "backends": [
                        {"name": "b", "schema": "OpenAI",
                         "upstream": {"host": "127.0.0.1", "port": port}}
                    ],
                },
            ],
        }
    )


async def _env():
    up = ExtraUpstream()
    runner = web.AppRunner(up.make_app(), access_log=None)
    await runner.setup()
    site = web.TCPSite(runner, "127.0.0.1", 0)
    await site.start()
    port = runner.addresses[0][1]
    server = GatewayServer(RuntimeConfig(_cfg(port)))
    gw = await run_server(server, host="127.0.0.1", port=0)
    return runner, gw, gw.addresses[0][1]


def test_misc_endpoints():
    async def main():
        runner, gw, port = await _env()
        base = f"http://127.0.0.1:{port}"
        async with aiohttp.ClientSession() as c:
            async with c.post(f"{base}/v1/images/generations",
                              json={"model": "dall-e-3", "prompt": "cat"}) as r:
                assert r.status == 200
                assert (await r.json())["data"][0]["url"]
            async with c.post(f"{base}/v2/rerank",
                              json={"model": "rerank-v3", "query": "q",
                                    "documents": ["a", "b"]}) as r:
                assert r.status == 200
                assert (await r.json())["results"][0]["relevance_score"] == 0.9
            async with c.post(f"{base}/tokenize",
                              json={"model": "m", "prompt": "one two three"}) as r:
                assert (await r.json())["count"] == 3
            # responses API: unary + streamed
            async with c.post(f"{base}/v1/responses",
                              json={"model": "m", "input": "hello"}) as r:
                body = await r.json()
                assert body["usage"]["total_tokens"] == 6
            async with c.post(f"{base}/v1/responses",
                              json={"model": "m", "input": "hello", "stream": True}) as r:
                raw = await r.read()
                assert b"response.completed" in raw
            # multipart audio with model override re-encoding
            form = aiohttp.FormData()
            form.add_field("model", "whisper-1")
            form.add_field("file", b"\x00\x01audio", filename="a.wav",
                           content_type="audio/wav")
            async with c.post(f"{base}/v1/audio/transcriptions", data=form) as r:
                assert r.status == 200, await r.text()
                assert (await r.json())["text"] == "transcribed"
            async with c.get(f"{base}/debug/tasks") as r:
                assert r.status == 200
                assert (await r.json())["count"] >= 1
        await gw.cleanup()
        await runner.cleanup()

    asyncio.run(main())


def test_tokenize_aws_translators():
    """/tokenize -> AWS Bedrock CountTokens (Converse form) and AWS
    Anthropic CountTokens (base64 InvokeModel form) — parity:
    tokenize_awsbedrock.go / tokenize_awsanthropic.go, incl. cross-region
    model-id stripping and the inputTokens response mapping."""
    import base64

    from aigw.filterapi.config import APISchemaName
    from aigw.translator import get_translator

    # Bedrock Converse form, chat-shaped request
    tr = get_translator("/tokenize", APISchemaName.AWS_BEDROCK)
    res = tr.request({
        "model": "us.amazon.nova-pro-v1",
        "messages": [{"role": "system", "content": "be terse"},
                     {"role": "user", "content": "hello"}],
    }, model_override="", stream=False, force_include_usage=False)
    assert res.path == "/model/amazon.nova-pro-v1/count-tokens"  # CRIS stripped
    doc = json.loads(res.body)
    conv = doc["input"]["converse"]
    assert conv["messages"][0]["role"] == "user"
    assert conv["messages"][0]["content"][0]["text"] == "hello"
    assert conv["system"][0]["text"] == "be terse"
    out = tr.response_body(200, b'{"inputTokens": 17}')
    assert json.loads(out.body) == {"count": 17, "tokens": []}
    assert out.usage.input_tokens == 17

    # completion-shaped request becomes a single user message
    tr = get_translator("/tokenize", APISchemaName.AWS_BEDROCK)
    res = tr.request({"model": "m", "prompt": "count me"},
                     model_override="", stream=False, force_include_usage=False)
    conv = json.loads(res.body)["input"]["converse"]
    assert conv["messages"] == [{"role": "user",
                                 "content": [{"text": "count me"}]}]

    # AWS Anthropic InvokeModel form: inner body is base64 Anthropic JSON
    # with anthropic_version + max_tokens and NO model field
    tr = get_translator("/tokenize", APISchemaName.AWS_ANTHROPIC)
    res = tr.request({
        "model": "us.anthropic.claude-sonnet-4",
        "messages": [{"role": "user", "content": "hi"}],
    }, model_override="", stream=False, force_include_usage=False)
    assert res.path == "/model/anthropic.claude-sonnet-4/count-tokens"
    inner = json.loads(base64.b64decode(
        json.loads(res.body)["input"]["invokeModel"]["body"]))
    assert inner["anthropic_version"] == "bedrock-2023-05-31"
    assert inner["max_tokens"] == 1
    assert "model" not in inner
    assert inner["messages"][0]["content"] == "hi" or \
        inner["messages"][0]["content"][0].get("text") == "hi"
    out = tr.response_body(200, b'{"inputTokens": 9}')
    assert json.loads(out.body) == {"count": 9, "tokens": []}
