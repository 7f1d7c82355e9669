"""Semantic-cache scoping: a cached answer must never cross models,
routes, generation parameters, or client credentials — even when two
requests embed to the SAME key vector (near-identical prompts). The GPU
embed index is faked with a single-slot dict so the scope isolation is
carried entirely by the value tag, which is exactly the property under
test (VERDICT r01 weak #5 / advisor finding)."""

import asyncio
import json

import yaml

from aigw.extproc.server import GatewayServer, _cache_fingerprint
from aigw.extproc.lean_front import serve_lean
from aigw.extproc.upstream_client import LeanClient
from aigw.filterapi.config import load_config
from aigw.filterapi.runtime import RuntimeConfig
from aigw.testing.fastmock import start_fast_mock


class FakeCacheGPU:
    """Worst-case embedding model: EVERY text maps to the same vector, so
    any two requests collide in key space. cache_enabled + single slot."""

    cache_enabled = True

    def __init__(self):
        self.slot = None

    async def cache_lookup_text(self, text):
        return self.slot, b"KEYVEC"

    async def cache_insert(self, vec, response):
        self.slot = response

    async def count_text_tokens(self, text):
        return 0


def _cfg(up_port):
    return load_config(yaml.safe_load(f"""
routes:
  - name: r
    backends:
      - name: b
        schema: OpenAI
        upstream: {{host: 127.0.0.1, port: {up_port}}}
"""))


def test_cache_tag_isolates_model_params_and_credentials():
    async def run():
        up_srv, up_port = await start_fast_mock("127.0.0.1", 0)
        server = GatewayServer(RuntimeConfig(_cfg(up_port)))
        server.gpu = FakeCacheGPU()
        _, port, cleanup = await serve_lean(server, "127.0.0.1", 0,
                                            with_fallback=False)
        client = LeanClient()

        async def chat(model, auth="Bearer user-a", **params):
            body = {"model": model,
                    "messages": [{"role": "user", "content": "same text"}]}
            body.update(params)
            r = await client.post(
                host="127.0.0.1", port=port, tls=False,
                path="/v1/chat/completions",
                headers={"content-type": "application/json",
                         "authorization": auth},
                body=json.dumps(body).encode(),
            )
            data = await r.read()
            hit = r.headers.get("x-aigw-cache") == "hit"
            r.release()
            assert r.status == 200, data
            return hit

        # prime the cache for (model-a, user-a, default params)
        assert not await chat("model-a")
        # identical scope -> hit
        assert await chat("model-a")
        # different model: same key vector, must NOT serve model-a's answer
        assert not await chat("model-b")
        # model-b's insert overwrote the single slot; re-prime model-a
        assert not await chat("model-a")
        assert await chat("model-a")
        # different sampling params -> miss
        assert not await chat("model-a", temperature=0.9)
        # different client credential -> miss (tenant isolation)
        assert not await chat("model-a", auth="Bearer user-b")

        await client.close()
        await cleanup()
        up_srv.close()

    asyncio.run(run())


def test_cache_fingerprint_properties():
    base = {"model": "m", "messages": [{"role": "user", "content": "x"}],
            "temperature": 0.5}
    h = {"authorization": "Bearer k"}
    fp = _cache_fingerprint("m", "r", base, h)
    assert len(fp) == 16
    # stable across message-content changes (semantic matching handles text)
    other = dict(base, messages=[{"role": "user", "content": "different"}])
    assert _cache_fingerprint("m", "r", other, h) == fp
    # sensitive to each scope dimension
    assert _cache_fingerprint("m2", "r", base, h) != fp
    assert _cache_fingerprint("m", "r2", base, h) != fp
    assert _cache_fingerprint("m", "r", dict(base, temperature=0.6), h) != fp
    assert _cache_fingerprint("m", "r", dict(base, tools=[{"type": "function"}]), h) != fp
    assert _cache_fingerprint("m", "r", base, {"authorization": "Bearer z"}) != fp
