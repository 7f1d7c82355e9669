"""TLS upstream end-to-end with a private CA (the BackendTLSPolicy
caCertificateRefs analogue): the lean client verifies the upstream's
chain against the configured trust anchor and sets SNI/hostname from
``upstream.hostname``. Verification is never disabled — a wrong CA must
fail the request (and fall back, if a fallback exists)."""

import asyncio
import json
import ssl
import subprocess

import pytest
import yaml
from aiohttp import web

from aigw.extproc.server import GatewayServer, run_server
from aigw.extproc.upstream_client import LeanClient
from aigw.filterapi.config import load_config
from aigw.filterapi.runtime import RuntimeConfig

OK = {
    "id": "x", "object": "chat.completion", "model": "m",
    "choices": [{"index": 0, "message": {"role": "assistant", "content": "tls-ok"},
                 "finish_reason": "stop"}],
    "usage": {"prompt_tokens": 1, "completion_tokens": 1, "total_tokens": 2},
}


def _mk_cert(tmp_path, cn="upstream.test"):
    key, crt = tmp_path / f"{cn}.key", tmp_path / f"{cn}.crt"
    subprocess.run(
        ["openssl", "req", "-x509", "-newkey", "rsa:2048", "-nodes",
         "-keyout", str(key), "-out", str(crt), "-days", "1",
         "-subj", f"/CN={cn}",
         "-addext", f"subjectAltName=DNS:{cn},IP:127.0.0.1"],
        check=True, capture_output=True,
    )
    return str(key), str(crt)


async def _tls_upstream(key, crt):
    async def chat(request):
        return web.json_response(OK)

    app = web.Application()
    app.router.add_post("/v1/chat/completions", chat)
    runner = web.AppRunner(app)
    await runner.setup()
    ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_SERVER)
    ctx.load_cert_chain(crt, key)
    site = web.TCPSite(runner, "127.0.0.1", 0, ssl_context=ctx)
    await site.start()
    return runner, site._server.sockets[0].getsockname()[1]


@pytest.mark.timeout(120)
def test_tls_upstream_with_private_ca(tmp_path):
    key, crt = _mk_cert(tmp_path)

    async def run():
        up_runner, up_port = await _tls_upstream(key, crt)
        cfg = load_config(yaml.safe_load(f"""
routes:
  - name: r
    backends:
      - name: b
        schema: OpenAI
        upstream:
          host: 127.0.0.1
          port: {up_port}
          tls: true
          hostname: upstream.test
          caFile: {crt}
"""))
        assert cfg.routes[0].backends[0].upstream.ca_file == crt
        server = GatewayServer(RuntimeConfig(cfg))
        runner = await run_server(server, host="127.0.0.1", port=0)
        port = runner.addresses[0][1]
        client = LeanClient()
        body = json.dumps({"model": "m", "messages": [{"role": "user", "content": "q"}]}).encode()
        r = await client.post(host="127.0.0.1", port=port, tls=False,
                              path="/v1/chat/completions",
                              headers={"content-type": "application/json"}, body=body)
        data = json.loads(await r.read())
        assert r.status == 200
        assert data["choices"][0]["message"]["content"] == "tls-ok"
        r.release()
        await client.close()
        await runner.cleanup()
        await server.close()
        await up_runner.cleanup()

    asyncio.run(run())


@pytest.mark.timeout(120)
def test_wrong_ca_fails_and_falls_back(tmp_path):
    key, crt = _mk_cert(tmp_path)
    _key2, crt2 = _mk_cert(tmp_path, cn="other.test")  # unrelated CA

    async def run():
        up_runner, up_port = await _tls_upstream(key, crt)

        async def plain_chat(request):
            return web.json_response(OK)

        plain = web.Application()
        plain.router.add_post("/v1/chat/completions", plain_chat)
        prunner = web.AppRunner(plain)
        await prunner.setup()
        psite = web.TCPSite(prunner, "127.0.0.1", 0)
        await psite.start()
        pport = psite._server.sockets[0].getsockname()[1]

        cfg = load_config(yaml.safe_load(f"""
routes:
  - name: r
    backends:
      - name: bad-ca
        schema: OpenAI
        upstream:
          host: 127.0.0.1
          port: {up_port}
          tls: true
          hostname: upstream.test
          caFile: {crt2}
      - name: plain-fallback
        priority: 1
        schema: OpenAI
        upstream: {{host: 127.0.0.1, port: {pport}}}
"""))
        server = GatewayServer(RuntimeConfig(cfg))
        runner = await run_server(server, host="127.0.0.1", port=0)
        port = runner.addresses[0][1]
        client = LeanClient()
        body = json.dumps({"model": "m", "messages": [{"role": "user", "content": "q"}]}).encode()
        r = await client.post(host="127.0.0.1", port=port, tls=False,
                              path="/v1/chat/completions",
                              headers={"content-type": "application/json"}, body=body)
        assert r.status == 200  # served by the fallback, not the bad-CA TLS origin
        assert json.loads(await r.read())["choices"][0]["message"]["content"] == "tls-ok"
        r.release()
        await client.close()
        await runner.cleanup()
        await server.close()
        await up_runner.cleanup()
        await prunner.cleanup()

    asyncio.run(run())


def test_backend_tls_policy_translates():
    from aigw.controller import translate_yaml

    cfg = translate_yaml(
        """
apiVersion: aigateway.envoyproxy.io/v1beta1
kind: AIGatewayRoute
metadata: {name: r, namespace: default}
spec:
  rules:
    - matches:
        - headers: [{type: Exact, name: x-ai-eg-model, value: m}]
      backendRefs: [{name: asb}]
---
apiVersion: aigateway.envoyproxy.io/v1beta1
kind: AIServiceBackend
metadata: {name: asb, namespace: default}
spec:
  schema: {name: OpenAI}
  backendRef: {name: up, kind: Backend, group: gateway.envoyproxy.io, port: 8443}
---
apiVersion: gateway.envoyproxy.io/v1alpha1
kind: Backend
metadata: {name: up, namespace: default}
spec:
  endpoints: [{ip: {address: 10.0.0.5, port: 8443}}]
---
apiVersion: gateway.networking.k8s.io/v1alpha3
kind: BackendTLSPolicy
metadata: {name: tls, namespace: default}
spec:
  targetRefs:
    - {group: gateway.envoyproxy.io, kind: Backend, name: up}
  validation:
    hostname: private.llm.internal
    caCertificateRefs:
      - {group: "", kind: ConfigMap, name: ca}
---
apiVersion: v1
kind: ConfigMap
metadata: {name: ca, namespace: default}
data:
  ca.crt: |
    -----BEGIN CERTIFICATE-----
    FAKEPEM
    -----END CERTIFICATE-----
"""
    )
    up = cfg.routes[0].backends[0].upstream
    assert up.tls is True
    assert up.hostname == "private.llm.internal"
    assert "FAKEPEM" in up.ca_pem
