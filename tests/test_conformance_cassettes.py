"""Schema-fidelity conformance suite driven by the vendored VCR
cassettes (tests/goldens/cassettes/ — real recorded provider traffic,
see README there).

Two layers, mirroring the reference's testopenai/VCR strategy
(/root/reference/tests/data-plane/vcr/, tests/internal/testopenai/):

1. END-TO-END REPLAY: every cassette request is POSTed through a real
   gateway (lean front) to a scripted upstream that replays the
   cassette's recorded response verbatim. The gateway must forward the
   request body semantically unchanged (OpenAI passthrough) and deliver
   the response back with byte-level JSON fidelity — unary bodies
   JSON-equal, SSE streams event-sequence-equal — for EVERY recorded
   shape: tool calls, multimodal content parts, logprobs, reasoning,
   json-mode, base64 embeddings, error envelopes.

2. TRANSLATION MATRIX SNAPSHOTS: every cassette request is translated
   through every registered provider translator and compared against
   committed goldens (tests/goldens/conformance_matrix.json,
   regenerated via scripts/regen_conformance.py). This locks the full
   translator matrix against the same exotic unions.
"""

from __future__ import annotations

import asyncio
import glob
import json
import os
import urllib.parse

import pytest
import yaml

from aigw.extproc.lean_front import serve_lean
from aigw.extproc.server import GatewayServer
from aigw.extproc.upstream_client import LeanClient
from aigw.filterapi import RuntimeConfig, load_config
from aigw.filterapi.config import APISchemaName
from aigw.translator import TranslationError, get_translator

CASSETTE_DIR = os.path.join(os.path.dirname(__file__), "goldens", "cassettes")
MATRIX_PATH = os.path.join(os.path.dirname(__file__), "goldens",
                           "conformance_matrix.json")

# endpoint -> provider schemas to snapshot (A.9 registry; unregistered
# pairs are skipped at runtime so the matrix mirrors the real registry)
MATRIX_SCHEMAS = {
    "/v1/chat/completions": [
        APISchemaName.OPENAI, APISchemaName.AWS_BEDROCK,
        APISchemaName.AWS_ANTHROPIC, APISchemaName.AZURE_OPENAI,
        APISchemaName.GCP_VERTEX_AI, APISchemaName.GCP_ANTHROPIC,
        APISchemaName.ANTHROPIC,
    ],
    "/v1/completions": [APISchemaName.OPENAI, APISchemaName.AZURE_OPENAI],
    "/v1/embeddings": [
        APISchemaName.OPENAI, APISchemaName.AWS_BEDROCK,
        APISchemaName.AZURE_OPENAI, APISchemaName.GCP_VERTEX_AI,
    ],
    "/v1/images/generations": [APISchemaName.OPENAI],
}


def load_cassettes():
    """[(name, path, request_body_bytes, status, resp_content_type,
    resp_body_bytes)] for every vendored interaction."""
    out = []
    for f in sorted(glob.glob(os.path.join(CASSETTE_DIR, "*.yaml"))):
        name = os.path.basename(f)[:-5]
        doc = yaml.safe_load(open(f, encoding="utf-8"))
        for it in doc.get("interactions", []):
            req = it["request"]
            resp = it["response"]
            path = urllib.parse.urlparse(req["url"]).path
            ctype = ""
            for k, v in (resp.get("headers") or {}).items():
                if k.lower() == "content-type":
                    ctype = v[0] if isinstance(v, list) else v
            req_ctype = ""
            for k, v in (req.get("headers") or {}).items():
                if k.lower() == "content-type":
                    req_ctype = v[0] if isinstance(v, list) else v
            out.append({
                "name": f"{name}#{it.get('id', 0)}",
                "path": path,
                "req_body": (req.get("body") or "").encode(),
                "req_ctype": req_ctype,
                "status": int(resp.get("code", 200)),
                "resp_ctype": ctype,
                "resp_body": (resp.get("body") or "").encode(),
            })
    return out


def _json_endpoint_cases():
    """Cassettes replayable through the gateway's JSON endpoints."""
    from aigw.extproc.server import JSON_ENDPOINTS

    return [c for c in load_cassettes()
            if c["path"] in JSON_ENDPOINTS and
            c["req_ctype"].startswith("application/json")]


# ---------------------------------------------------------------------------
# layer 2: translation-matrix snapshots


def build_matrix() -> dict:
    """case key -> {"path":..., "body":..., "headers":...} or {"error":...}.
    Pure function of the translators; used by the regen script and the
    comparison test."""
    matrix: dict = {}
    for c in load_cassettes():
        schemas = MATRIX_SCHEMAS.get(c["path"])
        if not schemas or not c["req_ctype"].startswith("application/json"):
            continue
        try:
            body = json.loads(c["req_body"])
        except ValueError:
            continue
        if not isinstance(body, dict):
            continue
        stream = bool(body.get("stream"))
        for schema in schemas:
            key = f'{c["name"]}|{c["path"]}|{schema.value}'
            try:
                tr = get_translator(
                    c["path"], schema,
                    gcp_project="proj", gcp_region="us-central1",
                )
            except TranslationError:
                continue
            try:
                res = tr.request(
                    json.loads(c["req_body"]),  # fresh copy per translator
                    model_override="",
                    stream=stream,
                    force_include_usage=False,
                )
                matrix[key] = {
                    "path": res.path,
                    "headers": dict(sorted(res.headers.items())),
                    "body": json.loads(res.body) if res.body else None,
                }
            except TranslationError as e:
                matrix[key] = {"error": str(e)}
    return matrix


def test_translation_matrix_matches_goldens():
    assert os.path.exists(MATRIX_PATH), (
        "run scripts/regen_conformance.py to create the matrix goldens"
    )
    want = json.load(open(MATRIX_PATH, encoding="utf-8"))
    got = build_matrix()
    assert len(got) >= 100, f"matrix shrank to {len(got)} cases"
    missing = set(want) - set(got)
    extra = set(got) - set(want)
    assert not missing, f"cases disappeared: {sorted(missing)[:5]}"
    assert not extra, f"unreviewed new cases: {sorted(extra)[:5]} (regen goldens)"
    for key in sorted(want):
        assert got[key] == want[key], (
            f"translation drifted for {key}:\n"
            f"want {json.dumps(want[key], sort_keys=True)[:400]}\n"
            f"got  {json.dumps(got[key], sort_keys=True)[:400]}"
        )


# ---------------------------------------------------------------------------
# layer 1: end-to-end replay through a real gateway


class ReplayUpstream:
    """Serves one cassette response per request, recording request bodies.
    SSE responses are replayed as chunked transfer split at event
    boundaries (the provider's real framing shape)."""

    def __init__(self):
        self.response = None  # (status, ctype, body)
        self.requests: list[bytes] = []
        self._server = None

    async def start(self):
        self._server = await asyncio.start_server(self._handle, "127.0.0.1", 0)
        return self._server.sockets[0].getsockname()[1]

    async def _handle(self, reader, writer):
        try:
            while True:
                head = b""
                while b"\r\n\r\n" not in head:
                    chunk = await reader.read(65536)
                    if not chunk:
                        return
                    head += chunk
                head_part, _, rest = head.partition(b"\r\n\r\n")
                clen = 0
                for line in head_part.split(b"\r\n"):
                    if line.lower().startswith(b"content-length:"):
                        clen = int(line.split(b":", 1)[1])
                while len(rest) < clen:
                    rest += await reader.read(65536)
                self.requests.append(rest[:clen])
                status, ctype, body = self.response
                if ctype.startswith("text/event-stream"):
                    writer.write(
                        b"HTTP/1.1 %d X\r\ncontent-type: %s\r\n"
                        b"transfer-encoding: chunked\r\n\r\n"
                        % (status, ctype.encode())
                    )
                    for ev in body.split(b"\n\n"):
                        if not ev.strip():
                            continue
                        data = ev + b"\n\n"
                        writer.write(b"%x\r\n" % len(data) + data + b"\r\n")
                        await writer.drain()
                    writer.write(b"0\r\n\r\n")
                else:
                    writer.write(
                        b"HTTP/1.1 %d X\r\ncontent-type: %s\r\n"
                        b"content-length: %d\r\n\r\n"
                        % (status, (ctype or "application/json").encode(), len(body))
                    )
                    writer.write(body)
                await writer.drain()
        except (ConnectionResetError, BrokenPipeError):
            pass
        finally:
            writer.close()

    def stop(self):
        self._server.close()


def _sse_events(raw: bytes) -> list:
    """data payloads of an SSE byte stream, JSON-decoded where possible."""
    events = []
    for block in raw.split(b"\n\n"):
        for line in block.split(b"\n"):
            if line.startswith(b"data:"):
                payload = line[5:].strip()
                try:
                    events.append(json.loads(payload))
                except ValueError:
                    events.append(payload.decode("utf-8", "replace"))
    return events


@pytest.mark.timeout(300)
def test_cassette_replay_end_to_end():
    cases = _json_endpoint_cases()
    assert len(cases) >= 35, f"only {len(cases)} replayable cassettes"

    async def run():
        up = ReplayUpstream()
        up_port = await up.start()
        cfg = load_config(yaml.safe_load(f"""
routes:
  - name: conformance
    backends:
      - name: openai
        schema: OpenAI
        upstream: {{host: 127.0.0.1, port: {up_port}}}
"""))
        server = GatewayServer(RuntimeConfig(cfg))
        _, gw_port, cleanup = await serve_lean(server, "127.0.0.1", 0,
                                               with_fallback=False)
        client = LeanClient()
        replayed = 0
        for c in cases:
            up.response = (c["status"], c["resp_ctype"], c["resp_body"])
            up.requests.clear()
            r = await client.post(
                host="127.0.0.1", port=gw_port, tls=False, path=c["path"],
                headers={"content-type": "application/json"},
                body=c["req_body"], timeout_s=30.0,
            )
            got = await r.read()
            r.release()
            if c["status"] >= 400:
                # provider-side errors: the gateway may also reject the
                # malformed request locally; either way the client must
                # see a 4xx and a JSON error envelope
                assert 400 <= r.status < 500, (c["name"], r.status, got[:200])
                assert json.loads(got), c["name"]
                replayed += 1
                continue
            assert r.status == c["status"], (c["name"], r.status, got[:300])
            if up.requests:
                # passthrough: the upstream saw the client's body
                # semantically unchanged
                assert json.loads(up.requests[0]) == json.loads(c["req_body"]), c["name"]
            if c["resp_ctype"].startswith("text/event-stream"):
                assert _sse_events(got) == _sse_events(c["resp_body"]), (
                    f'{c["name"]}: SSE event sequence drifted')
            else:
                assert json.loads(got) == json.loads(c["resp_body"]), (
                    f'{c["name"]}: unary body drifted')
            replayed += 1
        assert replayed == len(cases)
        await client.close()
        await cleanup()
        up.stop()

    asyncio.run(run())


@pytest.mark.timeout(300)
def test_cassette_replay_through_fast_front():
    """The native C++ front must preserve the same passthrough fidelity
    on its hot paths (chat/completions/embeddings)."""
    from aigw.extproc.fast_front import FastFront

    cases = [c for c in _json_endpoint_cases()
             if c["path"] in ("/v1/chat/completions", "/v1/completions",
                              "/v1/embeddings")]

    async def run():
        up = ReplayUpstream()
        up_port = await up.start()
        cfg = load_config(yaml.safe_load(f"""
routes:
  - name: conformance
    backends:
      - name: openai
        schema: OpenAI
        upstream: {{host: 127.0.0.1, port: {up_port}}}
"""))
        server = GatewayServer(RuntimeConfig(cfg))
        front = FastFront(server, server.runtime)
        gw_port = await front.start("127.0.0.1", 0)
        client = LeanClient()
        for c in cases:
            up.response = (c["status"], c["resp_ctype"], c["resp_body"])
            up.requests.clear()
            r = await client.post(
                host="127.0.0.1", port=gw_port, tls=False, path=c["path"],
                headers={"content-type": "application/json"},
                body=c["req_body"], timeout_s=30.0,
            )
            got = await r.read()
            r.release()
            if c["status"] >= 400:
                assert 400 <= r.status < 500, (c["name"], r.status, got[:200])
                continue
            assert r.status == c["status"], (c["name"], r.status, got[:300])
            if up.requests:
                assert json.loads(up.requests[0]) == json.loads(c["req_body"]), c["name"]
            if c["resp_ctype"].startswith("text/event-stream"):
                assert _sse_events(got) == _sse_events(c["resp_body"]), c["name"]
            else:
                assert json.loads(got) == json.loads(c["resp_body"]), c["name"]
        await client.close()
        await front.stop()
        up.stop()

    asyncio.run(run())
