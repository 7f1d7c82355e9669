"""MCP (Model Context Protocol) gateway: N upstream MCP servers behind one
streamable-HTTP endpoint.

Parity targets (internal/mcpproxy/):
- ``initialize`` fans out to every backend, merges capabilities, and seals
  the per-backend session ids into one stateless encrypted session blob
  (handlers.go:569, session.go);
- ``tools/list`` aggregates with ``<backend>__<tool>`` name prefixing and
  per-backend ToolSelector filtering (include/exclude exact + RE2-style
  regex, exclude wins — mcpconfig.go:81-98);
- ``tools/call`` strips the prefix and routes to the owning backend with
  that backend's own session id (handlers.go:728);
- ``resources/list``, ``prompts/list`` aggregate; ``resources/read`` routes
  by trying the owning backends; ``ping`` answers locally;
- upstream responses in either JSON or SSE framing are folded to the
  JSON-RPC response with the matching id.
"""

from __future__ import annotations

import asyncio
import json
import logging
import re
from typing import Optional

import aiohttp
from aiohttp import web

from aigw import internalapi
from aigw.filterapi.config import MCPBackend, MCPRoute
from aigw.mcp.session import SessionCrypto, SessionError
from aigw.mutator import apply_header_mutation
from aigw.translator.sse import SSEDecoder

logger = logging.getLogger("aigw.mcp")

PREFIX_SEP = "__"
PROTOCOL_VERSION = "2025-06-18"
_METHOD_KEY = web.RequestKey("aigw_mcp_method", str) if hasattr(web, "RequestKey") else "aigw_mcp_method"
_BACKEND_KEY = web.RequestKey("aigw_mcp_backend", str) if hasattr(web, "RequestKey") else "aigw_mcp_backend"
_TOOL_KEY = web.RequestKey("aigw_mcp_tool", str) if hasattr(web, "RequestKey") else "aigw_mcp_tool"


def _rpc_error(id_, code: int, message: str, status: int = 200) -> web.Response:
    return web.json_response(
        {"jsonrpc": "2.0", "id": id_, "error": {"code": code, "message": message}},
        status=status,
    )


class MCPProxy:
    def __init__(self, route: MCPRoute, session_seed: str, client_factory=None,
                 metrics=None, tracer=None, fallback_seed: str = ""):
        self.route = route
        self.crypto = SessionCrypto(
            session_seed, fallback_seeds=[fallback_seed] if fallback_seed else None
        )
        self.metrics = metrics  # aigw.metrics.GenAIMetrics or None
        self.tracer = tracer  # aigw.tracing.Tracer or None
        self._client_factory = client_factory
        self._session: Optional[aiohttp.ClientSession] = None
        self._jwt = None
        if getattr(route, "oauth", None) is not None:
            from aigw.mcp.jwt_auth import JWTValidator

            # raises at startup on unusable JWKS (config error, not a
            # silent open gate)
            self._jwt = JWTValidator.from_config(
                route.oauth.issuer, route.oauth.audiences,
                jwks_json=route.oauth.jwks, jwks_file=route.oauth.jwks_file,
            )
        self._tool_res: dict[str, list] = {}
        for b in route.backends:
            self._tool_res[b.name] = [re.compile(p) for p in b.tool_exclude + b.tool_include]
        self._authz = None
        if getattr(route, "authorization", None) is not None:
            from aigw.mcp.authorization import CompiledAuthorization

            # compiles CEL rules at startup: bad config fails loudly
            self._authz = CompiledAuthorization(route.authorization)

    async def _client(self) -> aiohttp.ClientSession:
        if self._session is None:
            if self._client_factory is not None:
                self._session = self._client_factory()
            else:
                self._session = aiohttp.ClientSession(
                    connector=aiohttp.TCPConnector(limit=0)
                )
        return self._session

    async def close(self) -> None:
        if self._session is not None:
            await self._session.close()
            self._session = None

    # ---- upstream I/O --------------------------------------------------------

    async def _call_backend(
        self, backend: MCPBackend, payload: dict, session_id: str = ""
    ) -> tuple[Optional[dict], str]:
        """POST a JSON-RPC message; returns (response message or None for
        notifications, new session id if the backend issued one)."""
        client = await self._client()
        headers = {
            "content-type": "application/json",
            "accept": "application/json, text/event-stream",
        }
        if session_id:
            headers[internalapi.MCP_SESSION_ID_HEADER] = session_id
        if backend.headers is not None:
            headers = apply_header_mutation(headers, backend.headers)
        if backend.auth is not None and backend.auth.api_key:
            headers["authorization"] = f"Bearer {backend.auth.api_key}"
        url = backend.upstream.base_url + backend.path
        async with client.post(url, json=payload, headers=headers) as resp:
            new_session = resp.headers.get(internalapi.MCP_SESSION_ID_HEADER, "")
            if resp.status == 202:
                return None, new_session
            if resp.status >= 400:
                body = await resp.text()
                raise RuntimeError(f"backend {backend.name} returned {resp.status}: {body[:200]}")
            ctype = resp.headers.get("content-type", "")
            if ctype.startswith("text/event-stream"):
                dec = SSEDecoder()
                want_id = payload.get("id")
                async for chunk in resp.content.iter_any():
                    for ev in dec.feed(chunk):
                        if not ev.data:
                            continue
                        try:
                            msg = json.loads(ev.data)
                        except ValueError:
                            continue
                        if msg.get("id") == want_id and ("result" in msg or "error" in msg):
                            return msg, new_session
                return None, new_session
            data = await resp.read()
            return (json.loads(data) if data else None), new_session

    # ---- tool filtering ------------------------------------------------------

    @staticmethod
    def _selected(backend: MCPBackend, tool: str) -> bool:
        """include/exclude exact + regex; exclude wins (mcpconfig.go:81-98)."""

        def matches(pats: list[str]) -> bool:
            for p in pats:
                if p == tool:
                    return True
                try:
                    if re.fullmatch(p, tool):
                        return True
                except re.error:
                    continue
            return False

        if backend.tool_exclude and matches(backend.tool_exclude):
            return False
        if backend.tool_include:
            return matches(backend.tool_include)
        return True

    # ---- handlers ------------------------------------------------------------

    def _authorize(self, request: web.Request):
        """Bearer / JWT gate; 401 carries the OAuth protected-resource
        metadata pointer (authorization.go WWW-Authenticate behavior)."""
        if self._jwt is not None:
            got = request.headers.get("authorization", "")
            if got.startswith("Bearer "):
                from aigw.mcp.jwt_auth import JWTError

                try:
                    self._jwt.validate(got[7:])
                    return None
                except JWTError:
                    pass
            return self._deny(request)
        if not self.route.bearer_token:
            return None
        got = request.headers.get("authorization", "")
        if got == f"Bearer {self.route.bearer_token}":
            return None
        return self._deny(request)

    def _deny(self, request: web.Request):
        resp = web.json_response(
            {"jsonrpc": "2.0", "id": None,
             "error": {"code": -32001, "message": "unauthorized"}},
            status=401,
        )
        if self.route.resource_metadata_url:
            resp.headers["www-authenticate"] = (
                f'Bearer resource_metadata="{self.route.resource_metadata_url}"'
            )
        return resp

    def _authorize_rules(self, request: web.Request, method: str, id_,
                         payload: dict):
        """Per-request CEL/scope/claim authorization after the
        authentication gate (authorization.go:755-783): deny answers 403
        with an insufficient_scope WWW-Authenticate challenge when the
        denial was scope-only."""
        from aigw.mcp.authorization import (
            build_activation,
            build_insufficient_scope_header,
            parse_unverified_claims,
        )

        params = payload.get("params") or {}
        backend_name = ""
        tool = ""
        if method in ("tools/call", "prompts/get"):
            prefixed = str(params.get("name", ""))
            be, stripped = self._backend_for(prefixed)
            if be is not None:
                backend_name, tool = be.name, stripped
            else:
                tool = prefixed
        claims = parse_unverified_claims(request.headers.get("authorization", ""))
        activation = build_activation(
            http_method=request.method,
            host=request.headers.get("host", ""),
            path=request.path,
            headers=dict(request.headers),
            mcp_method=method,
            backend=backend_name,
            tool=tool,
            params=params,
            claims=claims,
        )
        decision = self._authz.authorize(
            activation=activation, claims=claims, backend=backend_name, tool=tool
        )
        if decision.allowed:
            return None
        resp = _rpc_error(id_, -32001, "access denied", status=403)
        if decision.required_scopes:
            resp.headers["www-authenticate"] = build_insufficient_scope_header(
                decision.required_scopes, self._authz.resource_metadata_url
            )
        return resp

    @staticmethod
    def _cors(resp: web.Response) -> web.Response:
        # the MCP inspector runs in a browser on a different origin
        # (mcp_route_security_policy.go ensureCORSHeaders)
        resp.headers["Access-Control-Allow-Origin"] = "*"
        resp.headers["Access-Control-Allow-Methods"] = "GET"
        resp.headers["Access-Control-Allow-Headers"] = "mcp-protocol-version"
        return resp

    async def handle_protected_resource_metadata(self, request: web.Request) -> web.Response:
        """RFC 9728 OAuth protected-resource metadata, served WITHOUT the
        authn gate (buildOAuthProtectedResourceMetadataJSON :474-502)."""
        oauth = self.route.oauth
        if oauth is None:
            return web.Response(status=404)
        doc = {
            "resource": oauth.resource or self.route.path,
            "authorization_servers": [oauth.issuer] if oauth.issuer else [],
            "bearer_methods_supported": ["header"],
        }
        if oauth.resource_name:
            doc["resource_name"] = oauth.resource_name
        if oauth.scopes_supported:
            doc["scopes_supported"] = oauth.scopes_supported
        return self._cors(web.json_response(doc))

    async def handle_authorization_server_metadata(self, request: web.Request) -> web.Response:
        """RFC 8414 authorization-server metadata. The reference proxies
        the AS's own document with a hardcoded fallback
        (buildOAuthAuthServerMetadataJSON :504-560); without egress this
        gateway serves the fallback shape derived from the issuer."""
        oauth = self.route.oauth
        if oauth is None or not oauth.issuer:
            return web.Response(status=404)
        issuer = oauth.issuer.rstrip("/")
        doc = {
            "issuer": issuer,
            "authorization_endpoint": f"{issuer}/authorize",
            "token_endpoint": f"{issuer}/token",
            "registration_endpoint": f"{issuer}/register",
            "response_types_supported": ["code"],
            "grant_types_supported": ["authorization_code", "refresh_token"],
            "code_challenge_methods_supported": ["S256"],
            "token_endpoint_auth_methods_supported": ["client_secret_post",
                                                      "client_secret_basic", "none"],
        }
        if oauth.scopes_supported:
            doc["scopes_supported"] = oauth.scopes_supported
        return self._cors(web.json_response(doc))

    async def handle(self, request: web.Request) -> web.StreamResponse:
        import time as _time

        t0 = _time.monotonic()
        span = None
        if self.tracer is not None:
            span = self.tracer.start_span("mcp", dict(request.headers))
        resp = None
        try:
            resp = await self._handle_inner(request)
            return resp
        finally:
            method = request.get(_METHOD_KEY, "") or request.method
            if self.metrics is not None:
                outcome = "ok"
                if isinstance(resp, web.Response) and resp.status >= 400:
                    outcome = f"http_{resp.status}"
                elif resp is None:
                    outcome = "error"
                self.metrics.mcp_requests.labels(method=method, outcome=outcome).inc()
                self.metrics.mcp_duration.labels(method=method).observe(
                    _time.monotonic() - t0
                )
            if span is not None:
                # per-RPC attributes (tracing/mcp*.go parity): method,
                # route, resolved backend/tool, session presence, outcome
                span.set("mcp.method", method)
                span.set("mcp.route", self.route.name)
                span.set("mcp.backend", request.get(_BACKEND_KEY, "") or None)
                span.set("mcp.tool", request.get(_TOOL_KEY, "") or None)
                span.set("mcp.session",
                         bool(request.headers.get(internalapi.MCP_SESSION_ID_HEADER)))
                if isinstance(resp, web.Response) and resp.status >= 400:
                    self.tracer.end_span(span, error=f"http_{resp.status}")
                else:
                    self.tracer.end_span(span)

    async def _handle_inner(self, request: web.Request) -> web.StreamResponse:
        denied = self._authorize(request)
        if denied is not None:
            return denied
        if request.method == "GET":
            # server-initiated stream endpoint: no push support yet
            return web.Response(status=405, text="SSE server stream not supported")
        if request.method == "DELETE":
            return web.Response(status=202)  # stateless sessions: nothing to delete
        try:
            payload = json.loads(await request.read())
        except ValueError:
            return _rpc_error(None, -32700, "parse error", status=400)
        if isinstance(payload, list):
            return _rpc_error(None, -32600, "batch requests not supported", status=400)
        method = payload.get("method", "")
        request[_METHOD_KEY] = method
        if method in ("tools/call", "prompts/get"):
            prefixed = str((payload.get("params") or {}).get("name", ""))
            be, bare = self._backend_for(prefixed)
            if be is not None:
                request[_BACKEND_KEY] = be.name
                request[_TOOL_KEY] = bare
        id_ = payload.get("id")
        if self._authz is not None:
            denied = self._authorize_rules(request, method, id_, payload)
            if denied is not None:
                return denied
        token = request.headers.get(internalapi.MCP_SESSION_ID_HEADER, "")
        sessions: dict[str, str] = {}
        if token:
            try:
                sessions = self.crypto.open(token).get("s", {})
            except SessionError:
                return _rpc_error(id_, -32600, "invalid session", status=404)
        try:
            if method == "initialize":
                return await self._initialize(payload)
            if method == "ping":
                return web.json_response({"jsonrpc": "2.0", "id": id_, "result": {}})
            if method.startswith("notifications/"):
                await self._fan_out_notification(payload, sessions)
                return web.Response(status=202)
            if method == "tools/list":
                return await self._aggregate_list(payload, sessions, "tools", "name")
            if method == "prompts/list":
                return await self._aggregate_list(payload, sessions, "prompts", "name")
            if method == "resources/list":
                return await self._aggregate_list(payload, sessions, "resources", "name")
            if method in ("tools/call", "prompts/get"):
                return await self._routed_call(payload, sessions, key="name")
            if method == "resources/read":
                return await self._resources_read(payload, sessions)
            return _rpc_error(id_, -32601, f"method {method!r} not found")
        except Exception as e:
            logger.exception("mcp %s failed", method)
            return _rpc_error(id_, -32603, str(e))

    async def _initialize(self, payload: dict) -> web.Response:
        id_ = payload.get("id")
        results = await asyncio.gather(
            *(self._call_backend(b, payload) for b in self.route.backends),
            return_exceptions=True,
        )
        sessions: dict[str, str] = {}
        caps: dict = {}
        ok = 0
        for b, res in zip(self.route.backends, results):
            if isinstance(res, Exception):
                logger.warning("initialize failed for backend %s: %s", b.name, res)
                continue
            msg, new_session = res
            if new_session:
                sessions[b.name] = new_session
            if msg and "result" in msg:
                ok += 1
                for k, v in (msg["result"].get("capabilities") or {}).items():
                    if isinstance(v, dict):
                        caps.setdefault(k, {}).update(v)
                    else:
                        caps[k] = v
        if ok == 0:
            return _rpc_error(id_, -32603, "all MCP backends failed to initialize")
        token = self.crypto.seal({"s": sessions})
        resp = web.json_response(
            {
                "jsonrpc": "2.0",
                "id": id_,
                "result": {
                    "protocolVersion": PROTOCOL_VERSION,
                    "capabilities": caps,
                    "serverInfo": {"name": "aigw-mcp-gateway", "version": "0.1.0"},
                },
            }
        )
        resp.headers[internalapi.MCP_SESSION_ID_HEADER] = token
        return resp

    async def _fan_out_notification(self, payload: dict, sessions: dict[str, str]) -> None:
        await asyncio.gather(
            *(
                self._call_backend(b, payload, sessions.get(b.name, ""))
                for b in self.route.backends
            ),
            return_exceptions=True,
        )

    async def _aggregate_list(
        self, payload: dict, sessions: dict[str, str], list_key: str, name_key: str
    ) -> web.Response:
        id_ = payload.get("id")
        results = await asyncio.gather(
            *(
                self._call_backend(b, payload, sessions.get(b.name, ""))
                for b in self.route.backends
            ),
            return_exceptions=True,
        )
        merged: list = []
        for b, res in zip(self.route.backends, results):
            if isinstance(res, Exception):
                logger.warning("%s failed for backend %s: %s", list_key, b.name, res)
                continue
            msg, _ = res
            if not msg or "result" not in msg:
                continue
            for item in msg["result"].get(list_key) or []:
                name = item.get(name_key, "")
                if list_key == "tools" and not self._selected(b, name):
                    continue
                item = dict(item)
                item[name_key] = f"{b.name}{PREFIX_SEP}{name}"
                merged.append(item)
        return web.json_response(
            {"jsonrpc": "2.0", "id": id_, "result": {list_key: merged}}
        )

    def _backend_for(self, prefixed: str) -> tuple[Optional[MCPBackend], str]:
        name, sep, rest = prefixed.partition(PREFIX_SEP)
        if sep:
            for b in self.route.backends:
                if b.name == name:
                    return b, rest
        return None, prefixed

    async def _routed_call(self, payload: dict, sessions: dict[str, str], key: str) -> web.Response:
        id_ = payload.get("id")
        params = payload.get("params") or {}
        backend, bare = self._backend_for(params.get(key, ""))
        if backend is None:
            return _rpc_error(id_, -32602, f"unknown backend prefix in {params.get(key)!r}")
        if payload.get("method") == "tools/call" and not self._selected(backend, bare):
            return _rpc_error(id_, -32602, f"tool {bare!r} is not allowed")
        fwd = dict(payload)
        fwd["params"] = dict(params)
        fwd["params"][key] = bare
        msg, _ = await self._call_backend(backend, fwd, sessions.get(backend.name, ""))
        if msg is None:
            return _rpc_error(id_, -32603, "no response from backend")
        msg["id"] = id_
        return web.json_response(msg)

    async def _resources_read(self, payload: dict, sessions: dict[str, str]) -> web.Response:
        id_ = payload.get("id")
        last_err = "no backends"
        for b in self.route.backends:
            try:
                msg, _ = await self._call_backend(b, payload, sessions.get(b.name, ""))
                if msg and "result" in msg:
                    msg["id"] = id_
                    return web.json_response(msg)
                if msg and "error" in msg:
                    last_err = msg["error"].get("message", "error")
            except Exception as e:
                last_err = str(e)
        return _rpc_error(id_, -32002, f"resource not found: {last_err}")
