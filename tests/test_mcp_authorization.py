"""MCP authorization: CEL/scope/claim rules + OAuth scope challenges
(parity: internal/mcpproxy/authorization.go rule matching :186-242,
insufficient_scope WWW-Authenticate :466-478, 401-vs-403 split in
handlers.go:755-783). Covers the full 401/403 matrix end-to-end through
the proxy plus unit coverage of the CEL condition subset."""

import asyncio
import base64
import json

import aiohttp
import pytest
from aiohttp import web

from aigw.filterapi.config import load_config
from aigw.mcp.authorization import (
    CELCondition,
    CompiledAuthorization,
    MCPAuthzError,
    build_activation,
    build_insufficient_scope_header,
    parse_unverified_claims,
)
from aigw.mcp.proxy import MCPProxy
from tests.test_mcp import FakeMCPServer, _start


def _jwt(claims: dict) -> str:
    def b64(d):
        return base64.urlsafe_b64encode(json.dumps(d).encode()).rstrip(b"=").decode()

    return f"{b64({'alg': 'none'})}.{b64(claims)}.sig"


# ---------------------------------------------------------------------------
# CEL condition engine


def _activation(**over):
    base = build_activation(
        http_method="POST", host="gw.local", path="/mcp",
        headers={"X-Team": "infra", "authorization": "Bearer x"},
        mcp_method="tools/call", backend="alpha", tool="tool_a",
        params={"name": "alpha__tool_a", "arguments": {"level": 3}},
        claims={"sub": "alice", "groups": ["admins", "dev"],
                "scope": "mcp:read mcp:write"},
    )
    base.update(over)
    return base


def test_cel_condition_subset():
    act = _activation()
    cases_true = [
        'request.mcp.tool == "tool_a"',
        'request.mcp.backend == "alpha" && request.method == "POST"',
        'request.headers["x-team"] == "infra"',
        '"admins" in request.auth.jwt.claims.groups',
        '"mcp:write" in request.auth.jwt.scopes',
        'request.mcp.tool.startsWith("tool")',
        'request.host.endsWith(".local")',
        'request.path.contains("mcp")',
        'request.mcp.method.matches("tools/.*")',
        '!(request.mcp.tool == "other")',
        'request.mcp.params.arguments.level == 3',
        'request.auth.jwt.claims.sub == "alice" || false',
    ]
    for expr in cases_true:
        assert CELCondition(expr).evaluate(act), expr
    cases_false = [
        'request.mcp.tool == "tool_b"',
        '"root" in request.auth.jwt.claims.groups',
        'request.headers["x-missing"] == "v"',      # absent -> false
        'request.auth.jwt.claims.missing.deep == 1',  # nested absent
        'request.mcp.tool.startsWith("zzz")',
    ]
    for expr in cases_false:
        assert not CELCondition(expr).evaluate(act), expr

    # unsafe syntax rejected at compile time
    for bad in ("__import__('os')", "request.__class__", "open('/etc/passwd')",
                "[x for x in request]", "request.headers.pop('a')"):
        with pytest.raises(MCPAuthzError):
            CELCondition(bad)


def test_scope_and_claim_helpers():
    claims = parse_unverified_claims(
        "Bearer " + _jwt({"sub": "a", "scope": "s1 s2"}))
    assert claims["sub"] == "a"
    assert parse_unverified_claims("Bearer not-a-jwt") == {}
    assert parse_unverified_claims("") == {}
    h = build_insufficient_scope_header(["s1", "s2"], "https://gw/meta")
    assert h == ('Bearer error="insufficient_scope", scope="s1 s2", '
                 'resource_metadata="https://gw/meta", '
                 'error_description="The token is missing required scopes"')


# ---------------------------------------------------------------------------
# rule matching (CompiledAuthorization.authorize)


def _compile(doc: dict) -> CompiledAuthorization:
    cfg = load_config({
        "routes": [],
        "mcp": {"routes": [{
            "name": "m", "path": "/mcp",
            "backends": [{"name": "alpha", "upstream": {"host": "h", "port": 1}}],
            "authorization": doc,
        }]},
    })
    return CompiledAuthorization(cfg.mcp.routes[0].authorization)


def _decide(authz, *, claims=None, tool="tool_a", backend="alpha",
            headers=None, method="tools/call"):
    claims = claims or {}
    act = build_activation(
        http_method="POST", host="h", path="/mcp",
        headers=headers or {}, mcp_method=method, backend=backend,
        tool=tool, params={}, claims=claims)
    return authz.authorize(activation=act, claims=claims,
                           backend=backend, tool=tool)


def test_rule_matrix():
    # empty rules -> defaultAction verbatim
    assert _decide(_compile({"defaultAction": "Allow"})).allowed
    assert not _decide(_compile({"defaultAction": "Deny"})).allowed

    # scope rule: all scopes required; partial -> challenge with the set
    authz = _compile({"defaultAction": "Deny", "resourceMetadataUrl": "https://gw/meta",
                      "rules": [{"action": "Allow",
                                 "jwtScopes": ["mcp:read", "mcp:write"]}]})
    assert _decide(authz, claims={"scope": "mcp:read mcp:write extra"}).allowed
    d = _decide(authz, claims={"scope": "mcp:read"})
    assert not d.allowed and d.required_scopes == ["mcp:read", "mcp:write"]

    # claim rule: string-or-array handling, dotted paths
    authz = _compile({"defaultAction": "Deny", "rules": [
        {"action": "Allow", "jwtClaims": [{"name": "org.team",
                                           "values": ["infra"]}]}]})
    assert _decide(authz, claims={"org": {"team": "infra"}}).allowed
    assert _decide(authz, claims={"org": {"team": ["infra", "x"]}}).allowed
    assert not _decide(authz, claims={"org": {"team": "sales"}}).allowed

    # first match wins: deny before allow
    authz = _compile({"defaultAction": "Allow", "rules": [
        {"action": "Deny", "tools": ["alpha__tool_secret"]},
        {"action": "Allow"}]})
    assert not _decide(authz, tool="tool_secret").allowed
    assert _decide(authz, tool="tool_a").allowed

    # CEL condition gates the rule
    authz = _compile({"defaultAction": "Deny", "rules": [
        {"action": "Allow", "cel": 'request.headers["x-team"] == "infra"'}]})
    assert _decide(authz, headers={"X-Team": "infra"}).allowed
    assert not _decide(authz, headers={"X-Team": "sales"}).allowed


# ---------------------------------------------------------------------------
# end-to-end 401/403 matrix through the proxy


def test_mcp_authz_end_to_end():
    async def run():
        srv = FakeMCPServer("alpha", ["tool_a", "tool_secret"])
        runner, port = await _start(srv.handle)
        cfg = load_config({
            "routes": [],
            "mcp": {"routes": [{
                "name": "m", "path": "/mcp",
                "bearerToken": "gate-token",
                "resourceMetadataUrl": "https://gw/.well-known/oauth-protected-resource",
                "backends": [{"name": "alpha",
                              "upstream": {"host": "127.0.0.1", "port": port}}],
                "authorization": {
                    "defaultAction": "Deny",
                    "resourceMetadataUrl":
                        "https://gw/.well-known/oauth-protected-resource",
                    "rules": [
                        {"action": "Deny", "tools": ["alpha__tool_secret"]},
                        {"action": "Allow", "jwtScopes": ["mcp:call"],
                         "cel": 'request.mcp.method.startsWith("tools/")'},
                        {"action": "Allow",
                         "cel": 'request.mcp.method == "initialize"'},
                    ],
                },
            }]},
        })
        route = cfg.mcp.routes[0]
        proxy = MCPProxy(route, "seed")
        app = web.Application()
        app.router.add_route("*", "/mcp", proxy.handle)
        prunner = web.AppRunner(app)
        await prunner.setup()
        site = web.TCPSite(prunner, "127.0.0.1", 0)
        await site.start()
        pport = prunner.addresses[0][1]
        base = f"http://127.0.0.1:{pport}/mcp"

        # authn gate: wrong bearer -> 401 + resource metadata challenge
        async with aiohttp.ClientSession() as c:
            async with c.post(base, json={"jsonrpc": "2.0", "id": 1,
                                          "method": "initialize", "params": {}},
                              headers={"authorization": "Bearer wrong"}) as r:
                assert r.status == 401
                assert "resource_metadata=" in r.headers["www-authenticate"]

            # authn passes; initialize allowed by CEL rule. The gate token
            # is not a JWT, so rule evaluation sees empty claims.
            hdr_gate = {"authorization": "Bearer gate-token"}
            async with c.post(base, json={"jsonrpc": "2.0", "id": 1,
                                          "method": "initialize",
                                          "params": {"protocolVersion": "2025-06-18",
                                                     "capabilities": {}}},
                              headers=hdr_gate) as r:
                assert r.status == 200
                session = r.headers["mcp-session-id"]

            call = {"jsonrpc": "2.0", "id": 2, "method": "tools/call",
                    "params": {"name": "alpha__tool_a", "arguments": {}}}
            hdr = dict(hdr_gate, **{"mcp-session-id": session})

            # tools/call without scopes -> 403 + insufficient_scope challenge
            async with c.post(base, json=call, headers=hdr) as r:
                assert r.status == 403
                www = r.headers["www-authenticate"]
                assert 'error="insufficient_scope"' in www
                assert 'scope="mcp:call"' in www
                assert "oauth-protected-resource" in www

            # ping (no matching rule) -> 403 default deny, NO scope challenge
            async with c.post(base, json={"jsonrpc": "2.0", "id": 3,
                                          "method": "ping"}, headers=hdr) as r:
                assert r.status == 403
                assert "www-authenticate" not in r.headers

            # bearer gate must still pass; scopes ride a second JWT? No —
            # single Authorization header. Switch the route check to the
            # JWT itself is the realistic deployment; here the gate token
            # doubles as the scope carrier via a JWT-shaped token.
        await prunner.cleanup()
        await proxy.close()
        await runner.cleanup()

    asyncio.run(run())


def test_mcp_authz_scoped_jwt_allows():
    """With a JWT-shaped bearer that passes the authn gate (no static
    token configured) the scope rule admits the call and the deny rule
    still blocks the secret tool."""

    async def run():
        srv = FakeMCPServer("alpha", ["tool_a", "tool_secret"])
        runner, port = await _start(srv.handle)
        cfg = load_config({
            "routes": [],
            "mcp": {"routes": [{
                "name": "m", "path": "/mcp",
                "backends": [{"name": "alpha",
                              "upstream": {"host": "127.0.0.1", "port": port}}],
                "authorization": {
                    "defaultAction": "Deny",
                    "rules": [
                        {"action": "Deny", "tools": ["alpha__tool_secret"]},
                        {"action": "Allow", "jwtScopes": ["mcp:call"]},
                    ],
                },
            }]},
        })
        proxy = MCPProxy(cfg.mcp.routes[0], "seed")
        app = web.Application()
        app.router.add_route("*", "/mcp", proxy.handle)
        prunner = web.AppRunner(app)
        await prunner.setup()
        site = web.TCPSite(prunner, "127.0.0.1", 0)
        await site.start()
        pport = prunner.addresses[0][1]
        base = f"http://127.0.0.1:{pport}/mcp"
        tok = _jwt({"sub": "alice", "scope": "mcp:call"})
        hdr = {"authorization": f"Bearer {tok}"}

        async with aiohttp.ClientSession() as c:
            # the scope rule has no tool/CEL gate, so the scoped JWT
            # admits initialize
            async with c.post(base, json={"jsonrpc": "2.0", "id": 1,
                                          "method": "initialize",
                                          "params": {"protocolVersion": "2025-06-18",
                                                     "capabilities": {}}},
                              headers=hdr) as r:
                assert r.status == 200
                session = r.headers["mcp-session-id"]
            hdr_s = dict(hdr, **{"mcp-session-id": session})
            # scoped call to a permitted tool -> allowed end to end
            async with c.post(base, json={"jsonrpc": "2.0", "id": 2,
                                          "method": "tools/call",
                                          "params": {"name": "alpha__tool_a",
                                                     "arguments": {}}},
                              headers=hdr_s) as r:
                assert r.status == 200
            # the deny rule wins first for the secret tool, scopes or not
            async with c.post(base, json={"jsonrpc": "2.0", "id": 3,
                                          "method": "tools/call",
                                          "params": {"name": "alpha__tool_secret",
                                                     "arguments": {}}},
                              headers=hdr_s) as r:
                assert r.status == 403
            # a token WITHOUT the scope gets the challenge
            weak = {"authorization":
                    f"Bearer {_jwt({'sub': 'bob', 'scope': 'other'})}",
                    "mcp-session-id": session}
            async with c.post(base, json={"jsonrpc": "2.0", "id": 4,
                                          "method": "tools/call",
                                          "params": {"name": "alpha__tool_a",
                                                     "arguments": {}}},
                              headers=weak) as r:
                assert r.status == 403
                assert 'scope="mcp:call"' in r.headers["www-authenticate"]
        await prunner.cleanup()
        await proxy.close()
        await runner.cleanup()

    asyncio.run(run())


def test_wellknown_oauth_metadata_endpoints():
    """RFC 9728 protected-resource + RFC 8414 authorization-server
    metadata served outside the authn gate, with CORS for the browser
    inspector (mcp_route_security_policy.go :340-560)."""
    from aigw.extproc.server import GatewayServer
    from aigw.filterapi import RuntimeConfig
    from tests.test_mcp_jwt import _jwks

    async def run():
        cfg = load_config({
            "routes": [],
            "mcp": {"routes": [{
                "name": "m", "path": "/mcp",
                "bearerToken": "gate",
                "backends": [{"name": "a", "upstream": {"host": "h", "port": 1}}],
                "oauth": {"issuer": "https://as.example.com",
                          "jwks": _jwks(),
                          "resource": "https://gw.example.com/mcp",
                          "resourceName": "aigw mcp",
                          "scopesSupported": ["mcp:read", "mcp:call"]},
            }]},
        })
        server = GatewayServer(RuntimeConfig(cfg))
        app = server.make_app()
        runner = web.AppRunner(app)
        await runner.setup()
        site = web.TCPSite(runner, "127.0.0.1", 0)
        await site.start()
        port = runner.addresses[0][1]
        async with aiohttp.ClientSession() as c:
            # no Authorization header needed
            async with c.get(
                f"http://127.0.0.1:{port}/.well-known/oauth-protected-resource/mcp"
            ) as r:
                assert r.status == 200
                doc = await r.json()
                assert doc["resource"] == "https://gw.example.com/mcp"
                assert doc["authorization_servers"] == ["https://as.example.com"]
                assert doc["bearer_methods_supported"] == ["header"]
                assert doc["scopes_supported"] == ["mcp:read", "mcp:call"]
                assert r.headers["Access-Control-Allow-Origin"] == "*"
            async with c.get(
                f"http://127.0.0.1:{port}/.well-known/oauth-authorization-server/mcp"
            ) as r:
                assert r.status == 200
                doc = await r.json()
                assert doc["issuer"] == "https://as.example.com"
                assert doc["token_endpoint"] == "https://as.example.com/token"
                assert "S256" in doc["code_challenge_methods_supported"]
        await runner.cleanup()

    asyncio.run(run())
