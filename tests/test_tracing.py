"""Tracing tests (parity: internal/tracing semconv selection §A.3, span
lifecycle, upstream traceparent injection)."""

import asyncio

import aiohttp
import pytest

from aigw.extproc.server import GatewayServer, run_server
from aigw.filterapi import RuntimeConfig, load_config
from aigw.testing.mockupstream import start_mock_upstream
from aigw.tracing import InMemoryExporter, Tracer, tracing_from_env


def test_semconv_selection():
    assert Tracer(semconv="openinference").semconv == "openinference"
    assert Tracer(semconv="gen_ai").semconv == "gen_ai"
    with pytest.raises(ValueError):
        Tracer(semconv="wat")  # unknown value is a startup error, not fallback


def test_tracing_from_env():
    assert tracing_from_env({"OTEL_SDK_DISABLED": "true"}) is None
    assert tracing_from_env({}) is None
    t = tracing_from_env(
        {
            "OTEL_TRACES_EXPORTER": "console",
            "AI_GATEWAY_TRACING_SEMCONV": "gen_ai",
            "AIGW_SPAN_REQUEST_HEADER_ATTRIBUTES": "x-user-id:user.id",
        }
    )
    assert t is not None and t.semconv == "gen_ai"
    assert t.header_attributes == {"x-user-id": "user.id"}
    with pytest.raises(ValueError):
        tracing_from_env({"OTEL_TRACES_EXPORTER": "console",
                          "AI_GATEWAY_TRACING_SEMCONV": "bogus"})


def test_span_parent_from_traceparent():
    t = Tracer(exporter=InMemoryExporter())
    span = t.start_span("chat", {"traceparent": "00-" + "ab" * 16 + "-" + "cd" * 8 + "-01"})
    assert span.trace_id == "ab" * 16
    assert span.parent_span_id == "cd" * 8
    t.end_span(span)
    assert t.exporter.spans[0].status == "OK"


def test_gateway_emits_spans_and_injects_traceparent():
    async def main():
        mock, up_runner, up_port = await start_mock_upstream()
        mock.record = True
        cfg = load_config(
            {
                "version": "v1",
                "routes": [
                    {
                        "name": "r",
                        "backends": [
                            {"name": "b", "schema": "OpenAI",
                             "upstream": {"host": "127.0.0.1", "port": up_port}}
                        ],
                    }
                ],
            }
        )
        exporter = InMemoryExporter()
        tracer = Tracer(semconv="gen_ai", exporter=exporter,
                        header_attributes={"x-user-id": "user.id"})
        server = GatewayServer(RuntimeConfig(cfg), tracer=tracer)
        gw_runner = await run_server(server, host="127.0.0.1", port=0)
        port = gw_runner.addresses[0][1]
        async with aiohttp.ClientSession() as client:
            async with client.post(
                f"http://127.0.0.1:{port}/v1/chat/completions",
                json={"model": "m", "messages": [{"role": "user", "content": "hi"}]},
                headers={"x-user-id": "alice"},
            ) as r:
                assert r.status == 200
        span = exporter.spans[0]
        assert span.attributes["gen_ai.request.model"] == "m"
        assert span.attributes["gen_ai.usage.output_tokens"] == 16
        assert span.attributes["user.id"] == "alice"
        assert span.end_ns > span.start_ns
        # traceparent joined the upstream request
        up_headers = {k.lower(): v for k, v in mock.requests[-1]["headers"].items()}
        assert up_headers["traceparent"].startswith(f"00-{span.trace_id}-")
        await gw_runner.cleanup()
        await up_runner.cleanup()

    asyncio.run(main())


def test_session_id_header_attribute_default():
    """Unset span header-attrs default to agent-session-id -> session.id;
    an explicitly EMPTY value clears the default
    (requestheaderattrs/resolve.go semantics)."""
    from aigw.tracing.tracing import tracing_from_env as build_tracer

    env = {"OTEL_EXPORTER_OTLP_ENDPOINT": "http://127.0.0.1:9",
           "OTEL_TRACES_EXPORTER": "otlp"}
    t = build_tracer(env)
    assert t.header_attributes == {"agent-session-id": "session.id"}

    t = build_tracer({**env, "AIGW_SPAN_REQUEST_HEADER_ATTRIBUTES": ""})
    assert t.header_attributes == {}

    t = build_tracer({**env,
                      "AIGW_SPAN_REQUEST_HEADER_ATTRIBUTES":
                          "x-team:team.id, x-env:deploy.env"})
    assert t.header_attributes == {"x-team": "team.id", "x-env": "deploy.env"}
