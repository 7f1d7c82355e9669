"""Filter-config schema / runtime / watcher tests (parity:
internal/filterapi/filterconfig_test.go, watcher.go)."""

import asyncio
import textwrap

import pytest
import yaml

from aigw.filterapi import (
    APISchemaName,
    ConfigError,
    ConfigWatcher,
    LLMRequestCostType,
    RuntimeConfig,
    load_config,
)

BASIC = textwrap.dedent(
    """
    version: v1
    uuid: abc-123
    llmRequestCosts:
      - metadataKey: llm_total_token
        type: TotalToken
      - metadataKey: my_cel
        type: CEL
        cel: "input_tokens + output_tokens * 2"
    routes:
      - name: r-gpt
        headers:
          - name: x-ai-eg-model
            value: gpt-4o-mini
        backends:
          - name: openai
            schema: OpenAI
            upstream: {host: 127.0.0.1, port: 8080}
            auth: {apiKey: sk-test}
          - name: bedrock-fallback
            priority: 1
            schema: {name: AWSBedrock}
            upstream: {host: 127.0.0.1, port: 8081, hostname: bedrock.us-east-1.amazonaws.com}
            auth: {awsAccessKeyId: AK, awsSecretAccessKey: SK, awsRegion: us-east-1}
            modelNameOverride: anthropic.claude-3
      - name: r-any
        backends:
          - name: default
            upstream: {host: 127.0.0.1, port: 8080}
    models:
      - name: gpt-4o-mini
        ownedBy: openai
      - name: scoped-model
        hosts: [tenant-a.example.com]
    rateLimits:
      - name: global-tokens
        metadataKey: llm_total_token
        limit: 100000
        windowS: 60
    """
)


def test_load_basic():
    cfg = load_config(yaml.safe_load(BASIC))
    assert cfg.uuid == "abc-123"
    assert len(cfg.routes) == 2
    r = cfg.routes[0]
    assert r.backends[0].schema.name is APISchemaName.OPENAI
    assert r.backends[1].schema.name is APISchemaName.AWS_BEDROCK
    assert r.backends[1].priority == 1
    assert r.backends[1].model_name_override == "anthropic.claude-3"
    assert cfg.llm_request_costs[1].type is LLMRequestCostType.CEL


def test_route_selection_and_tiers():
    rc = RuntimeConfig(load_config(yaml.safe_load(BASIC)))
    cr = rc.select_route({"x-ai-eg-model": "gpt-4o-mini"})
    assert cr is not None and cr.route.name == "r-gpt"
    assert len(cr.tiers) == 2
    assert [b.name for b in cr.tiers[0]] == ["openai"]
    assert [b.name for b in cr.tiers[1]] == ["bedrock-fallback"]
    # unmatched model falls to the catch-all route
    cr2 = rc.select_route({"x-ai-eg-model": "unknown"})
    assert cr2.route.name == "r-any"


def test_models_by_host():
    rc = RuntimeConfig(load_config(yaml.safe_load(BASIC)))
    assert [m.name for m in rc.models_for_host("example.org")] == ["gpt-4o-mini"]
    assert sorted(m.name for m in rc.models_for_host("tenant-a.example.com:443")) == [
        "gpt-4o-mini",
        "scoped-model",
    ]


@pytest.mark.parametrize(
    "mutation",
    [
        {"version": "v999"},
        {"routes": [{"name": "x", "backends": []}]},
        {"routes": [{"name": "x", "backends": [{"noName": True}]}]},
        {"llmRequestCosts": [{"metadataKey": "k", "type": "CEL", "cel": "bad ("}]},
        {"rateLimits": [{"name": "rl", "metadataKey": "nope", "limit": 1}]},
        {"bogusField": 1},
    ],
)
def test_invalid_configs(mutation):
    base = yaml.safe_load(BASIC)
    base.update(mutation)
    with pytest.raises(ConfigError):
        load_config(base)


def test_watcher_reload(tmp_path):
    p = tmp_path / "config.yaml"
    p.write_text(BASIC)
    seen = []

    w = ConfigWatcher(str(p), lambda rc: seen.append(rc.uuid), tick_s=0.01)
    rc0 = w.load_once()
    assert rc0.uuid == "abc-123"

    async def run():
        await w.start()
        await asyncio.sleep(0.05)
        p.write_text(BASIC.replace("abc-123", "def-456"))
        for _ in range(100):
            await asyncio.sleep(0.02)
            if seen:
                break
        # a broken config must NOT propagate
        p.write_text("version: v1\nuuid: broken\nroutes: [{name: x, backends: []}]\n")
        await asyncio.sleep(0.1)
        await w.stop()

    asyncio.run(run())
    assert seen == ["def-456"]


def test_regex_header_match():
    cfg = load_config(
        {
            "version": "v1",
            "routes": [
                {
                    "name": "re",
                    "headers": [{"name": "x-ai-eg-model", "regex": "gpt-.*"}],
                    "backends": [{"name": "b", "upstream": {"host": "h", "port": 1}}],
                }
            ],
        }
    )
    rc = RuntimeConfig(cfg)
    assert rc.select_route({"x-ai-eg-model": "gpt-4o"}) is not None
    assert rc.select_route({"x-ai-eg-model": "claude"}) is None


def test_credential_override_config_and_strip_set():
    """credentialOverride parses from YAML config and every configured
    override header name (plus the reserved default AWS trio) lands in the
    runtime egress strip set."""
    from aigw.filterapi.config import load_config
    from aigw.filterapi.runtime import RuntimeConfig

    cfg = load_config({
        "routes": [{
            "name": "r",
            "backends": [
                {"name": "a", "schema": "OpenAI",
                 "upstream": {"host": "h", "port": 80},
                 "auth": {"apiKey": "sk",
                          "credentialOverride": {"headerName": "X-Client-Key",
                                                 "fallbackToConfigured": True}}},
                {"name": "b", "schema": "AWSBedrock",
                 "upstream": {"host": "h2", "port": 80},
                 "auth": {"awsAccessKeyId": "AK", "awsSecretAccessKey": "s",
                          "credentialOverride": {"headerName": "x-corp-aws-"}}},
            ],
        }],
    })
    a = cfg.routes[0].backends[0].auth
    assert a.credential_override.header_name == "X-Client-Key"
    assert a.credential_override.fallback_to_configured is True
    rt = RuntimeConfig(cfg)
    assert "x-client-key" in rt.override_strip_headers
    for h in ("x-corp-aws-access-key-id", "x-corp-aws-secret-access-key",
              "x-corp-aws-session-token"):
        assert h in rt.override_strip_headers
    # reserved defaults are always stripped, configured or not
    assert "x-aigw-aws-access-key-id" in rt.override_strip_headers
