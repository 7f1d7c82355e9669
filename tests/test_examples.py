"""Every example config loads, compiles to a RuntimeConfig, and the
full.yaml surface exercises each feature family."""

import glob
import os

import pytest

from aigw.filterapi.config import load_config_file
from aigw.filterapi.runtime import RuntimeConfig

EXAMPLES = sorted(
    p for p in glob.glob(
        os.path.join(os.path.dirname(__file__), "..", "examples", "*.yaml")
    )
    if "crd_" not in os.path.basename(p)  # CRD bundles have their own test
)


@pytest.mark.parametrize("path", EXAMPLES, ids=[os.path.basename(p) for p in EXAMPLES])
def test_example_loads_and_compiles(path, monkeypatch):
    for var in ("OPENAI_API_KEY", "AZURE_OPENAI_API_KEY", "ANTHROPIC_API_KEY",
                "AWS_ACCESS_KEY_ID", "AWS_SECRET_ACCESS_KEY", "AIGW_MCP_SEED"):
        monkeypatch.setenv(var, "test-" + var.lower())
    cfg = load_config_file(path)
    rc = RuntimeConfig(cfg)
    assert rc.config is cfg


def test_full_example_surface(monkeypatch):
    for var in ("OPENAI_API_KEY", "AZURE_OPENAI_API_KEY", "ANTHROPIC_API_KEY",
                "AWS_ACCESS_KEY_ID", "AWS_SECRET_ACCESS_KEY", "AIGW_MCP_SEED"):
        monkeypatch.setenv(var, "x")
    cfg = load_config_file(
        os.path.join(os.path.dirname(__file__), "..", "examples", "full.yaml")
    )
    r0 = cfg.routes[0]
    assert r0.backends[0].stream_idle_timeout_s == 10
    assert r0.backends[0].max_concurrency == 512
    assert r0.backends[2].priority == 1
    assert r0.backends[2].body_mutation.remove == ["stream_options"]
    assert cfg.routes[1].endpoint_picker
    assert any(m.hosts for m in cfg.models)
    assert cfg.rate_limits[1].key_headers == ["x-org-id"]
    assert cfg.mcp.routes[0].oauth.issuer.startswith("https://")
    assert cfg.mcp.routes[0].backends[0].tool_exclude == ["delete_.*"]
    # the CEL cost compiles and references reasoning_tokens
    assert any(c.type.value == "CEL" for c in cfg.llm_request_costs)


def test_crd_bundle_example_runs_with_env_expansion(monkeypatch):
    monkeypatch.setenv("OPENAI_API_KEY", "sk-live-123")
    from aigw.cli.main import _load_any_config

    cfg, watch, crd_mode = _load_any_config(
        os.path.join(os.path.dirname(__file__), "..", "examples", "crd_bundle.yaml")
    )
    assert watch is not None and crd_mode  # CRD bundles hot-reload via re-translation
    b = cfg.routes[0].backends[0]
    assert b.auth.api_key == "sk-live-123"
    assert b.stream_idle_timeout_s == 15.0
    assert cfg.models[0].owned_by == "my-org"
