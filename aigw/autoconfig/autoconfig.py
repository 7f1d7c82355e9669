"""Zero-config startup: synthesize a gateway config from provider env vars.

Parity with internal/autoconfig (config.go:74, openai.go:22): when
``aigw run`` is started without a config file, OPENAI_API_KEY /
ANTHROPIC_API_KEY / AZURE_OPENAI_API_KEY (+ *_BASE_URL overrides) produce
routes for the corresponding providers; every declared model routes by the
``x-ai-eg-model`` header exactly like a translated CRD bundle would.
"""

from __future__ import annotations

import os
from urllib.parse import urlparse

from aigw.filterapi.config import (
    APISchema,
    APISchemaName,
    Backend,
    BackendAuth,
    Config,
    HeaderMatch,
    LLMRequestCost,
    LLMRequestCostType,
    Route,
    Upstream,
)


def _upstream_from_url(url: str, default_host: str, default_port: int = 443) -> Upstream:
    if not url:
        return Upstream(host=default_host, port=default_port, tls=default_port == 443,
                        hostname=default_host)
    u = urlparse(url)
    port = u.port or (443 if u.scheme == "https" else 80)
    return Upstream(
        host=u.hostname or default_host,
        port=port,
        tls=u.scheme == "https",
        hostname=u.hostname or default_host,
        path_prefix=u.path.rstrip("/"),
    )


def config_from_env(env: dict | None = None) -> Config:
    env = env if env is not None else dict(os.environ)
    routes: list[Route] = []

    if env.get("OPENAI_API_KEY"):
        up = _upstream_from_url(env.get("OPENAI_BASE_URL", ""), "api.openai.com")
        routes.append(
            Route(
                name="openai",
                headers=[HeaderMatch(name="x-ai-eg-model", regex="gpt-.*|o[0-9].*|davinci.*|text-embedding.*")],
                backends=[
                    Backend(
                        name="openai",
                        schema=APISchema(name=APISchemaName.OPENAI),
                        upstream=up,
                        auth=BackendAuth(api_key=env["OPENAI_API_KEY"]),
                    )
                ],
            )
        )
    if env.get("ANTHROPIC_API_KEY"):
        up = _upstream_from_url(env.get("ANTHROPIC_BASE_URL", ""), "api.anthropic.com")
        routes.append(
            Route(
                name="anthropic",
                headers=[HeaderMatch(name="x-ai-eg-model", regex="claude.*")],
                backends=[
                    Backend(
                        name="anthropic",
                        schema=APISchema(name=APISchemaName.ANTHROPIC),
                        upstream=up,
                        auth=BackendAuth(anthropic_api_key=env["ANTHROPIC_API_KEY"]),
                    )
                ],
            )
        )
    if env.get("AZURE_OPENAI_API_KEY") and env.get("AZURE_OPENAI_ENDPOINT"):
        up = _upstream_from_url(env["AZURE_OPENAI_ENDPOINT"], "")
        routes.append(
            Route(
                name="azure-openai",
                headers=[],
                backends=[
                    Backend(
                        name="azure-openai",
                        schema=APISchema(
                            name=APISchemaName.AZURE_OPENAI,
                            version=env.get("AZURE_OPENAI_API_VERSION", "2025-01-01-preview"),
                        ),
                        upstream=up,
                        auth=BackendAuth(azure_api_key=env["AZURE_OPENAI_API_KEY"]),
                    )
                ],
            )
        )
    if not routes:
        raise ValueError(
            "no provider credentials in environment (OPENAI_API_KEY / "
            "ANTHROPIC_API_KEY / AZURE_OPENAI_API_KEY+AZURE_OPENAI_ENDPOINT) "
            "and no config file given"
        )
    return Config(
        uuid="autoconfig",
        routes=routes,
        llm_request_costs=[
            LLMRequestCost(metadata_key="llm_total_token", type=LLMRequestCostType.TOTAL_TOKEN)
        ],
    )
