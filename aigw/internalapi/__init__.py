"""Shared constants bridging the config, control and data planes.

Mirrors the role of the reference's internal/internalapi/internalapi.go:20-92
(header names, metadata namespaces, ports) — the values are kept
wire-compatible so configs and clients written for the reference behave
identically against this gateway.
"""

# Header carrying the model name extracted from the request body; route
# rules match on it (reference: internalapi.ModelNameHeaderKeyDefault).
MODEL_NAME_HEADER = "x-ai-eg-model"

# Original request path, stashed before provider translation rewrites it.
ORIGINAL_PATH_HEADER = "x-ai-eg-original-path"

# Hostname of the selected upstream (used e.g. as the SigV4 signing host).
UPSTREAM_HOST_HEADER = "x-ai-eg-upstream-host"

# Backend selected for the current try (fallback re-selection rewrites it).
BACKEND_NAME_HEADER = "x-ai-eg-backend-name"

# MCP proxy internals.
MCP_BACKEND_HEADER = "x-ai-eg-mcp-backend"
MCP_SESSION_ID_HEADER = "mcp-session-id"

# Metadata namespaces (kept for config compatibility; this gateway stores
# per-request metadata on the request context rather than in Envoy dynamic
# metadata).
AI_GATEWAY_METADATA_NAMESPACE = "io.envoy.ai_gateway"
EXTENSION_METADATA_NAMESPACE = "aigateway.envoy.io"

# Well-known per-request metadata keys (reference: extproc/processor_impl.go
# buildDynamicMetadata :917 and token-latency merge :739-749).
META_INPUT_TOKENS = "llm_input_token"
META_OUTPUT_TOKENS = "llm_output_token"
META_TOTAL_TOKENS = "llm_total_token"
META_CACHED_INPUT_TOKENS = "llm_cached_input_token"
META_BACKEND_NAME = "ai_service_backend_name"
META_MODEL_NAME = "model_name_override"
META_ROUTE_NAME = "route_name"
META_TTFT = "token_latency_ttft"
META_ITL = "token_latency_itl"

# Headers a client must never be allowed to spoof: stripped at ingress
# (reference: extproc/server.go:439-455).
INTERNAL_HEADERS = (
    MODEL_NAME_HEADER,
    ORIGINAL_PATH_HEADER,
    UPSTREAM_HOST_HEADER,
    BACKEND_NAME_HEADER,
    MCP_BACKEND_HEADER,
)

# Per-request credential override (reference: internalapi.go:77-92).
# Override headers are only honored when the backend config sets
# credentialOverride (filterapi CredentialOverride); the header NAME is
# config-supplied. For AWS the configured value is a PREFIX and the three
# SigV4 header names derive from it, default prefix below.
AWS_CREDENTIAL_OVERRIDE_HEADER_PREFIX = "x-aigw-aws-"


def aws_credential_override_header_names(
    prefix: str = AWS_CREDENTIAL_OVERRIDE_HEADER_PREFIX,
) -> tuple[str, str, str]:
    """(access-key-id, secret-access-key, session-token) header names for a
    configured prefix (reference: AWSCredentialOverrideHeaderNames,
    internalapi.go:86-92)."""
    return (
        prefix + "access-key-id",
        prefix + "secret-access-key",
        prefix + "session-token",
    )

# Default ports (reference: mainlib/main.go:104-121, internalapi.go:48-52).
DEFAULT_LISTEN_PORT = 1975
DEFAULT_ADMIN_PORT = 1064
DEFAULT_MCP_PORT = 9856

# OpenAI / Anthropic / Cohere endpoint path registry — the 16 paths the
# reference registers (cmd/extproc/mainlib/main.go:326-354). Prefixes are
# configurable per deployment; these are the defaults.
OPENAI_PREFIX_DEFAULT = ""
ANTHROPIC_PREFIX_DEFAULT = "/anthropic"
COHERE_PREFIX_DEFAULT = "/cohere"
