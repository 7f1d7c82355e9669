"""Gateway filter-config schema.

This is the versioned contract between the control plane (controller /
``aigw translate`` / autoconfig) and the data plane (the per-GPU gateway
shard), mirroring the reference's internal/filterapi/filterconfig.go:24-220:
version+UUID gate, global and route-scoped LLM request costs, backends with
schema+auth+mutations, declared models, and MCP config. It is deliberately
not tied to Kubernetes (filterconfig.go:6-12) so the data plane can be
tested standalone.

One deliberate divergence: because this gateway has no Envoy in front of it,
each backend carries its concrete ``upstream`` address (the reference
resolves backends to Envoy clusters via xDS instead). Rate-limit rules are
also first-class here (the reference delegates enforcement to the external
Envoy ratelimit service + Redis; we enforce in-process with RCCL-synced
counters — SURVEY.md §5.8).
"""

from __future__ import annotations

import enum
from dataclasses import dataclass, field
from typing import Optional

import yaml

CURRENT_CONFIG_VERSION = "v1"


class ConfigError(ValueError):
    pass


class APISchemaName(str, enum.Enum):
    """Provider API schema names (filterconfig.go VersionedAPISchema)."""

    OPENAI = "OpenAI"
    AWS_BEDROCK = "AWSBedrock"
    AWS_ANTHROPIC = "AWSAnthropic"
    AZURE_OPENAI = "AzureOpenAI"
    GCP_VERTEX_AI = "GCPVertexAI"
    GCP_ANTHROPIC = "GCPAnthropic"
    ANTHROPIC = "Anthropic"
    COHERE = "Cohere"


@dataclass
class APISchema:
    name: APISchemaName = APISchemaName.OPENAI
    version: str = ""  # e.g. Azure api-version, GCP region qualifier


class LLMRequestCostType(str, enum.Enum):
    INPUT_TOKEN = "InputToken"
    OUTPUT_TOKEN = "OutputToken"
    TOTAL_TOKEN = "TotalToken"
    CACHED_INPUT_TOKEN = "CachedInputToken"
    CEL = "CEL"


@dataclass
class LLMRequestCost:
    """A named cost extracted per request (filterconfig.go LLMRequestCost)."""

    metadata_key: str
    type: LLMRequestCostType = LLMRequestCostType.OUTPUT_TOKEN
    cel: str = ""


@dataclass
class HeaderMatch:
    """Route-rule header match. value="" + regex="" means presence match."""

    name: str
    value: str = ""
    regex: str = ""


@dataclass
class HeaderMutation:
    set: dict[str, str] = field(default_factory=dict)
    remove: list[str] = field(default_factory=list)


@dataclass
class BodyMutation:
    """JSON field set/remove on the request body (bodymutator, sjson paths)."""

    set: dict[str, object] = field(default_factory=dict)  # dotted path -> value
    remove: list[str] = field(default_factory=list)


@dataclass
class OIDCSource:
    """OIDC client-credentials source for credential rotation
    (api/v1beta1 backendsecurity_policy.go BSPOIDC)."""

    issuer: str = ""
    client_id: str = ""
    client_secret: str = ""
    client_secret_file: str = ""
    audience: str = ""
    scopes: list[str] = field(default_factory=list)
    token_endpoint: str = ""  # skip discovery (tests / fixed deployments)


@dataclass
class CredentialRotation:
    """Background credential rotation for this backend's auth
    (internal/controller/rotators/): the rotator refreshes the
    credential FILE the auth handler reads (api_key_file /
    aws_credentials_file), which is the reference's Secret-mount
    contract collapsed to one process.

    kind: aws_oidc | azure | gcp_oidc.
    """

    kind: str = ""
    out_file: str = ""  # defaults to the auth's credential file
    pre_rotation_window_s: float = 300.0
    oidc: Optional[OIDCSource] = None
    # aws_oidc (rotators/aws_oidc_rotator.go)
    aws_role_arn: str = ""
    aws_region: str = ""
    sts_endpoint: str = ""  # test/override hook
    # azure (rotators/azure_token_rotator.go)
    azure_tenant_id: str = ""
    azure_client_id: str = ""
    azure_client_secret: str = ""
    azure_scope: str = ""
    azure_authority: str = ""
    # gcp_oidc (rotators/gcp_oidc_token_rotator.go)
    gcp_project_number: str = ""
    gcp_pool: str = ""
    gcp_provider: str = ""
    gcp_service_account: str = ""
    gcp_sts_endpoint: str = ""
    gcp_iam_endpoint: str = ""


@dataclass
class CredentialOverride:
    """Per-request credential sourcing for a backend (filterconfig.go
    CredentialOverride :221-240). Override headers are only honored when
    this is configured; otherwise client-sent override headers are
    stripped at egress and ignored.

    header_name: request header carrying the credential. For AWS it is a
    PREFIX, not a full name — the three SigV4 header names derive from it
    (internalapi.aws_credential_override_header_names).
    fallback_to_configured: when the source header is absent, True falls
    back to the static credential; False makes the gateway answer 401.
    """

    header_name: str = ""
    fallback_to_configured: bool = False


@dataclass
class BackendAuth:
    """Upstream credential injection (filterconfig.go BackendAuth).

    Exactly one of the kinds should be populated.
    """

    api_key: str = ""  # -> Authorization: Bearer (backendauth/api_key.go)
    anthropic_api_key: str = ""  # -> x-api-key (anthropicapikey.go)
    azure_api_key: str = ""  # -> api-key (azureapikey.go)
    azure_access_token: str = ""  # -> Authorization: Bearer (azure.go)
    gcp_access_token: str = ""  # -> Authorization: Bearer + URL rewrite (gcp.go)
    gcp_project: str = ""
    gcp_region: str = ""
    aws_access_key_id: str = ""  # -> SigV4 signing (aws.go:86-160)
    aws_secret_access_key: str = ""
    aws_session_token: str = ""
    aws_region: str = ""
    # File-based credentials, re-read on change (mtime-cached): the hot-
    # rotation analogue of the reference's BSP rotators, which refresh
    # OIDC/exchanged credentials into mounted Secrets
    # (internal/controller/rotators/). The file content replaces api_key /
    # the AWS credential trio (INI credentials-file format for AWS).
    api_key_file: str = ""
    aws_credentials_file: str = ""
    # Per-request credential sourcing; None = overrides disabled.
    credential_override: Optional["CredentialOverride"] = None
    # Background OIDC/token-exchange rotation of the credential file.
    rotation: Optional["CredentialRotation"] = None

    @property
    def kind(self) -> str:
        if self.api_key_file:
            return "api_key"
        if self.aws_credentials_file:
            return "aws_access_key_id"
        for k in (
            "api_key",
            "anthropic_api_key",
            "azure_api_key",
            "azure_access_token",
            "gcp_access_token",
            "aws_access_key_id",
        ):
            if getattr(self, k):
                return k
        return ""


@dataclass
class Upstream:
    """Concrete upstream address for a backend (replaces Envoy clusters)."""

    host: str = "127.0.0.1"
    port: int = 443
    tls: bool = False
    path_prefix: str = ""  # prepended to the translated path
    hostname: str = ""  # Host header / TLS SNI / SigV4 signing host; defaults to host
    # Custom trust anchor for private upstreams (the BackendTLSPolicy
    # caCertificateRefs analogue): PEM file path or inline PEM. Empty =
    # system trust store. Verification is never disabled.
    ca_file: str = ""
    ca_pem: str = ""

    @property
    def authority(self) -> str:
        h = self.hostname or self.host
        return f"{h}:{self.port}"

    @property
    def base_url(self) -> str:
        scheme = "https" if self.tls else "http"
        return f"{scheme}://{self.host}:{self.port}{self.path_prefix}"


@dataclass
class Backend:
    """One upstream provider binding (filterconfig.go Backend)."""

    name: str
    schema: APISchema = field(default_factory=APISchema)
    upstream: Upstream = field(default_factory=Upstream)
    weight: int = 1
    priority: int = 0  # 0 = primary; >0 = fallback tiers (ai_gateway_route.go:387-397)
    model_name_override: str = ""
    auth: Optional[BackendAuth] = None
    header_mutation: Optional[HeaderMutation] = None
    body_mutation: Optional[BodyMutation] = None
    # Per-try timeout seconds (extensionserver/post_translate_modify.go:206-300)
    timeout_s: float = 60.0
    # Per-try idle timeout: max gap without upstream bytes. Before the
    # first response byte it resets the try and fallback proceeds; after
    # bytes have flowed the stream is cut (ai_gateway_route.go
    # StreamIdleTimeout -> per_try_idle_timeout). 0 = off.
    stream_idle_timeout_s: float = 0.0
    # Circuit breaking: max in-flight requests to this backend (parity with
    # the Envoy cluster circuit breakers the reference relies on; 0 = off).
    # A saturated backend counts as a failed attempt and fallback proceeds.
    max_concurrency: int = 0
    # Replica telemetry for the endpoint picker: poll this backend's
    # metrics endpoint (vLLM Prometheus format) so KV-occupancy scoring
    # runs on REAL replica load instead of gateway-local estimates
    # (InferencePool EPP inputs, extensionserver/inferencepool.go:39-54).
    telemetry: Optional["BackendTelemetry"] = None


@dataclass
class BackendTelemetry:
    """Metrics scrape config for one replica backend."""

    path: str = "/metrics"
    interval_s: float = 1.0
    # Prometheus gauge names (vLLM defaults); override for other servers
    kv_usage_metric: str = "vllm:gpu_cache_usage_perc"  # 0..1 fraction
    running_metric: str = "vllm:num_requests_running"
    waiting_metric: str = "vllm:num_requests_waiting"
    # KV block capacity used to convert the usage fraction into blocks
    kv_total: float = 100000.0


@dataclass
class RateLimitRule:
    """Token-budget rule enforced by the gateway (reference: QuotaPolicy +
    Envoy ratelimit service; here enforced in-process, RCCL-synced)."""

    name: str
    metadata_key: str  # which LLMRequestCost feeds this bucket
    limit: int  # tokens per window, across ALL shards
    window_s: float = 60.0
    key_headers: list[str] = field(default_factory=list)  # bucket per header value


@dataclass
class InferencePool:
    """Dynamic pool membership (InferencePool analogue). The reference
    resolves pool members behind a k8s selector and rewrites the Envoy
    cluster per member (extensionserver/inferencepool.go:39-54,
    post_cluster_modify.go:32); the single-node analogue re-resolves a
    DNS name (headless-service style A records) or a members file on an
    interval and swaps the runtime with the refreshed member set."""

    # DNS name resolved to A records; every address becomes a member
    service: str = ""
    # newline-separated host:port list (airgapped / file-managed pools);
    # takes precedence over `service` when both are set
    members_file: str = ""
    # member port when resolving via DNS (members_file carries its own)
    port: int = 0
    interval_s: float = 5.0
    # template applied to every resolved member
    schema: APISchema = field(default_factory=APISchema)
    telemetry: Optional["BackendTelemetry"] = None
    auth: Optional[BackendAuth] = None
    timeout_s: float = 60.0
    max_concurrency: int = 0


@dataclass
class Route:
    """One routing rule (filterconfig.go Config.Rules).

    Matched in order of declaration against the model header + any extra
    header matches; the first match wins (HTTPRoute semantics).
    """

    name: str
    backends: list[Backend] = field(default_factory=list)
    headers: list[HeaderMatch] = field(default_factory=list)
    model_name_override: str = ""
    request_costs: list[LLMRequestCost] = field(default_factory=list)
    header_mutation: Optional[HeaderMutation] = None
    # Retry budget across priority tiers (numAttemptsPerPriority analogue).
    retries: int = 1
    # InferencePool-style endpoint picking: rank same-priority backends by
    # the on-GPU KV-occupancy scorer instead of weighted random
    # (replaces the reference's external EPP service —
    # extensionserver/inferencepool.go:39-54).
    endpoint_picker: bool = False
    # Dynamic member resolution; resolved members are APPENDED to the
    # static `backends` list by aigw.extproc.pool.PoolManager
    pool: Optional[InferencePool] = None


@dataclass
class Model:
    """Declared model served from /v1/models (models_processor.go:40-62)."""

    name: str
    owned_by: str = "aigw"
    created_at: int = 0
    hosts: list[str] = field(default_factory=list)  # empty = unscoped


@dataclass
class MCPBackend:
    name: str
    upstream: Upstream = field(default_factory=Upstream)
    path: str = "/mcp"
    tool_include: list[str] = field(default_factory=list)
    tool_exclude: list[str] = field(default_factory=list)
    headers: Optional[HeaderMutation] = None
    auth: Optional[BackendAuth] = None


@dataclass
class MCPOAuth:
    """JWT validation of client access tokens (mcp_route.go securityPolicy
    oauth: issuer/audiences/jwks). JWKS is inline JSON or a file path —
    remote JWKS discovery needs egress and is not available here."""

    issuer: str = ""
    audiences: list[str] = field(default_factory=list)
    jwks: str = ""       # inline JWKS JSON (localJWKS analogue)
    jwks_file: str = ""
    # OAuth protected-resource metadata (RFC 9728) served at
    # /.well-known/oauth-protected-resource (mcp_route_security_policy.go
    # buildOAuthProtectedResourceMetadataJSON :474-502)
    resource: str = ""
    resource_name: str = ""
    scopes_supported: list[str] = field(default_factory=list)


@dataclass
class MCPAuthorizationRule:
    """One authorization rule (mcpconfig.go MCPRouteAuthorizationRule
    :126-170): action + optional CEL condition + optional tool target +
    optional JWT source (scopes: ALL required; claims: each must exist
    with one of the allowed values, dotted paths supported)."""

    action: str = "Allow"  # Allow | Deny
    cel: str = ""
    tools: list[str] = field(default_factory=list)  # backend__tool names
    jwt_scopes: list[str] = field(default_factory=list)
    jwt_claims: list[dict] = field(default_factory=list)  # {name, values}


@dataclass
class MCPAuthorization:
    """Route-level authorization (mcpconfig.go MCPRouteAuthorization
    :101-115): first matching rule wins, else defaultAction; scope-only
    denials answer 403 with an insufficient_scope WWW-Authenticate
    challenge pointing at resourceMetadataURL."""

    default_action: str = "Deny"  # Allow | Deny
    rules: list[MCPAuthorizationRule] = field(default_factory=list)
    resource_metadata_url: str = ""


@dataclass
class MCPRoute:
    name: str
    path: str = "/mcp"
    backends: list[MCPBackend] = field(default_factory=list)
    oauth: Optional[MCPOAuth] = None
    # client authorization at the gateway (mcpconfig.go:102-117): requests
    # must carry this bearer token; failures answer 401 with an OAuth
    # protected-resource-metadata WWW-Authenticate challenge when
    # resource_metadata_url is set
    bearer_token: str = ""
    resource_metadata_url: str = ""
    # fine-grained CEL/scope/claim rules evaluated per request after the
    # authentication gate (authorization.go)
    authorization: Optional[MCPAuthorization] = None


@dataclass
class MCPConfig:
    routes: list[MCPRoute] = field(default_factory=list)
    session_seed: str = "aigw-mcp"
    # previous seed(s) still accepted for DEcryption during seed rotation
    # (mainlib --mcpFallbackSessionEncryptionSeed)
    fallback_session_seed: str = ""


@dataclass
class Config:
    """Root gateway config (filterconfig.go Config)."""

    version: str = CURRENT_CONFIG_VERSION
    uuid: str = ""
    model_name_header_key: str = "x-ai-eg-model"
    routes: list[Route] = field(default_factory=list)
    models: list[Model] = field(default_factory=list)
    llm_request_costs: list[LLMRequestCost] = field(default_factory=list)
    rate_limits: list[RateLimitRule] = field(default_factory=list)
    mcp: Optional[MCPConfig] = None


# --- YAML (de)serialization -------------------------------------------------


def _dc(cls, d: dict, ctx: str):
    if not isinstance(d, dict):
        raise ConfigError(f"{ctx}: expected mapping, got {type(d).__name__}")
    fields = {f: t for f, t in cls.__dataclass_fields__.items()}  # noqa: C416
    kwargs = {}
    for key, val in d.items():
        pykey = _snake(key)
        if pykey not in fields:
            raise ConfigError(f"{ctx}: unknown field {key!r}")
        kwargs[pykey] = val
    return kwargs


def _snake(name: str) -> str:
    out = []
    for i, ch in enumerate(name):
        if ch.isupper():
            if i and (not name[i - 1].isupper() or (i + 1 < len(name) and name[i + 1].islower())):
                out.append("_")
            out.append(ch.lower())
        else:
            out.append(ch)
    return "".join(out)


def _parse_schema(d, ctx) -> APISchema:
    if isinstance(d, str):
        d = {"name": d}
    kw = _dc(APISchema, d, ctx)
    try:
        kw["name"] = APISchemaName(kw.get("name", "OpenAI"))
    except ValueError as e:
        raise ConfigError(f"{ctx}: unknown schema {kw.get('name')!r}") from e
    return APISchema(**kw)


def _parse_cost(d, ctx) -> LLMRequestCost:
    kw = _dc(LLMRequestCost, d, ctx)
    if "metadata_key" not in kw:
        raise ConfigError(f"{ctx}: metadataKey is required")
    kw["type"] = LLMRequestCostType(kw.get("type", "OutputToken"))
    c = LLMRequestCost(**kw)
    if c.type is LLMRequestCostType.CEL:
        if not c.cel:
            raise ConfigError(f"{ctx}: CEL cost requires 'cel' expression")
        from aigw.llmcost import compile_program
        from aigw.llmcost.cel import CostExpressionError

        try:
            compile_program(c.cel)  # validate at load time (gateway.go:201)
        except CostExpressionError as e:
            raise ConfigError(f"{ctx}: invalid CEL expression: {e}") from e
    return c


def _parse_backend_auth(d, ctx) -> BackendAuth:
    kw = _dc(BackendAuth, d, ctx)
    if kw.get("credential_override") is not None:
        kw["credential_override"] = CredentialOverride(
            **_dc(CredentialOverride, kw["credential_override"],
                  f"{ctx}.credentialOverride")
        )
    if kw.get("rotation") is not None:
        rkw = _dc(CredentialRotation, kw["rotation"], f"{ctx}.rotation")
        if rkw.get("oidc") is not None:
            rkw["oidc"] = OIDCSource(**_dc(OIDCSource, rkw["oidc"],
                                           f"{ctx}.rotation.oidc"))
        rot = CredentialRotation(**rkw)
        if rot.kind not in ("aws_oidc", "azure", "gcp_oidc"):
            raise ConfigError(f"{ctx}.rotation: unknown kind {rot.kind!r}")
        kw["rotation"] = rot
    return BackendAuth(**kw)


def _parse_backend(d, ctx) -> Backend:
    kw = _dc(Backend, d, ctx)
    if "name" not in kw:
        raise ConfigError(f"{ctx}: backend name is required")
    if "schema" in kw:
        kw["schema"] = _parse_schema(kw["schema"], f"{ctx}.schema")
    if "upstream" in kw:
        kw["upstream"] = Upstream(**_dc(Upstream, kw["upstream"], f"{ctx}.upstream"))
    if kw.get("auth") is not None:
        kw["auth"] = _parse_backend_auth(kw["auth"], f"{ctx}.auth")
    if kw.get("header_mutation") is not None:
        kw["header_mutation"] = HeaderMutation(
            **_dc(HeaderMutation, kw["header_mutation"], f"{ctx}.headerMutation")
        )
    if kw.get("body_mutation") is not None:
        kw["body_mutation"] = BodyMutation(
            **_dc(BodyMutation, kw["body_mutation"], f"{ctx}.bodyMutation")
        )
    if kw.get("telemetry") is not None:
        kw["telemetry"] = BackendTelemetry(
            **_dc(BackendTelemetry, kw["telemetry"], f"{ctx}.telemetry")
        )
    return Backend(**kw)


def _parse_route(d, ctx) -> Route:
    kw = _dc(Route, d, ctx)
    if "name" not in kw:
        raise ConfigError(f"{ctx}: route name is required")
    kw["backends"] = [
        _parse_backend(b, f"{ctx}.backends[{i}]") for i, b in enumerate(kw.get("backends", []))
    ]
    kw["headers"] = [
        HeaderMatch(**_dc(HeaderMatch, h, f"{ctx}.headers[{i}]"))
        for i, h in enumerate(kw.get("headers", []))
    ]
    kw["request_costs"] = [
        _parse_cost(c, f"{ctx}.requestCosts[{i}]") for i, c in enumerate(kw.get("request_costs", []))
    ]
    if kw.get("header_mutation") is not None:
        kw["header_mutation"] = HeaderMutation(
            **_dc(HeaderMutation, kw["header_mutation"], f"{ctx}.headerMutation")
        )
    if kw.get("pool") is not None:
        pkw = _dc(InferencePool, kw["pool"], f"{ctx}.pool")
        if "schema" in pkw:
            pkw["schema"] = _parse_schema(pkw["schema"], f"{ctx}.pool.schema")
        if pkw.get("auth") is not None:
            pkw["auth"] = _parse_backend_auth(pkw["auth"], f"{ctx}.pool.auth")
        if pkw.get("telemetry") is not None:
            pkw["telemetry"] = BackendTelemetry(
                **_dc(BackendTelemetry, pkw["telemetry"], f"{ctx}.pool.telemetry")
            )
        pool = InferencePool(**pkw)
        if not pool.service and not pool.members_file:
            raise ConfigError(f"{ctx}.pool: service or membersFile is required")
        if pool.service and pool.port <= 0:
            raise ConfigError(f"{ctx}.pool: port is required with service")
        kw["pool"] = pool
    return Route(**kw)


def load_config(data: object) -> Config:
    """Build a Config from parsed YAML/JSON, validating every field."""

    if isinstance(data, (str, bytes)):
        data = yaml.safe_load(data)
    if not isinstance(data, dict):
        raise ConfigError("config root must be a mapping")
    kw = _dc(Config, data, "config")
    version = kw.get("version", CURRENT_CONFIG_VERSION)
    if version != CURRENT_CONFIG_VERSION:
        # Version gate against mixed-version corruption during rolling
        # upgrades (filterconfig.go:26-31).
        raise ConfigError(f"unsupported config version {version!r}")
    kw["routes"] = [_parse_route(r, f"routes[{i}]") for i, r in enumerate(kw.get("routes", []))]
    kw["models"] = [
        Model(**_dc(Model, m, f"models[{i}]")) for i, m in enumerate(kw.get("models", []))
    ]
    kw["llm_request_costs"] = [
        _parse_cost(c, f"llmRequestCosts[{i}]")
        for i, c in enumerate(kw.get("llm_request_costs", []))
    ]
    kw["rate_limits"] = [
        RateLimitRule(**_dc(RateLimitRule, r, f"rateLimits[{i}]"))
        for i, r in enumerate(kw.get("rate_limits", []))
    ]
    if kw.get("mcp") is not None:
        mkw = _dc(MCPConfig, kw["mcp"], "mcp")
        routes = []
        for i, r in enumerate(mkw.get("routes", [])):
            rkw = _dc(MCPRoute, r, f"mcp.routes[{i}]")
            backends = []
            for j, b in enumerate(rkw.get("backends", [])):
                bkw = _dc(MCPBackend, b, f"mcp.routes[{i}].backends[{j}]")
                if "upstream" in bkw:
                    bkw["upstream"] = Upstream(**_dc(Upstream, bkw["upstream"], "upstream"))
                if bkw.get("headers") is not None:
                    bkw["headers"] = HeaderMutation(**_dc(HeaderMutation, bkw["headers"], "headers"))
                if bkw.get("auth") is not None:
                    bkw["auth"] = _parse_backend_auth(bkw["auth"], "auth")
                backends.append(MCPBackend(**bkw))
            rkw["backends"] = backends
            if rkw.get("oauth") is not None:
                rkw["oauth"] = MCPOAuth(
                    **_dc(MCPOAuth, rkw["oauth"], f"mcp.routes[{i}].oauth")
                )
            if rkw.get("authorization") is not None:
                akw = _dc(MCPAuthorization, rkw["authorization"],
                          f"mcp.routes[{i}].authorization")
                akw["rules"] = [
                    MCPAuthorizationRule(**_dc(
                        MCPAuthorizationRule, r,
                        f"mcp.routes[{i}].authorization.rules[{j}]"))
                    for j, r in enumerate(akw.get("rules", []))
                ]
                auth_cfg = MCPAuthorization(**akw)
                if auth_cfg.default_action not in ("Allow", "Deny"):
                    raise ConfigError(
                        f"mcp.routes[{i}].authorization: defaultAction must "
                        "be Allow or Deny")
                rkw["authorization"] = auth_cfg
            routes.append(MCPRoute(**rkw))
        mkw["routes"] = routes
        kw["mcp"] = MCPConfig(**mkw)
    cfg = Config(**kw)
    _validate(cfg)
    return cfg


def _validate(cfg: Config) -> None:
    cost_keys = {c.metadata_key for c in cfg.llm_request_costs}
    for r in cfg.routes:
        if not r.backends and r.pool is None:
            # a pool route may start empty: members arrive from the
            # first PoolManager resolution sweep before traffic
            raise ConfigError(f"route {r.name!r} has no backends")
        for c in r.request_costs:
            cost_keys.add(c.metadata_key)
        names = [b.name for b in r.backends]
        if len(set(names)) != len(names):
            raise ConfigError(f"route {r.name!r} has duplicate backend names")
    for rl in cfg.rate_limits:
        if rl.metadata_key not in cost_keys and rl.metadata_key not in (
            "llm_input_token",
            "llm_output_token",
            "llm_total_token",
        ):
            raise ConfigError(
                f"rateLimit {rl.name!r} references unknown cost metadata key {rl.metadata_key!r}"
            )
        if rl.limit <= 0:
            raise ConfigError(f"rateLimit {rl.name!r}: limit must be positive")


def _camel(s: str) -> str:
    parts = s.split("_")
    return parts[0] + "".join(p.title() for p in parts[1:])


def config_to_dict(obj) -> object:
    """Serialize a Config (or any nested part) back to camelCase YAML-ready
    dicts, omitting fields left at their defaults."""
    import dataclasses
    import enum as _enum

    if dataclasses.is_dataclass(obj):
        out = {}
        for f in dataclasses.fields(obj):
            v = getattr(obj, f.name)
            default = (
                f.default
                if f.default is not dataclasses.MISSING
                else (f.default_factory() if f.default_factory is not dataclasses.MISSING else dataclasses.MISSING)
            )
            if default is not dataclasses.MISSING and v == default:
                continue
            out[_camel(f.name)] = config_to_dict(v)
        return out
    if isinstance(obj, _enum.Enum):
        return obj.value
    if isinstance(obj, list):
        return [config_to_dict(x) for x in obj]
    if isinstance(obj, dict):
        return {k: config_to_dict(v) for k, v in obj.items()}
    return obj


def dump_config_yaml(cfg: Config) -> str:
    d = config_to_dict(cfg)
    d.setdefault("version", cfg.version)
    return yaml.safe_dump(d, sort_keys=False)


def _expand_env(text: str) -> str:
    """``${VAR}`` substitution from the environment (the reference's
    substitution.aigw.run/env annotations, cmd/aigw/run.go:52-55). Unset
    variables are left verbatim so configs fail loudly downstream."""
    import os
    import re as _re

    return _re.sub(
        r"\$\{(\w+)\}", lambda m: os.environ.get(m.group(1), m.group(0)), text
    )


def load_config_file(path: str) -> Config:
    with open(path, "r", encoding="utf-8") as f:
        return load_config(yaml.safe_load(_expand_env(f.read())))
