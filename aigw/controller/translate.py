"""CRD-compatible declarative config surface.

Accepts the SAME YAML documents a user feeds Envoy AI Gateway —
``AIGatewayRoute``, ``AIServiceBackend``, ``BackendSecurityPolicy``,
Envoy Gateway ``Backend``, ``BackendTrafficPolicy`` (Global rateLimit),
``GatewayConfig``, ``MCPRoute``, core ``Secret`` — and compiles them into
one aigw filter Config, playing the role of the reference's Gateway
reconciler (internal/controller/gateway.go:88: aggregate routes, backends
and BSPs attached to a Gateway into one filterapi.Config). Kinds that only
exist to steer Envoy (GatewayClass, EnvoyProxy, ClientTrafficPolicy,
HTTPRoute, Deployment, Service, ...) are accepted and ignored, so the
reference's example bundles (examples/basic/*.yaml, token_ratelimit/...)
translate as-is.

This is the offline path (``aigw translate``) and the load path of
``aigw run`` — equivalent to cmd/aigw/translate.go:41 running the
reconcilers against a fake client.
"""

from __future__ import annotations

import base64
import logging
from typing import Optional

import yaml

logger = logging.getLogger("aigw.controller")

from aigw.filterapi.config import (
    APISchema,
    APISchemaName,
    Backend,
    BackendAuth,
    Config,
    ConfigError,
    HeaderMatch,
    LLMRequestCost,
    LLMRequestCostType,
    MCPBackend,
    MCPConfig,
    MCPRoute,
    Model,
    RateLimitRule,
    Route,
    Upstream,
)

_IGNORED_KINDS = {
    "GatewayClass",
    "EnvoyProxy",
    "ClientTrafficPolicy",
    "HTTPRoute",
    "HTTPRouteFilter",
    "Deployment",
    "Service",
    "ServiceAccount",
    "ConfigMap",
    "Namespace",
    "ReferenceGrant",
    "EnvoyExtensionPolicy",
    "SecurityPolicy",
    "InferencePool",
    "EnvoyPatchPolicy",
    "GRPCRoute",
    "Job",
    "Pod",
}

_UNIT_SECONDS = {"Second": 1.0, "Minute": 60.0, "Hour": 3600.0, "Day": 86400.0}


class _Bundle:
    def __init__(self, docs: list[dict]):
        self.by_kind: dict[str, list[dict]] = {}
        for d in docs:
            if not isinstance(d, dict) or "kind" not in d:
                continue
            self.by_kind.setdefault(d["kind"], []).append(d)

    def find(self, kind: str, name: str, namespace: Optional[str] = None) -> Optional[dict]:
        for d in self.by_kind.get(kind, []):
            md = d.get("metadata") or {}
            if md.get("name") == name and (
                namespace is None or md.get("namespace", "default") == namespace
            ):
                return d
        return None


def _secret_value(bundle: _Bundle, name: str, namespace: Optional[str], key_hint=None) -> str:
    sec = bundle.find("Secret", name, namespace)
    if sec is None:
        raise ConfigError(f"Secret {name!r} not found in bundle")
    data = {}
    for k, v in (sec.get("stringData") or {}).items():
        data[k] = v
    for k, v in (sec.get("data") or {}).items():
        data.setdefault(k, base64.b64decode(v).decode("utf-8"))
    if not data:
        raise ConfigError(f"Secret {name!r} has no data")
    if key_hint:
        for k in key_hint:
            if k in data:
                return data[k]
    return next(iter(data.values()))


def _parse_aws_credentials_file(text: str) -> dict[str, str]:
    out: dict[str, str] = {}
    for line in text.splitlines():
        line = line.strip()
        if not line or line.startswith(("#", "[", ";")):
            continue
        k, _, v = line.partition("=")
        out[k.strip().lower()] = v.strip()
    return out


def _bsp_for_backend(bundle: _Bundle, ns: str, asb_name: str) -> Optional[BackendAuth]:
    """Find the BackendSecurityPolicy targeting an AIServiceBackend and
    build the auth config (controller/gateway.go:846 bspToFilterAPIBackendAuth)."""
    for bsp in bundle.by_kind.get("BackendSecurityPolicy", []):
        spec = bsp.get("spec") or {}
        targets = spec.get("targetRefs") or []
        if not any(t.get("name") == asb_name for t in targets):
            continue
        t = spec.get("type")
        if t == "APIKey":
            ref = (spec.get("apiKey") or {}).get("secretRef") or {}
            inline = (spec.get("apiKey") or {}).get("inline")
            key = inline or _secret_value(
                bundle, ref.get("name", ""), ref.get("namespace", ns), ["apiKey", "api-key"]
            )
            return BackendAuth(api_key=key.strip())
        if t == "AnthropicAPIKey":
            ref = (spec.get("anthropicAPIKey") or {}).get("secretRef") or {}
            key = _secret_value(
                bundle, ref.get("name", ""), ref.get("namespace", ns), ["apiKey", "api-key"]
            )
            return BackendAuth(anthropic_api_key=key.strip())
        if t == "AzureAPIKey":
            ref = (spec.get("azureAPIKey") or {}).get("secretRef") or {}
            key = _secret_value(bundle, ref.get("name", ""), ref.get("namespace", ns), ["apiKey"])
            return BackendAuth(azure_api_key=key.strip())
        if t == "AzureCredentials":
            az = spec.get("azureCredentials") or {}
            ref = (az.get("clientSecretRef") or {})
            token = _secret_value(
                bundle, ref.get("name", ""), ref.get("namespace", ns), ["client-secret"]
            )
            return BackendAuth(azure_access_token=token.strip())
        if t == "AWSCredentials":
            aws = spec.get("awsCredentials") or {}
            region = aws.get("region", "us-east-1")
            cf = (aws.get("credentialsFile") or {}).get("secretRef") or {}
            if cf:
                creds = _parse_aws_credentials_file(
                    _secret_value(bundle, cf.get("name", ""), cf.get("namespace", ns),
                                  ["credentials"])
                )
                return BackendAuth(
                    aws_access_key_id=creds.get("aws_access_key_id", ""),
                    aws_secret_access_key=creds.get("aws_secret_access_key", ""),
                    aws_session_token=creds.get("aws_session_token", ""),
                    aws_region=region,
                )
            # OIDC rotation reduces to env/file credentials at runtime; use
            # region-only auth so SigV4 still signs with overrides.
            return BackendAuth(aws_region=region, aws_access_key_id="anonymous",
                               aws_secret_access_key="anonymous")
        if t == "GCPCredentials":
            gcp = spec.get("gcpCredentials") or {}
            ref = (gcp.get("credentialsFileRef") or gcp.get("secretRef") or {})
            token = ""
            if ref:
                token = _secret_value(
                    bundle, ref.get("name", ""), ref.get("namespace", ns),
                    ["token", "access-token", "credentials"],
                )
            return BackendAuth(
                gcp_access_token=token.strip(),
                gcp_project=gcp.get("projectName", gcp.get("project", "")),
                gcp_region=gcp.get("region", ""),
            )
    return None


def _apply_tls_policy(bundle: _Bundle, ns: str, backend_name: str, up: Upstream) -> Upstream:
    """BackendTLSPolicy targeting this Backend: TLS on, SNI/verify hostname
    from validation.hostname, and a private trust anchor from
    caCertificateRefs (ConfigMap ca.crt) when not wellKnownCACertificates."""
    for pol in bundle.by_kind.get("BackendTLSPolicy", []):
        spec = pol.get("spec") or {}
        refs = spec.get("targetRefs") or []
        if not any(r.get("kind") == "Backend" and r.get("name") == backend_name
                   for r in refs):
            continue
        v = spec.get("validation") or {}
        up.tls = True
        if v.get("hostname"):
            up.hostname = v["hostname"]
        for cref in v.get("caCertificateRefs") or []:
            cm = bundle.find(cref.get("kind", "ConfigMap"), cref.get("name", ""), ns)
            if cm is not None:
                pem = (cm.get("data") or {}).get("ca.crt", "")
                if pem:
                    up.ca_pem = pem
        break
    return up


def _resolve_upstream(bundle: _Bundle, ns: str, backend_ref: dict) -> Upstream:
    name = backend_ref.get("name", "")
    kind = backend_ref.get("kind", "Backend")
    if kind == "Service":
        svc = bundle.find("Service", name, ns)
        port = backend_ref.get("port", 80)
        host = f"{name}.{ns}.svc.cluster.local" if svc is None else name
        return Upstream(host=host, port=port, hostname=host)
    eb = bundle.find("Backend", name, ns)
    if eb is None:
        raise ConfigError(f"Backend {name!r} not found in bundle")
    eps = (eb.get("spec") or {}).get("endpoints") or []
    if not eps:
        raise ConfigError(f"Backend {name!r} has no endpoints")
    ep = eps[0]
    if "fqdn" in ep:
        host = ep["fqdn"].get("hostname", "")
        port = ep["fqdn"].get("port", 443)
        up = Upstream(host=host, port=port, tls=port == 443, hostname=host)
        return _apply_tls_policy(bundle, ns, name, up)
    if "ip" in ep:
        up = Upstream(host=ep["ip"].get("address", ""), port=ep["ip"].get("port", 80))
        return _apply_tls_policy(bundle, ns, name, up)
    if "unix" in ep:
        raise ConfigError("unix-socket backends are not supported")
    raise ConfigError(f"Backend {name!r}: unknown endpoint type")


def _parse_costs(costs: list) -> list[LLMRequestCost]:
    out = []
    for c in costs or []:
        t = c.get("type", "OutputToken")
        # v1beta1 extra types fold into CEL so the core enum stays small
        extra = {
            "CacheCreationInputToken": "cache_creation_input_tokens",
            "ReasoningToken": "reasoning_tokens",
        }
        if t in extra:
            out.append(
                LLMRequestCost(
                    metadata_key=c["metadataKey"],
                    type=LLMRequestCostType.CEL,
                    cel=extra[t],
                )
            )
            continue
        out.append(
            LLMRequestCost(
                metadata_key=c["metadataKey"],
                type=LLMRequestCostType(t),
                cel=c.get("cel", ""),
            )
        )
    return out


def translate_crds(docs: list[dict]) -> Config:
    # typed validation first (the kube-apiserver's CRD/CEL gate analogue,
    # aigw/controller/crds.py): malformed bundles fail with field-level
    # errors before any compilation
    from aigw.controller.crds import CRDValidationError, validate_bundle

    try:
        validate_bundle(docs)
    except CRDValidationError as e:
        raise ConfigError(str(e)) from e
    return _translate_crds_validated(docs)


def _translate_crds_validated(docs: list[dict]) -> Config:
    bundle = _Bundle(docs)
    routes: list[Route] = []
    models: list[Model] = []
    seen_models: set[str] = set()
    global_costs: list[LLMRequestCost] = []
    rate_limits: list[RateLimitRule] = []
    mcp_routes: list[MCPRoute] = []

    for gc in bundle.by_kind.get("GatewayConfig", []):
        global_costs.extend(_parse_costs((gc.get("spec") or {}).get("llmRequestCosts")))

    for route_doc in bundle.by_kind.get("AIGatewayRoute", []):
        md = route_doc.get("metadata") or {}
        ns = md.get("namespace", "default")
        spec = route_doc.get("spec") or {}
        route_costs = _parse_costs(spec.get("llmRequestCosts"))
        for i, rule in enumerate(spec.get("rules") or []):
            # per-rule /v1/models attribution (ai_gateway_route.go
            # ModelsOwnedBy/ModelsCreatedAt — rule-level fields)
            owned_by = rule.get("modelsOwnedBy") or "aigw"
            created_at = 0
            if rule.get("modelsCreatedAt"):
                import datetime as _dt

                try:
                    created_at = int(
                        _dt.datetime.fromisoformat(
                            str(rule["modelsCreatedAt"]).replace("Z", "+00:00")
                        ).timestamp()
                    )
                except ValueError:
                    created_at = 0
            headers: list[HeaderMatch] = []
            for match in rule.get("matches") or []:
                for h in match.get("headers") or []:
                    mtype = h.get("type", "Exact")
                    if mtype == "RegularExpression":
                        headers.append(HeaderMatch(name=h["name"], regex=h.get("value", "")))
                    else:
                        headers.append(HeaderMatch(name=h["name"], value=h.get("value", "")))
                    if h.get("name") == "x-ai-eg-model" and mtype == "Exact":
                        v = h.get("value", "")
                        if v and v not in seen_models:
                            seen_models.add(v)
                            models.append(
                                Model(name=v, owned_by=owned_by, created_at=created_at)
                            )
            backends: list[Backend] = []
            for bref in rule.get("backendRefs") or []:
                asb_name = bref.get("name", "")
                asb = bundle.find("AIServiceBackend", asb_name, ns)
                if asb is None:
                    # cross-file reference (bundles are often split across
                    # files); the controller leaves such rules unresolved
                    # rather than failing the whole Gateway
                    logger.warning(
                        "translate: AIServiceBackend %r not in bundle; skipping ref",
                        asb_name,
                    )
                    continue
                aspec = asb.get("spec") or {}
                schema_d = aspec.get("schema") or {}
                schema = APISchema(
                    name=APISchemaName(schema_d.get("name", "OpenAI")),
                    version=schema_d.get("version", "") or "",
                )
                try:
                    upstream = _resolve_upstream(bundle, ns, aspec.get("backendRef") or {})
                except ConfigError as e:
                    logger.warning("translate: %s; skipping ref %r", e, asb_name)
                    continue
                auth = _bsp_for_backend(bundle, ns, asb_name)
                timeout = 60.0
                timeouts = rule.get("timeouts") or {}
                if timeouts.get("request"):
                    timeout = _parse_duration(timeouts["request"])
                idle = 0.0
                if rule.get("streamIdleTimeout"):
                    idle = _parse_duration(rule["streamIdleTimeout"])
                backends.append(
                    Backend(
                        name=asb_name,
                        schema=schema,
                        upstream=upstream,
                        weight=bref.get("weight", 1),
                        priority=bref.get("priority", 0),
                        model_name_override=bref.get("modelNameOverride", ""),
                        auth=auth,
                        timeout_s=timeout,
                        stream_idle_timeout_s=idle,
                    )
                )
            if backends:
                routes.append(
                    Route(
                        name=f"{md.get('name','route')}-rule-{i}",
                        headers=headers,
                        backends=backends,
                        model_name_override=rule.get("modelNameOverride", ""),
                        request_costs=route_costs,
                    )
                )

    # BackendTrafficPolicy Global rate limits -> in-process token buckets
    for btp in bundle.by_kind.get("BackendTrafficPolicy", []):
        md = btp.get("metadata") or {}
        rl = ((btp.get("spec") or {}).get("rateLimit") or {})
        if rl.get("type") != "Global":
            continue
        for i, rule in enumerate((rl.get("global") or {}).get("rules") or []):
            limit = rule.get("limit") or {}
            cost = ((rule.get("cost") or {}).get("response") or {})
            meta = (cost.get("metadata") or {})
            key = meta.get("key", "llm_total_token")
            key_headers = []
            for sel in rule.get("clientSelectors") or []:
                for h in sel.get("headers") or []:
                    if h.get("type") == "Distinct" or "value" not in h:
                        key_headers.append(h.get("name", ""))
            rate_limits.append(
                RateLimitRule(
                    name=f"{md.get('name','rl')}-{i}",
                    metadata_key=key,
                    limit=limit.get("requests", 0) or 1,
                    window_s=_UNIT_SECONDS.get(limit.get("unit", "Hour"), 3600.0),
                    key_headers=key_headers,
                )
            )

    # QuotaPolicy (v1alpha1) -> same bucket model
    for qp in bundle.by_kind.get("QuotaPolicy", []):
        md = qp.get("metadata") or {}
        spec = qp.get("spec") or {}
        for i, rule in enumerate(spec.get("rules") or []):
            limit = rule.get("limit") or {}
            rate_limits.append(
                RateLimitRule(
                    name=f"{md.get('name','quota')}-{i}",
                    metadata_key=rule.get("metadataKey", "llm_total_token"),
                    limit=limit.get("tokens", limit.get("requests", 1)),
                    window_s=_parse_duration(limit.get("window", "1h")),
                    key_headers=rule.get("keyHeaders") or [],
                )
            )

    for mr in bundle.by_kind.get("MCPRoute", []):
        md = mr.get("metadata") or {}
        ns = md.get("namespace", "default")
        spec = mr.get("spec") or {}
        mcp_backends = []
        for bref in spec.get("backendRefs") or []:
            upstream = _resolve_upstream(bundle, ns, bref)
            sel = bref.get("toolSelector") or {}
            auth = None
            sp = bref.get("securityPolicy") or {}
            if sp.get("apiKey"):
                ref = (sp["apiKey"].get("secretRef") or {})
                inline = sp["apiKey"].get("inline")
                key = inline or _secret_value(
                    bundle, ref.get("name", ""), ref.get("namespace", ns),
                    ["apiKey", "api-key", "token"],
                )
                auth = BackendAuth(api_key=key.strip())
            mcp_backends.append(
                MCPBackend(
                    name=bref.get("name", ""),
                    upstream=upstream,
                    path=bref.get("path", "/mcp"),
                    # v1beta1 uses include/exclude + includeRegex/excludeRegex
                    # (mcp_route.go ToolSelector); our selector treats entries
                    # as exact-or-regex, so the four lists merge
                    tool_include=(sel.get("include") or []) + (sel.get("includeRegex") or []),
                    tool_exclude=(sel.get("exclude") or []) + (sel.get("excludeRegex") or []),
                    auth=auth,
                )
            )
        # route-level OAuth JWT validation (securityPolicy.oauth ->
        # MCPOAuth): inline/local JWKS only; remote JWKS needs egress, so
        # it is reported loudly rather than silently leaving the gate open
        oauth = None
        rsp = spec.get("securityPolicy") or {}
        oa = rsp.get("oauth")
        if oa:
            jwks = oa.get("jwks") or {}
            local = jwks.get("localJWKS") or {}
            inline = local.get("inline", "")
            if not inline and jwks.get("remoteJWKS"):
                raise ConfigError(
                    f"MCPRoute {md.get('name')!r}: remoteJWKS requires network "
                    "egress; provide jwks.localJWKS.inline instead"
                )
            from aigw.filterapi.config import MCPOAuth

            oauth = MCPOAuth(
                issuer=oa.get("issuer", ""),
                audiences=oa.get("audiences") or [],
                jwks=inline,
            )
        mcp_routes.append(
            MCPRoute(name=md.get("name", "mcp"), path=spec.get("path", "/mcp"),
                     backends=mcp_backends, oauth=oauth)
        )

    handled = {
        "AIGatewayRoute", "AIServiceBackend", "BackendSecurityPolicy", "Backend",
        "BackendTrafficPolicy", "GatewayConfig", "QuotaPolicy", "MCPRoute", "Secret",
        "BackendTLSPolicy",
        "Gateway",
    }
    for kind, docs_of_kind in bundle.by_kind.items():
        if kind in handled or kind in _IGNORED_KINDS:
            continue
        # Unknown kinds in OUR API group are typos and must fail loudly;
        # everything else (RBAC, operators, arbitrary k8s objects that ride
        # along in real bundles) is skipped like the controller would.
        if any(
            "aigateway.envoyproxy.io" in (d.get("apiVersion") or "") for d in docs_of_kind
        ):
            raise ConfigError(f"unsupported aigateway.envoyproxy.io kind: {kind}")
        logger.info("translate: skipping %d %s object(s)", len(docs_of_kind), kind)

    cfg = Config(
        uuid="translated",
        routes=routes,
        models=models,
        llm_request_costs=global_costs,
        rate_limits=rate_limits,
        mcp=MCPConfig(routes=mcp_routes) if mcp_routes else None,
    )
    return cfg


def _parse_duration(s) -> float:
    if isinstance(s, (int, float)):
        return float(s)
    s = str(s).strip()
    units = {"ms": 0.001, "s": 1.0, "m": 60.0, "h": 3600.0, "d": 86400.0}
    for suffix in ("ms", "s", "m", "h", "d"):
        if s.endswith(suffix):
            return float(s[: -len(suffix)]) * units[suffix]
    return float(s)


def translate_yaml(text: str) -> Config:
    docs = [d for d in yaml.safe_load_all(text) if d]
    return translate_crds(docs)
