"""Typed CRD API surface with kubebuilder-rule validation.

The reference's user-facing API is the CRD types in api/v1beta1 (typed Go
structs with kubebuilder CEL validation enforced by the kube-apiserver;
tests/crdcel/). This module is the MI355X-native equivalent: pydantic
models for the spec of every AIGW kind plus explicit validators encoding
the same XValidation rules, applied by the translate path BEFORE
compilation so a malformed bundle fails with a compact field-level error
instead of a KeyError mid-translate. Non-AIGW kinds (GatewayClass,
HTTPRoute, Deployment, ...) pass through unvalidated, exactly as the
reference's reconcilers ignore them.

Rules mirrored (api/v1beta1, verified against the reference's markers):
- AIGatewayRoute: targetRefs only Gateway (ai_gateway_route.go:62);
  rule names unique and 'route-not-found' reserved (:97, :213);
  InferencePool and AIServiceBackend refs cannot mix in one rule, one
  InferencePool max (:214-215); group+kind together, only
  inference.networking.k8s.io/InferencePool (:314-315); priority 0-65535;
  weight >= 0.
- AIServiceBackend: backendRef must be gateway.envoyproxy.io/Backend or
  a core Service (ai_service_backend.go:66; Service accepted here since
  this gateway resolves addresses itself).
- BackendSecurityPolicy: exactly the credential field matching `type`
  (backendsecurity_policy.go:50-55).
- MCPRoute: backendRefs require name; toolSelector include/exclude lists.
"""

from __future__ import annotations

from typing import Optional

from pydantic import BaseModel, ConfigDict, Field, model_validator

_cfg = ConfigDict(extra="allow", populate_by_name=True)


class CRDValidationError(ValueError):
    pass


class _Spec(BaseModel):
    model_config = _cfg


class TargetRef(_Spec):
    kind: Optional[str] = None
    name: Optional[str] = None
    group: Optional[str] = None


class BackendRef(_Spec):
    name: str
    kind: Optional[str] = None
    group: Optional[str] = None
    namespace: Optional[str] = None
    weight: Optional[int] = Field(default=None, ge=0)
    priority: Optional[int] = Field(default=None, ge=0, le=65535)
    modelNameOverride: Optional[str] = None

    @model_validator(mode="after")
    def _inference_pool_rules(self):
        has_g, has_k = self.group is not None, self.kind is not None
        if has_g != has_k:
            raise ValueError("group and kind must be specified together")
        if has_g and not (self.group == "inference.networking.k8s.io"
                          and self.kind == "InferencePool"):
            raise ValueError(
                "only InferencePool from inference.networking.k8s.io group is supported")
        return self

    @property
    def is_inference_pool(self) -> bool:
        return self.kind == "InferencePool"


class RouteRuleMatchHeader(_Spec):
    name: str
    value: Optional[str] = None
    type: Optional[str] = None


class RouteRuleMatch(_Spec):
    headers: list[RouteRuleMatchHeader] = []


class AIGatewayRouteRule(_Spec):
    name: Optional[str] = None
    matches: list[RouteRuleMatch] = []
    backendRefs: list[BackendRef] = []
    modelNameOverride: Optional[str] = None
    timeouts: Optional[dict] = None

    @model_validator(mode="after")
    def _rule_checks(self):
        if self.name == "route-not-found":
            raise ValueError("rule name route-not-found is reserved")
        pools = [r for r in self.backendRefs if r.is_inference_pool]
        if pools and len(pools) != len(self.backendRefs):
            raise ValueError(
                "cannot mix InferencePool and AIServiceBackend references in the same rule")
        if len(pools) > 1:
            raise ValueError("only one InferencePool backend is allowed per rule")
        return self


class AIGatewayRouteSpec(_Spec):
    targetRefs: list[TargetRef] = []
    parentRefs: list[TargetRef] = []
    rules: list[AIGatewayRouteRule] = []
    llmRequestCosts: list[dict] = []

    @model_validator(mode="after")
    def _route_checks(self):
        for t in self.targetRefs:
            if t.kind is not None and t.kind != "Gateway":
                raise ValueError("only Gateway is supported")
        named = [r.name for r in self.rules if r.name]
        if len(named) != len(set(named)):
            raise ValueError("rule name must be unique within the route")
        return self


class AIServiceBackendSpec(_Spec):
    schema_: dict = Field(alias="schema")
    backendRef: dict

    @model_validator(mode="after")
    def _backend_ref_kind(self):
        kind = self.backendRef.get("kind", "Backend")
        group = self.backendRef.get("group", "gateway.envoyproxy.io")
        if (kind, group) not in (("Backend", "gateway.envoyproxy.io"),
                                 ("Service", ""), ("Service", "core")):
            raise ValueError(
                "BackendRef must be a Backend resource of Envoy Gateway "
                "(or a core Service on this gateway)")
        name = self.schema_.get("name")
        if name not in ("OpenAI", "AWSBedrock", "AWSAnthropic", "AzureOpenAI",
                        "GCPVertexAI", "GCPAnthropic", "Anthropic", "Cohere"):
            raise ValueError(f"unknown API schema {name!r}")
        return self


_BSP_FIELDS = {
    "APIKey": "apiKey",
    "AWSCredentials": "awsCredentials",
    "AzureAPIKey": "azureAPIKey",
    "AzureCredentials": "azureCredentials",
    "GCPCredentials": "gcpCredentials",
    "AnthropicAPIKey": "anthropicAPIKey",
}


class BackendSecurityPolicySpec(_Spec):
    type: str
    targetRefs: list[TargetRef] = []
    apiKey: Optional[dict] = None
    awsCredentials: Optional[dict] = None
    azureAPIKey: Optional[dict] = None
    azureCredentials: Optional[dict] = None
    gcpCredentials: Optional[dict] = None
    anthropicAPIKey: Optional[dict] = None

    @model_validator(mode="after")
    def _exactly_one_credential(self):
        want = _BSP_FIELDS.get(self.type)
        if want is None:
            raise ValueError(f"unknown BackendSecurityPolicy type {self.type!r}")
        for t, f in _BSP_FIELDS.items():
            v = getattr(self, f)
            if f == want and v is None:
                raise ValueError(
                    f"When type is {self.type}, the {want} field must be set")
            if f != want and v is not None:
                raise ValueError(
                    f"When type is {self.type}, only {want} field should be set")
        return self


class MCPBackendRef(_Spec):
    name: str
    kind: Optional[str] = None
    path: Optional[str] = None
    toolSelector: Optional[dict] = None
    securityPolicy: Optional[dict] = None


class MCPRouteSpec(_Spec):
    parentRefs: list[TargetRef] = []
    targetRefs: list[TargetRef] = []
    path: Optional[str] = None
    backendRefs: list[MCPBackendRef] = []
    securityPolicy: Optional[dict] = None


class GatewayConfigSpec(_Spec):
    llmRequestCosts: list[dict] = []


class QuotaPolicySpec(_Spec):
    targetRefs: list[TargetRef] = []
    rules: list[dict] = []


_SPEC_MODELS = {
    "AIGatewayRoute": AIGatewayRouteSpec,
    "AIServiceBackend": AIServiceBackendSpec,
    "BackendSecurityPolicy": BackendSecurityPolicySpec,
    "MCPRoute": MCPRouteSpec,
    "GatewayConfig": GatewayConfigSpec,
    "QuotaPolicy": QuotaPolicySpec,
}


def validate_crd(doc: dict):
    """Validate one CRD document; returns the typed spec (or None for
    kinds without a model). Raises CRDValidationError with
    kind/name-qualified messages."""
    kind = doc.get("kind", "")
    model = _SPEC_MODELS.get(kind)
    if model is None:
        return None
    name = (doc.get("metadata") or {}).get("name", "?")
    spec = doc.get("spec")
    if not isinstance(spec, dict):
        raise CRDValidationError(f"{kind}/{name}: spec is required")
    try:
        return model.model_validate(spec)
    except CRDValidationError:
        raise
    except Exception as e:
        if hasattr(e, "errors"):
            parts = []
            for err in e.errors()[:3]:
                loc = ".".join(str(x) for x in err.get("loc", ()))
                parts.append(f"{loc}: {err.get('msg', 'invalid')}"
                             if loc else err.get("msg", "invalid"))
            raise CRDValidationError(f"{kind}/{name}: " + "; ".join(parts)) from e
        raise CRDValidationError(f"{kind}/{name}: {e}") from e


def validate_bundle(docs: list[dict]) -> int:
    """Validate every AIGW document in a bundle; returns the count of
    validated specs."""
    n = 0
    for doc in docs:
        if isinstance(doc, dict) and validate_crd(doc) is not None:
            n += 1
    return n
