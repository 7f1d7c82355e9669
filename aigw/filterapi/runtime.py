"""Runtime compilation of a filter Config.

Mirrors internal/filterapi/runtime.go:28-60 — the static Config is compiled
once at load time into ready-to-use objects (auth handlers, cost programs,
regex matchers, model registry) so the per-request hot path does zero
parsing. A RuntimeConfig is immutable; config reload builds a fresh one and
swaps a single reference (extproc/server.go:81-88 semantics: in-flight
requests keep the runtime they resolved).
"""

from __future__ import annotations

import re
from dataclasses import dataclass, field
from typing import Optional

from aigw.llmcost import CostProgram, CostVars
from aigw.filterapi.config import (
    Backend,
    Config,
    LLMRequestCost,
    LLMRequestCostType,
    Route,
)


@dataclass
class CompiledCost:
    metadata_key: str
    type: LLMRequestCostType
    program: Optional[CostProgram] = None

    def evaluate(self, v: CostVars) -> int:
        t = self.type
        if t is LLMRequestCostType.INPUT_TOKEN:
            return v.input_tokens
        if t is LLMRequestCostType.OUTPUT_TOKEN:
            return v.output_tokens
        if t is LLMRequestCostType.TOTAL_TOKEN:
            return v.total_tokens
        if t is LLMRequestCostType.CACHED_INPUT_TOKEN:
            return v.cached_input_tokens
        assert self.program is not None
        return self.program.evaluate(v)


@dataclass
class CompiledMatch:
    name: str
    value: str = ""
    regex: Optional[re.Pattern] = None

    def matches(self, headers: dict[str, str]) -> bool:
        got = headers.get(self.name)
        if got is None:
            return False
        if self.regex is not None:
            return self.regex.fullmatch(got) is not None
        if self.value == "":
            return True  # presence match
        return got == self.value


@dataclass
class CompiledRoute:
    route: Route
    matches: list[CompiledMatch]
    costs: list[CompiledCost]
    # backends sorted into priority tiers: tiers[0] = primary weighted set
    tiers: list[list[Backend]] = field(default_factory=list)

    def matches_headers(self, headers: dict[str, str]) -> bool:
        return all(m.matches(headers) for m in self.matches)


def _compile_cost(c: LLMRequestCost) -> CompiledCost:
    prog = CostProgram(c.cel) if c.type is LLMRequestCostType.CEL else None
    return CompiledCost(metadata_key=c.metadata_key, type=c.type, program=prog)


class RuntimeConfig:
    """Immutable compiled view of a Config."""

    def __init__(self, cfg: Config):
        self.config = cfg
        self.uuid = cfg.uuid
        self.model_header = cfg.model_name_header_key
        self.global_costs = [_compile_cost(c) for c in cfg.llm_request_costs]
        self.routes: list[CompiledRoute] = []
        for r in cfg.routes:
            matches = []
            for h in r.headers:
                matches.append(
                    CompiledMatch(
                        name=h.name.lower(),
                        value=h.value,
                        regex=re.compile(h.regex) if h.regex else None,
                    )
                )
            tiers: dict[int, list[Backend]] = {}
            for b in r.backends:
                tiers.setdefault(b.priority, []).append(b)
            self.routes.append(
                CompiledRoute(
                    route=r,
                    matches=matches,
                    costs=self.global_costs + [_compile_cost(c) for c in r.request_costs],
                    tiers=[tiers[k] for k in sorted(tiers)],
                )
            )
        # Model registry for /v1/models (models_processor.go:40-62).
        self.models = list(cfg.models)
        self.models_by_host: dict[str, list] = {}
        self.unscoped_models = []
        for m in self.models:
            if m.hosts:
                for h in m.hosts:
                    self.models_by_host.setdefault(h, []).append(m)
            else:
                self.unscoped_models.append(m)
        self.rate_limits = list(cfg.rate_limits)
        # Every credential-override header name configured anywhere, plus
        # the reserved default AWS trio: the server strips these at egress
        # so a client-sent override can never leak to an upstream that did
        # not opt into per-request credentials (credential_override.go; the
        # reference's controller builds the same list as
        # CredentialOverride.InputHeadersToRemove).
        from aigw import internalapi
        from aigw.backendauth.auth import override_header_names

        strip: set[str] = set(internalapi.aws_credential_override_header_names())
        for r in cfg.routes:
            for b in r.backends:
                if b.auth is not None:
                    strip.update(h.lower() for h in override_header_names(b.auth))
        self.override_strip_headers = frozenset(strip)

    def select_route(self, headers: dict[str, str]) -> Optional[CompiledRoute]:
        """First-match route selection over lowercase header dict."""
        for cr in self.routes:
            if cr.matches_headers(headers):
                return cr
        return None

    def models_for_host(self, host: str):
        host = host.split(":")[0]
        scoped = self.models_by_host.get(host)
        if scoped:
            return scoped + self.unscoped_models
        return self.unscoped_models
