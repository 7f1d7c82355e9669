"""MCP route authorization: CEL-style rules + OAuth scope challenges.

Parity target: internal/mcpproxy/authorization.go (:60-280 rule matching,
:466-478 WWW-Authenticate challenge) and the config schema in
internal/filterapi/mcpconfig.go:101-170. Rules carry an action
(Allow/Deny), an optional CEL condition over the ``request`` activation,
an optional target (tool list, ``backend__tool`` names), and an optional
JWT source (required scopes — ALL must be present — and claims — each
must exist with at least one allowed value). First matching rule wins;
no match falls back to ``defaultAction``. A denial that failed only on
scopes produces an ``insufficient_scope`` WWW-Authenticate challenge
carrying the smallest sufficient scope set and the route's OAuth
protected-resource metadata URL.

The CEL condition engine is a whitelisted-AST transpile of the subset the
reference's MCP configs exercise: attribute chains on ``request``
(``request.mcp.tool``, ``request.auth.jwt.claims.groups``), header
indexing (``request.headers["x-team"]``), string/number literals,
``== != < <= > >=``, ``in``, ``&& || !``, parentheses, and the CEL
string methods ``startsWith/endsWith/contains/matches``. Missing
attributes evaluate to None (CEL optional semantics: comparisons against
absent values are false rather than errors).
"""

from __future__ import annotations

import ast
import base64
import json
import re
from dataclasses import dataclass, field
from typing import Optional

from aigw.filterapi.config import MCPAuthorization, MCPAuthorizationRule


class MCPAuthzError(ValueError):
    """Bad authorization config (compile-time)."""


# ---------------------------------------------------------------------------
# CEL-subset condition engine


class _Missing:
    """Absent attribute: all comparisons false, truthiness false."""

    def __getattr__(self, name):
        return self

    def __getitem__(self, key):
        return self

    def __bool__(self):
        return False

    def __eq__(self, other):
        return False

    def __ne__(self, other):  # CEL: absent != x is an error -> treat false
        return False

    def __contains__(self, item):
        return False

    def __iter__(self):
        return iter(())

    def __hash__(self):
        return 0


MISSING = _Missing()


class _View:
    """Attribute/index view over nested dicts with MISSING for absences."""

    __slots__ = ("_d",)

    def __init__(self, d):
        self._d = d

    def __getattr__(self, name):
        if name.startswith("_"):
            raise AttributeError(name)
        return _wrap(self._d.get(name, MISSING)) if isinstance(self._d, dict) else MISSING

    def __getitem__(self, key):
        if isinstance(self._d, dict):
            return _wrap(self._d.get(key, MISSING))
        if isinstance(self._d, list):
            try:
                return _wrap(self._d[key])
            except (IndexError, TypeError):
                return MISSING
        return MISSING

    def __eq__(self, other):
        return self._d == (other._d if isinstance(other, _View) else other)

    def __ne__(self, other):
        return not self.__eq__(other)

    def __contains__(self, item):
        if isinstance(item, _View):
            item = item._d
        try:
            return item in self._d
        except TypeError:
            return False

    def __bool__(self):
        return bool(self._d)

    def __hash__(self):
        return hash(str(self._d))


class _CelStr(str):
    """str + the CEL string methods, so conditions can call
    ``request.mcp.tool.startsWith("alpha__")`` etc."""

    def startsWith(self, a):  # noqa: N802 - CEL spelling
        return self.startswith(a)

    def endsWith(self, a):  # noqa: N802
        return self.endswith(a)

    def contains(self, a):
        return a in self

    def matches(self, a):
        return re.search(a, self) is not None


def _wrap(v):
    if isinstance(v, (dict, list)):
        return _View(v)
    if isinstance(v, str):
        return _CelStr(v)
    return v


_STR_METHODS = {
    "startsWith": lambda s, a: isinstance(s, str) and s.startswith(a),
    "endsWith": lambda s, a: isinstance(s, str) and s.endswith(a),
    "contains": lambda s, a: isinstance(s, str) and a in s,
    "matches": lambda s, a: isinstance(s, str) and re.search(a, s) is not None,
}

_ALLOWED_NODES = (
    ast.Expression, ast.BoolOp, ast.And, ast.Or, ast.UnaryOp, ast.Not,
    ast.Compare, ast.Eq, ast.NotEq, ast.Lt, ast.LtE, ast.Gt, ast.GtE,
    ast.In, ast.NotIn, ast.Attribute, ast.Subscript, ast.Index, ast.Name,
    ast.Load, ast.Constant, ast.Call, ast.List, ast.Tuple,
)


class CELCondition:
    """One compiled boolean condition over the request activation."""

    def __init__(self, expression: str):
        self.expression = expression
        py = self._transpile(expression)
        try:
            tree = ast.parse(py, mode="eval")
        except SyntaxError as e:
            raise MCPAuthzError(f"cannot parse CEL condition {expression!r}: {e}")
        self._validate(tree)
        self._code = compile(tree, "<mcp-cel>", "eval")

    @staticmethod
    def _transpile(expr: str) -> str:
        # protect string literals, then rewrite CEL operators to Python
        literals: list[str] = []

        def stash(m: re.Match) -> str:
            literals.append(m.group(0))
            return f"\x00{len(literals) - 1}\x00"

        s = re.sub(r"'(?:[^'\\]|\\.)*'|\"(?:[^\"\\]|\\.)*\"", stash, expr)
        s = s.replace("&&", " and ").replace("||", " or ")
        s = re.sub(r"!(?!=)", " not ", s)
        s = re.sub(r"\btrue\b", "True", s)
        s = re.sub(r"\bfalse\b", "False", s)
        s = re.sub(r"\bnull\b", "None", s)
        s = re.sub(r"\x00(\d+)\x00", lambda m: literals[int(m.group(1))], s)
        return s.strip()

    def _validate(self, tree: ast.AST) -> None:
        for node in ast.walk(tree):
            if not isinstance(node, _ALLOWED_NODES):
                raise MCPAuthzError(
                    f"disallowed syntax {type(node).__name__} in CEL "
                    f"condition {self.expression!r}")
            if isinstance(node, ast.Name) and node.id not in ("request", "True",
                                                              "False", "None"):
                raise MCPAuthzError(
                    f"unknown variable {node.id!r} in {self.expression!r}")
            if isinstance(node, ast.Attribute) and node.attr.startswith("_"):
                raise MCPAuthzError(
                    f"disallowed attribute {node.attr!r} in {self.expression!r}")
            if isinstance(node, ast.Call):
                if not (isinstance(node.func, ast.Attribute)
                        and node.func.attr in _STR_METHODS):
                    raise MCPAuthzError(
                        f"disallowed call in {self.expression!r}")

    def evaluate(self, activation: dict) -> bool:
        env = {"request": _View(activation), "__builtins__": {}}
        try:
            return bool(eval(self._code, env))  # noqa: S307 - whitelisted AST
        except Exception:
            # evaluation errors skip the rule (authorization.go:203-206)
            return False


# CEL string methods resolve as Call-on-Attribute; _View exposes the
# whitelisted names so `request.mcp.tool.startsWith("a")` works. For
# values that unwrap to raw str (headers), _View wraps dict/list only —
# so string leaves stay str; method calls on them go through _View via
# the attribute chain ONLY when the parent was a _View. Header/claim
# string leaves therefore need the methods on the VIEW of their parent:
# `request.headers["x"].startsWith(...)` returns a str from __getitem__,
# which has no startsWith — wrap strings in a subclass instead.
def _add_view_methods():
    for name, fn in _STR_METHODS.items():
        def make(fn):
            def method(self, *args):
                base = self._d if isinstance(self, _View) else self
                return fn(base, *args)
            return method
        setattr(_View, name, make(fn))


_add_view_methods()


# ---------------------------------------------------------------------------
# claims / scopes helpers (authorization.go:396-464)


def _claim_path(claims: dict, name: str):
    cur = claims
    for part in name.split("."):
        if not isinstance(cur, dict) or part not in cur:
            return None
        cur = cur[part]
    return cur


def _claim_has_allowed(value, allowed: list[str]) -> bool:
    if isinstance(value, str):
        return value in allowed
    if isinstance(value, list):
        return any(isinstance(v, str) and v in allowed for v in value)
    return False


def extract_scopes(claims: dict) -> set[str]:
    """'scope' (space-separated string) or 'scp' (string or list)."""
    out: set[str] = set()
    for key in ("scope", "scp"):
        v = claims.get(key)
        if isinstance(v, str):
            out.update(v.split())
        elif isinstance(v, list):
            out.update(x for x in v if isinstance(x, str))
    return out


def parse_unverified_claims(authorization_header: str) -> dict:
    """Claims of the bearer JWT WITHOUT verification — the verification
    gate runs earlier (route OAuth / bearer check), exactly like the
    reference trusts Envoy's jwt_authn before this layer
    (authorization.go:135-152)."""
    if not authorization_header.startswith("Bearer "):
        return {}
    parts = authorization_header[7:].split(".")
    if len(parts) != 3:
        return {}
    try:
        pad = parts[1] + "=" * (-len(parts[1]) % 4)
        claims = json.loads(base64.urlsafe_b64decode(pad))
        return claims if isinstance(claims, dict) else {}
    except Exception:
        return {}


def build_insufficient_scope_header(scopes: list[str],
                                    resource_metadata: str) -> str:
    """authorization.go buildInsufficientScopeHeader (:466-478)."""
    parts = ['Bearer error="insufficient_scope"',
             f'scope="{" ".join(scopes)}"']
    if resource_metadata:
        parts.append(f'resource_metadata="{resource_metadata}"')
    parts.append('error_description="The token is missing required scopes"')
    return ", ".join(parts)


# ---------------------------------------------------------------------------
# compiled authorization


@dataclass
class _CompiledRule:
    action_allow: bool
    cel: Optional[CELCondition]
    tools: list[str]
    scopes: list[str]
    claims: list[dict]


@dataclass
class AuthzDecision:
    allowed: bool
    required_scopes: list[str] = field(default_factory=list)


class CompiledAuthorization:
    def __init__(self, cfg: MCPAuthorization):
        self.default_allow = cfg.default_action == "Allow"
        self.resource_metadata_url = cfg.resource_metadata_url
        self.rules: list[_CompiledRule] = []
        for r in cfg.rules:
            if r.action not in ("Allow", "Deny"):
                raise MCPAuthzError(f"unknown rule action {r.action!r}")
            self.rules.append(_CompiledRule(
                action_allow=r.action == "Allow",
                cel=CELCondition(r.cel) if r.cel.strip() else None,
                tools=list(r.tools),
                scopes=list(r.jwt_scopes),
                claims=[dict(c) for c in r.jwt_claims],
            ))

    def authorize(self, *, activation: dict, claims: dict,
                  backend: str, tool: str) -> AuthzDecision:
        """authorization.go authorizeRequestWith (:186-242): first match
        wins; scope-only failures accumulate the challenge scope set."""
        if not self.rules:
            return AuthzDecision(self.default_allow)
        scopes = extract_scopes(claims)
        clean_claims = {k: v for k, v in claims.items() if k not in ("scope", "scp")}
        challenge: list[str] = []
        prefixed = f"{backend}__{tool}" if backend and tool else tool
        for rule in self.rules:
            if rule.cel is not None and not rule.cel.evaluate(activation):
                continue
            if rule.tools and prefixed not in rule.tools and tool not in rule.tools:
                continue
            if not rule.scopes and not rule.claims:
                return AuthzDecision(rule.action_allow)
            claims_ok = all(
                _claim_has_allowed(_claim_path(clean_claims, c.get("name", "")),
                                   c.get("values", []))
                for c in rule.claims
            )
            if not claims_ok:
                continue
            if set(rule.scopes) <= scopes:
                return AuthzDecision(rule.action_allow)
            if rule.action_allow and (not challenge or
                                      len(rule.scopes) < len(challenge)):
                challenge = list(rule.scopes)
        return AuthzDecision(self.default_allow, challenge)


def build_activation(*, http_method: str, host: str, path: str,
                     headers: dict[str, str], mcp_method: str, backend: str,
                     tool: str, params, claims: dict) -> dict:
    """buildCELActivation (:246-300): lowercased single-value headers,
    request.auth.jwt.{claims,scopes}, request.mcp.*."""
    lower = {k.lower(): v for k, v in headers.items()}
    return {
        "method": http_method,
        "host": host,
        "path": path,
        "headers": lower,
        "auth": {"jwt": {"claims": {k: v for k, v in claims.items()
                                    if k not in ("scope", "scp")},
                         "scopes": sorted(extract_scopes(claims))}},
        "mcp": {"method": mcp_method, "backend": backend, "tool": tool,
                "params": params if isinstance(params, dict) else {}},
    }
