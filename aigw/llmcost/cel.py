"""CEL-compatible cost-expression engine.

The reference compiles/evaluates google CEL programs over the variables
``model, backend, route_name, input_tokens, cached_input_tokens,
cache_creation_input_tokens, output_tokens, total_tokens, reasoning_tokens``
(internal/llmcostcel/cel.go:19-48). This module accepts the same expression
surface actually exercised by the reference's configs and tests — integer
arithmetic, comparisons, ``&&/||/!``, the CEL ternary ``cond ? a : b``,
string equality against ``model``/``backend``/``route_name``, parentheses,
and the ``uint()/int()/double()`` casts — by transpiling to a whitelisted
Python AST compiled once at config-load time (mirroring cel.go's
compile-at-config-validation, evaluate-per-request split).

Programs are validated at compile time with a probe evaluation, exactly as
the controller does (gateway.go:201), so a bad expression is rejected at
config load rather than at request time.
"""

from __future__ import annotations

import ast
import re
from dataclasses import dataclass
from typing import Union

Number = Union[int, float]

# Variables available to cost programs (llmcostcel/cel.go:19-29).
_INT_VARS = (
    "input_tokens",
    "cached_input_tokens",
    "cache_creation_input_tokens",
    "output_tokens",
    "total_tokens",
    "reasoning_tokens",
)
_STR_VARS = ("model", "backend", "route_name")
_ALLOWED_NAMES = frozenset(_INT_VARS + _STR_VARS)
_ALLOWED_CALLS = frozenset({"uint", "int", "double", "min", "max", "size"})
# CEL string member functions, evaluated on the three string variables
# (cel-go string extensions the reference's env exposes)
_ALLOWED_METHODS = frozenset({"startsWith", "endsWith", "contains", "matches"})

_ALLOWED_NODES = (
    ast.Expression, ast.BinOp, ast.UnaryOp, ast.BoolOp, ast.Compare,
    ast.IfExp, ast.Call, ast.Name, ast.Load, ast.Constant,
    ast.Attribute, ast.List, ast.Tuple,
    ast.Add, ast.Sub, ast.Mult, ast.Div, ast.Mod,
    ast.USub, ast.UAdd, ast.Not,
    ast.And, ast.Or,
    ast.Eq, ast.NotEq, ast.Lt, ast.LtE, ast.Gt, ast.GtE,
    ast.In, ast.NotIn,
)


class CostExpressionError(ValueError):
    """Raised when a cost expression fails to compile or is unsafe."""


@dataclass
class CostVars:
    """Evaluation context for one request (llmcostcel/cel.go:34-48)."""

    model: str = ""
    backend: str = ""
    route_name: str = ""
    input_tokens: int = 0
    cached_input_tokens: int = 0
    cache_creation_input_tokens: int = 0
    output_tokens: int = 0
    total_tokens: int = 0
    reasoning_tokens: int = 0

    def as_dict(self) -> dict:
        return {
            "model": self.model,
            "backend": self.backend,
            "route_name": self.route_name,
            "input_tokens": self.input_tokens,
            "cached_input_tokens": self.cached_input_tokens,
            "cache_creation_input_tokens": self.cache_creation_input_tokens,
            "output_tokens": self.output_tokens,
            "total_tokens": self.total_tokens,
            "reasoning_tokens": self.reasoning_tokens,
        }


# --- CEL → Python source transpilation -------------------------------------

_TERNARY_RE = re.compile(r"\?|:")


def _transpile(expr: str) -> str:
    """Rewrite CEL surface syntax into Python expression syntax.

    Handles: `&&`→`and`, `||`→`or`, `!x`→`not x` (but not `!=`), and the
    ternary `c ? a : b` → `(a) if (c) else (b)` with nesting support.
    String literals are protected from rewriting.
    """
    # Split out string literals so operators inside them are untouched.
    parts: list[str] = []
    lits: list[str] = []
    i = 0
    while i < len(expr):
        ch = expr[i]
        if ch in ("'", '"'):
            j = i + 1
            while j < len(expr):
                if expr[j] == "\\":
                    j += 2
                    continue
                if expr[j] == ch:
                    break
                j += 1
            if j >= len(expr):
                raise CostExpressionError(f"unterminated string literal in {expr!r}")
            lits.append(expr[i : j + 1])
            parts.append(f"\x00{len(lits) - 1}\x00")
            i = j + 1
        else:
            parts.append(ch)
            i += 1
    s = "".join(parts)

    s = s.replace("&&", " and ").replace("||", " or ")
    s = re.sub(r"!(?!=)", " not ", s)
    s = _rewrite_ternaries(s)

    def _restore(m: re.Match) -> str:
        return lits[int(m.group(1))]

    return re.sub("\x00(\\d+)\x00", _restore, s)


def _rewrite_ternaries(s: str) -> str:
    """Convert every `c ? a : b` (possibly nested) to Python ternaries."""
    while True:
        q = _find_top_level(s, "?")
        if q < 0:
            return s
        c = _matching_colon(s, q)
        cond, then, other = s[:q], s[q + 1 : c], s[c + 1 :]
        # The condition extends left to the start of the current
        # parenthesized group; CEL's ?: binds loosest, as does Python's.
        then = _rewrite_ternaries(then)
        other = _rewrite_ternaries(other)
        s = f"(({then}) if ({cond}) else ({other}))"


def _find_top_level(s: str, ch: str) -> int:
    depth = 0
    for i, c in enumerate(s):
        if c == "(":
            depth += 1
        elif c == ")":
            depth -= 1
        elif c == ch and depth == 0:
            return i
    return -1


def _matching_colon(s: str, q: int) -> int:
    depth = 0
    pending = 0
    for i in range(q + 1, len(s)):
        c = s[i]
        if c == "(":
            depth += 1
        elif c == ")":
            depth -= 1
        elif c == "?" and depth == 0:
            pending += 1
        elif c == ":" and depth == 0:
            if pending == 0:
                return i
            pending -= 1
    raise CostExpressionError(f"ternary '?' without matching ':' in {s!r}")


def _validate(tree: ast.AST, expr: str) -> None:
    for node in ast.walk(tree):
        if not isinstance(node, _ALLOWED_NODES):
            raise CostExpressionError(
                f"disallowed construct {type(node).__name__} in cost expression {expr!r}"
            )
        if isinstance(node, ast.Call):
            fn_ok = (isinstance(node.func, ast.Name) and node.func.id in _ALLOWED_CALLS) or (
                isinstance(node.func, ast.Attribute)
                and node.func.attr in _ALLOWED_METHODS)
            if not fn_ok:
                raise CostExpressionError(f"disallowed call in cost expression {expr!r}")
            if node.keywords:
                raise CostExpressionError("keyword arguments not allowed")
        if isinstance(node, ast.Attribute) and node.attr not in _ALLOWED_METHODS:
            raise CostExpressionError(
                f"disallowed attribute {node.attr!r} in cost expression {expr!r}")
        if isinstance(node, ast.Name) and node.id not in _ALLOWED_NAMES | _ALLOWED_CALLS:
            raise CostExpressionError(
                f"unknown variable {node.id!r} in cost expression {expr!r}"
            )
        if isinstance(node, ast.Constant) and not isinstance(node.value, (int, float, str)):
            raise CostExpressionError(f"disallowed literal {node.value!r}")


_CALL_ENV = {"uint": int, "int": int, "double": float, "min": min, "max": max,
             "size": len}


class _CelStr(str):
    """str + the CEL member functions (cel-go spelling) so expressions
    like model.startsWith('gpt-4') compile without exposing Python's
    attribute surface (the AST whitelist rejects every other attr)."""

    def startsWith(self, a):  # noqa: N802 - CEL spelling
        return self.startswith(a)

    def endsWith(self, a):  # noqa: N802
        return self.endswith(a)

    def contains(self, a):
        return a in self

    def matches(self, a):
        return re.search(a, self) is not None


class CostProgram:
    """A compiled cost expression. Thread-safe and reusable across requests."""

    __slots__ = ("expression", "_code")

    def __init__(self, expression: str):
        self.expression = expression
        py_src = _transpile(expression)
        try:
            tree = ast.parse(py_src, mode="eval")
        except SyntaxError as e:
            raise CostExpressionError(f"cannot parse cost expression {expression!r}: {e}") from e
        _validate(tree, expression)
        self._code = compile(tree, f"<cost:{expression}>", "eval")
        # Probe-evaluate so invalid programs fail at config-load time, like
        # the reference controller's CEL validation (gateway.go:201).
        self.evaluate(CostVars(model="m", backend="b", route_name="r"))

    def evaluate(self, v: CostVars) -> int:
        env = dict(_CALL_ENV)
        for k, val in v.as_dict().items():
            env[k] = _CelStr(val) if isinstance(val, str) else val
        try:
            out = eval(self._code, {"__builtins__": {}}, env)  # noqa: S307 - AST whitelisted
        except ZeroDivisionError:
            raise
        except Exception as e:  # pragma: no cover - defensive
            raise CostExpressionError(f"evaluating {self.expression!r}: {e}") from e
        if isinstance(out, bool) or isinstance(out, str):
            raise CostExpressionError(
                f"cost expression {self.expression!r} must produce a number, got {out!r}"
            )
        if isinstance(out, float):
            out = int(out)
        if out < 0:
            raise CostExpressionError(
                f"cost expression {self.expression!r} produced negative cost {out}"
            )
        return out


def compile_program(expression: str) -> CostProgram:
    return CostProgram(expression)


def evaluate(expression: str, v: CostVars) -> int:
    return CostProgram(expression).evaluate(v)
