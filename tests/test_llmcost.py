"""Cost-expression engine tests (parity with internal/llmcostcel/cel_test.go)."""

import pytest

from aigw.llmcost import CostProgram, CostVars
from aigw.llmcost.cel import CostExpressionError


V = CostVars(
    model="gpt-4o",
    backend="openai",
    route_name="r1",
    input_tokens=100,
    cached_input_tokens=10,
    cache_creation_input_tokens=5,
    output_tokens=40,
    total_tokens=140,
    reasoning_tokens=7,
)


@pytest.mark.parametrize(
    "expr,expected",
    [
        ("input_tokens + output_tokens", 140),
        ("input_tokens * 2 + output_tokens", 240),
        ("total_tokens", 140),
        ("uint(input_tokens)", 100),
        ("double(output_tokens) * 1.5", 60),
        ("input_tokens - cached_input_tokens", 90),
        ("reasoning_tokens + cache_creation_input_tokens", 12),
        ("model == 'gpt-4o' ? input_tokens : output_tokens", 100),
        ("model == 'other' ? input_tokens : output_tokens", 40),
        ("backend == 'openai' && input_tokens > 50 ? 1 : 0", 1),
        ("backend == 'azure' || route_name == 'r1' ? 2 : 3", 2),
        ("!(input_tokens > 1000) ? 5 : 6", 5),
        # nested ternary
        (
            "model == 'a' ? 1 : model == 'gpt-4o' ? input_tokens + 1 : 3",
            101,
        ),
        ("min(input_tokens, output_tokens)", 40),
        ("max(input_tokens, output_tokens)", 100),
    ],
)
def test_evaluate(expr, expected):
    assert CostProgram(expr).evaluate(V) == expected


@pytest.mark.parametrize(
    "expr",
    [
        "__import__('os')",
        "input_tokens.__class__",
        "open('/etc/passwd')",
        "unknown_var + 1",
        "lambda: 1",
        "[1,2][0]",
        "model",  # string result is an error
        "input_tokens ?",  # syntax error
    ],
)
def test_rejects(expr):
    with pytest.raises(CostExpressionError):
        CostProgram(expr)


def test_negative_cost_rejected():
    p = CostProgram("input_tokens - output_tokens")
    with pytest.raises(CostExpressionError):
        p.evaluate(CostVars(input_tokens=1, output_tokens=5))


def test_string_literal_with_operators():
    p = CostProgram("model == 'a && b ? c : d' ? 1 : 2")
    assert p.evaluate(V) == 2


def test_cel_string_functions_and_membership():
    """cel-go string extensions the reference env exposes: member
    functions on the string vars, list membership, size()."""
    v = CostVars(model="gpt-4o-mini", backend="openai", route_name="prod",
                 input_tokens=100, output_tokens=10, total_tokens=110)
    from aigw.llmcost.cel import evaluate

    cases = {
        "model.startsWith('gpt-4') ? input_tokens * 3 : input_tokens": 300,
        "model.endsWith('mini') ? 1 : 2": 1,
        "model.contains('4o') ? output_tokens : 0": 10,
        "model.matches('gpt-[0-9]+o') ? 7 : 8": 7,
        "model in ['gpt-4o-mini', 'gpt-4o'] ? 5 : 6": 5,
        "backend in ['azure'] ? 1 : 0": 0,
        "size(model) + size(route_name)": 11 + 4,
    }
    for expr, want in cases.items():
        assert evaluate(expr, v) == want, expr
    # attribute surface stays closed
    for bad in ("model.__class__", "model.upper()", "model.join(['x'])"):
        with pytest.raises(CostExpressionError):
            evaluate(bad, v)
