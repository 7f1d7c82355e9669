"""RateLimiter unit semantics: fixed-window reset, boundary-crossing
admission (cost at completion, like the reference's Redis flow),
retry-after math, and per-header bucket isolation."""

from aigw.filterapi.config import RateLimitRule
from aigw.ratelimit.limiter import RateLimiter


class Clock:
    def __init__(self):
        self.t = 1000.0

    def __call__(self):
        return self.t


def _mk(limit=100, window_s=60, key_headers=None, clock=None):
    rule = RateLimitRule(name="r", metadata_key="llm_total_token",
                         limit=limit, window_s=window_s,
                         key_headers=key_headers or [])
    return RateLimiter([rule], clock=clock or Clock()), rule


def test_boundary_request_succeeds_then_denied():
    clock = Clock()
    lim, _ = _mk(limit=100, clock=clock)
    assert lim.check({}).allowed
    lim.charge({}, {"llm_total_token": 99})
    # 99 < 100: next request admitted, charge crosses the boundary
    assert lim.check({}).allowed
    lim.charge({}, {"llm_total_token": 50})
    d = lim.check({})
    assert not d.allowed and d.rule == "r"
    assert 0 < d.retry_after_s <= 60


def test_window_reset():
    clock = Clock()
    lim, _ = _mk(limit=10, window_s=60, clock=clock)
    lim.charge({}, {"llm_total_token": 10})
    assert not lim.check({}).allowed
    clock.t += 61
    assert lim.check({}).allowed
    lim.charge({}, {"llm_total_token": 10})
    assert not lim.check({}).allowed


def test_retry_after_shrinks_with_time():
    clock = Clock()
    lim, _ = _mk(limit=1, window_s=60, clock=clock)
    lim.charge({}, {"llm_total_token": 5})
    first = lim.check({}).retry_after_s
    clock.t += 20
    second = lim.check({}).retry_after_s
    assert abs((first - second) - 20) < 1e-6


def test_key_headers_isolate_buckets():
    clock = Clock()
    lim, _ = _mk(limit=10, key_headers=["x-org-id"], clock=clock)
    a = {"x-org-id": "a"}
    b = {"x-org-id": "b"}
    lim.charge(a, {"llm_total_token": 10})
    assert not lim.check(a).allowed
    assert lim.check(b).allowed          # other org unaffected
    assert lim.check({}).allowed         # missing header = its own bucket


def test_unrelated_cost_key_not_charged():
    clock = Clock()
    lim, _ = _mk(limit=10, clock=clock)
    lim.charge({}, {"other_key": 500})
    assert lim.check({}).allowed


def test_remote_spend_counts_against_budget():
    clock = Clock()
    lim, rule = _mk(limit=10, clock=clock)
    lim.charge({}, {"llm_total_token": 4})
    deltas = lim.collect_deltas()
    assert deltas == {("r", ""): 4}
    assert lim.collect_deltas() == {}    # drained
    # all-reduce says global spend is 12 (8 remote + our 4)
    lim.apply_remote(("r", ""), 12, 4)
    assert not lim.check({}).allowed
