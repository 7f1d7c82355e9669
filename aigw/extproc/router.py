"""Backend selection: weighted pick within a priority tier, tier-ordered
fallback, bounded retries.

Reproduces the observable behavior of the reference's Envoy wiring
(SURVEY.md §5.3): backendRef ``weight`` splits traffic inside a tier;
``priority`` orders fallback tiers; on a retriable failure the next attempt
re-selects a backend (preferring untried ones) and the caller re-translates
the ORIGINAL request for it (processor_impl.go:334-336 onRetry semantics).
"""

from __future__ import annotations

import random
from typing import Iterator, Optional

from aigw.filterapi.config import Backend
from aigw.filterapi.runtime import CompiledRoute

# Upstream statuses that trigger fallback to the next backend, mirroring
# Envoy's retriable-status-codes defaults used by the reference examples
# (examples/provider_fallback/fallback.yaml retry policy).
RETRIABLE_STATUSES = frozenset({429, 500, 502, 503, 504})


def _weighted_order(backends: list[Backend], rng: random.Random) -> list[Backend]:
    """Weighted sampling without replacement (Efraimidis-Spirakis keys)."""
    return sorted(
        backends,
        key=lambda b: -(rng.random() ** (1.0 / max(b.weight, 1e-9))),
    )


def backend_attempts(
    route: CompiledRoute, rng: Optional[random.Random] = None
) -> Iterator[Backend]:
    """Yield backends in attempt order: weighted order inside tier 0, then
    each fallback tier. The caller bounds total attempts with route.retries
    (attempts = retries + 1)."""
    rng = rng or random
    for tier in route.tiers:
        if len(tier) == 1:
            yield tier[0]
        else:
            yield from _weighted_order(tier, rng)


def max_attempts(route: CompiledRoute) -> int:
    total_backends = sum(len(t) for t in route.tiers)
    return min(route.route.retries + 1, total_backends) if total_backends else 0
