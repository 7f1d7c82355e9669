"""Replica telemetry for the endpoint picker.

The reference's InferencePool wiring feeds an external endpoint-picker
service with live replica state (extensionserver/inferencepool.go:39-54,
post_cluster_modify.go); this gateway runs the picker on-GPU
(KV-occupancy scorer kernel) and this module supplies its INPUTS: a
background poller scrapes each backend's Prometheus metrics (vLLM gauge
names by default — gpu_cache_usage_perc, num_requests_running,
num_requests_waiting) and publishes per-backend stats rows
[kv_used, kv_total, queue, active] that the dispatch path reads instead
of gateway-local in-flight estimates. Stale rows (no successful scrape
within 3 poll intervals) fall back to the local estimates so a dead
metrics endpoint degrades gracefully."""

from __future__ import annotations

import asyncio
import logging
import time
from typing import Optional

logger = logging.getLogger("aigw.telemetry")


def parse_prometheus_gauge(text: str, name: str) -> Optional[float]:
    """Sum of all samples of `name` (labels ignored): vLLM exports one
    sample per served model."""
    total = None
    for line in text.splitlines():
        line = line.strip()
        if not line or line.startswith("#"):
            continue
        if not line.startswith(name):
            continue
        rest = line[len(name):]
        if rest and rest[0] not in (" ", "\t", "{"):
            continue  # prefix collision with a longer metric name
        value_part = rest.rsplit(" ", 1)[-1] if " " in rest else ""
        try:
            v = float(value_part)
        except ValueError:
            continue
        # a NaN/Inf gauge from a misbehaving replica would poison the
        # scorer's min() comparison; treat it as absent
        if v != v or v in (float("inf"), float("-inf")):
            continue
        total = v if total is None else total + v
    return total


class ReplicaTelemetry:
    """Polls every backend with a telemetry config; rows are keyed by
    backend name and timestamped for freshness checks."""

    def __init__(self, runtime, *, client=None):
        self.targets = []  # (backend, telemetry)
        for cr in runtime.routes:
            for tier in cr.tiers:
                for b in tier:
                    if b.telemetry is not None:
                        self.targets.append(b)
        self.rows: dict[str, tuple[float, list[float]]] = {}  # name -> (ts, row)
        self.scrape_errors: dict[str, str] = {}
        self._client = client
        self._task: Optional[asyncio.Task] = None

    def fresh_row(self, backend_name: str, max_age_s: float) -> Optional[list[float]]:
        got = self.rows.get(backend_name)
        if got is None:
            return None
        ts, row = got
        if time.monotonic() - ts > max_age_s:
            return None
        return row

    async def scrape_once(self) -> int:
        """One poll sweep (also the unit-test surface); returns the number
        of successfully refreshed rows."""
        import aiohttp

        if self._client is None:
            self._client = aiohttp.ClientSession(
                timeout=aiohttp.ClientTimeout(total=5))
        n = 0
        for b in self.targets:
            t = b.telemetry
            scheme = "https" if b.upstream.tls else "http"
            url = f"{scheme}://{b.upstream.host}:{b.upstream.port}{t.path}"
            try:
                async with self._client.get(url) as r:
                    text = await r.text()
                if r.status != 200:
                    raise RuntimeError(f"status {r.status}")
                usage = parse_prometheus_gauge(text, t.kv_usage_metric) or 0.0
                running = parse_prometheus_gauge(text, t.running_metric) or 0.0
                waiting = parse_prometheus_gauge(text, t.waiting_metric) or 0.0
                row = [usage * t.kv_total, t.kv_total, waiting, running]
                self.rows[b.name] = (time.monotonic(), row)
                self.scrape_errors.pop(b.name, None)
                n += 1
            except Exception as e:
                self.scrape_errors[b.name] = str(e)
                logger.debug("telemetry scrape failed for %s: %s", b.name, e)
        return n

    async def start(self) -> None:
        if not self.targets:
            return

        interval = min(b.telemetry.interval_s for b in self.targets)

        async def loop():
            while True:
                await self.scrape_once()
                await asyncio.sleep(interval)

        self._task = asyncio.ensure_future(loop())

    async def stop(self) -> None:
        if self._task is not None:
            self._task.cancel()
            try:
                await self._task
            except asyncio.CancelledError:
                pass
            self._task = None
        if self._client is not None:
            await self._client.close()
            self._client = None
