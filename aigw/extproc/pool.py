"""Dynamic InferencePool member resolution.

The reference resolves pool members behind a k8s label selector and
rewrites the Envoy cluster's endpoints per member, with the EPP fed by
the inference-extension protocol (extensionserver/inferencepool.go:39-54,
post_cluster_modify.go:32). This single-node analogue re-resolves a
route's `pool` source — a DNS name (headless-service style: one A record
per replica) or a members file (`host:port` per line) — on an interval.
When membership changes, the manager rebuilds the Config with the
resolved members appended to the route's static backends and swaps the
runtime through the same path as the file watcher, so in-flight requests
keep the config they resolved and the telemetry poller re-registers the
new member set.
"""

from __future__ import annotations

import asyncio
import copy
import logging
import socket
from typing import Optional

from aigw.filterapi import RuntimeConfig
from aigw.filterapi.config import Backend, InferencePool, Upstream

logger = logging.getLogger("aigw.pool")

# injected members are tagged by name so a rebase (external config
# reload, next resolve tick) can strip and re-add them
MEMBER_PREFIX = "pool:"


def member_name(route_name: str, host: str, port: int) -> str:
    return f"{MEMBER_PREFIX}{route_name}:{host}:{port}"


async def resolve_members(pool: InferencePool) -> list[tuple[str, int]]:
    """Resolve the pool source to a sorted (host, port) member list."""
    if pool.members_file:
        members = []
        try:
            with open(pool.members_file, "r", encoding="utf-8") as fh:
                for line in fh:
                    line = line.strip()
                    if not line or line.startswith("#"):
                        continue
                    host, _, port_s = line.rpartition(":")
                    if not host:
                        host, port_s = line, str(pool.port)
                    members.append((host, int(port_s)))
        except OSError as exc:
            raise RuntimeError(f"pool members file: {exc}") from exc
        return sorted(set(members))
    loop = asyncio.get_running_loop()
    infos = await loop.getaddrinfo(
        pool.service, pool.port, family=socket.AF_INET,
        type=socket.SOCK_STREAM)
    return sorted({(info[4][0], pool.port) for info in infos})


def build_member_backend(route_name: str, pool: InferencePool,
                         host: str, port: int) -> Backend:
    return Backend(
        name=member_name(route_name, host, port),
        schema=copy.deepcopy(pool.schema),
        upstream=Upstream(host=host, port=port),
        auth=copy.deepcopy(pool.auth),
        telemetry=copy.deepcopy(pool.telemetry),
        timeout_s=pool.timeout_s,
        max_concurrency=pool.max_concurrency,
    )


class PoolManager:
    """Background resolver for every route with a `pool` spec.

    Reads the CURRENT runtime config each tick (so an external hot
    reload rebases cleanly: static backends come from the new config,
    members are re-injected on the next tick) and swaps the runtime only
    when membership actually changed.
    """

    def __init__(self, server, swap_cb=None):
        self.server = server
        # swap callback: the CLI passes its reload chain so EVERY front
        # (including the native fast front's route table) follows
        # membership changes, same as a file-watcher reload
        self.swap_cb = swap_cb or server.swap_runtime
        self._task: Optional[asyncio.Task] = None
        self._last: dict[str, list[tuple[str, int]]] = {}
        self.resolve_errors: dict[str, str] = {}

    def _pool_routes(self):
        return [r for r in self.server.runtime.config.routes
                if r.pool is not None]

    async def resolve_once(self) -> bool:
        """One resolution sweep; returns True if the runtime was swapped."""
        routes = self._pool_routes()
        if not routes:
            return False
        desired: dict[str, list[tuple[str, int]]] = {}
        for r in routes:
            try:
                desired[r.name] = await resolve_members(r.pool)
                self.resolve_errors.pop(r.name, None)
            except (RuntimeError, OSError, socket.gaierror, ValueError) as exc:
                # resolution failure keeps the LAST known member set
                # (parity: the reference keeps serving the previous
                # endpoint set while the informer is disconnected)
                self.resolve_errors[r.name] = str(exc)
                logger.warning("pool %s: resolve failed: %s", r.name, exc)
                desired[r.name] = self._last.get(r.name, [])
        current: dict[str, list[tuple[str, int]]] = {}
        for r in routes:
            current[r.name] = sorted(
                (b.upstream.host, b.upstream.port)
                for b in r.backends if b.name.startswith(MEMBER_PREFIX))
        if desired == current:
            self._last = desired
            return False
        cfg = copy.deepcopy(self.server.runtime.config)
        for r in cfg.routes:
            if r.pool is None:
                continue
            static = [b for b in r.backends
                      if not b.name.startswith(MEMBER_PREFIX)]
            members = [build_member_backend(r.name, r.pool, host, port)
                       for host, port in desired.get(r.name, [])]
            r.backends = static + members
        res = self.swap_cb(RuntimeConfig(cfg))
        if res is not None and hasattr(res, "__await__"):
            await res
        await self.server.refresh_telemetry()
        self._last = desired
        logger.info("pool membership updated: %s",
                    {k: len(v) for k, v in desired.items()})
        return True

    async def start(self) -> None:
        if not self._pool_routes():
            return
        await self.resolve_once()  # members present before traffic

        async def loop() -> None:
            while True:
                interval = min((r.pool.interval_s
                                for r in self._pool_routes()), default=5.0)
                await asyncio.sleep(interval)
                try:
                    await self.resolve_once()
                except Exception:  # noqa: BLE001 — poller must survive
                    logger.exception("pool resolve sweep failed")

        self._task = asyncio.get_running_loop().create_task(loop())

    async def stop(self) -> None:
        if self._task is not None:
            self._task.cancel()
            try:
                await self._task
            except asyncio.CancelledError:
                pass
            self._task = None
