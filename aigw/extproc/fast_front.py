"""Native fast front: the C++ data-plane server (csrc/fastpath.cpp)
composed from a RuntimeConfig.

The fast front serves the OpenAI passthrough hot loop (frame → scan →
route → rate-limit → auth → upstream → usage-tap relay) entirely in
native threads; everything it cannot serve natively — provider
translation, multipart audio, MCP, admin endpoints — rides a loopback
aiohttp instance running the full Python app, exactly like the lean
front's fallback. CPython is out of the per-request path.

Eligibility is decided per route at composition time (a matched but
ineligible route is relayed to the fallback per request):
- backend schema is OpenAI (byte-level passthrough) or AzureOpenAI
  (deployments-path rewrite + api-key header); other providers need the
  Python translators;
- backend auth is None, a static/file API key, or an Azure API key
  (SigV4 and vendor token flows stay in Python);
- plain HTTP upstream (TLS upstreams stay in Python);
- no endpoint picker, no body mutations, no tracing.
Whole-config blockers (any present -> FastFrontUnsupported, callers fall
back to the Python fronts): non-model header matches, keyed or CEL rate
rules (the native limiter is fixed-window per rule, like the Python
limiter's un-keyed buckets).
"""

from __future__ import annotations

import logging
from typing import Optional

from aiohttp import web

from aigw.filterapi.config import APISchemaName, LLMRequestCostType
from aigw.filterapi.runtime import RuntimeConfig

try:
    import torch  # noqa: F401 - loads libc10/libamdhip64 for the extension
    import aigw_fast as _fast
except ImportError:  # pragma: no cover - built by setup.py everywhere
    _fast = None

logger = logging.getLogger("aigw.fast_front")


class FastFrontUnsupported(RuntimeError):
    """The config needs features only the Python fronts provide."""


def _backend_eligible(b) -> tuple[bool, Optional[dict]]:
    azure = b.schema.name is APISchemaName.AZURE_OPENAI
    if b.schema.name is not APISchemaName.OPENAI and not azure:
        return False, None
    if b.upstream.tls or b.upstream.path_prefix:
        return False, None
    if b.body_mutation is not None or b.header_mutation is not None:
        return False, None
    if b.max_concurrency:
        return False, None
    auth = b.auth
    bearer = ""
    api_key_file = ""
    if auth is not None:
        if auth.credential_override is not None:
            return False, None
        kind = auth.kind
        if kind == "api_key":
            bearer = auth.api_key
            api_key_file = auth.api_key_file
        elif azure and kind == "azure_api_key":
            bearer = auth.azure_api_key
        elif kind == "":
            pass
        else:
            return False, None
    return True, {
        "name": b.name,
        "host": b.upstream.host,
        "port": b.upstream.port,
        "bearer": bearer,
        "api_key_file": api_key_file,
        "model_override": b.model_name_override or "",
        "azure": azure,
        "azure_api_version": b.schema.version or "",
        "weight": float(b.weight),
        "priority": int(b.priority),
        "timeout_s": float(b.timeout_s),
    }


def route_specs(runtime: RuntimeConfig) -> list[dict]:
    """Native route-table descriptors for a RuntimeConfig (shared by
    initial composition and hot reload); raises FastFrontUnsupported on
    whole-config blockers."""
    specs: list[dict] = []
    for cr in runtime.routes:
        model_match = ""
        for m in cr.matches:
            if m.name == runtime.model_header and m.regex is None and m.value:
                model_match = m.value
            elif m.regex is not None or m.value or m.name != runtime.model_header:
                raise FastFrontUnsupported(
                    f"route {cr.route.name!r} matches on non-model headers"
                )
        eligible = not cr.route.endpoint_picker
        has_costs = bool(cr.costs)
        for c in cr.costs:
            if c.type is LLMRequestCostType.CEL:
                eligible = False  # CEL costs need the Python evaluator
        backends = []
        for tier in cr.tiers:
            for b in tier:
                ok, desc = _backend_eligible(b)
                if not ok:
                    eligible = False
                    break
                backends.append(desc)
            if not eligible:
                break
        if cr.route.header_mutation is not None:
            eligible = False
        if cr.route.model_name_override:
            for d in backends:
                if not d["model_override"]:
                    d["model_override"] = cr.route.model_name_override
        specs.append({
            "name": cr.route.name, "model_match": model_match,
            "retries": int(cr.route.retries), "has_costs": has_costs,
            "eligible": eligible,
            "backends": backends if eligible else [],
        })
    return specs


def build_fast_server(runtime: RuntimeConfig):
    """Compose a FastServer from a RuntimeConfig; raises
    FastFrontUnsupported when the config as a whole cannot be served."""
    if _fast is None:
        raise FastFrontUnsupported("aigw_fast extension not built")
    cfg = runtime.config
    for rule in cfg.rate_limits:
        if rule.key_headers:
            raise FastFrontUnsupported(
                f"rate rule {rule.name!r} uses key headers (python fronts only)"
            )
    srv = _fast.FastServer()
    for spec in route_specs(runtime):
        srv.add_route(spec["name"], spec["model_match"], spec["retries"],
                      spec["has_costs"], spec["eligible"], spec["backends"])
    for rule in cfg.rate_limits:
        srv.add_rate_rule(rule.name, int(rule.limit), float(rule.window_s),
                          rule.metadata_key)
    return srv


class FastFront:
    """Lifecycle wrapper: fallback aiohttp app + native server + a
    metrics bridge exposing the native counters on the Python /metrics."""

    def __init__(self, server, runtime: RuntimeConfig, *, gpu_socket: str = "",
                 gpu_window_us: int = 100, gpu_max_batch: int = 256,
                 gpu_direct: bool = False, gpu_device: int = 0,
                 gpu_cache: bool = False, cache_capacity: int = 65536,
                 cache_threshold: float = 0.92, cache_seed: int = 7,
                 cache_index_dtype: str = "bf16",
                 n_merges: int = 32768, tokenizer_seed: int = 1355):
        # `server` is the Python GatewayServer used for cold paths
        self.py_server = server
        self.runtime = runtime
        self.fast = build_fast_server(runtime)
        self.gpu_socket = gpu_socket
        self.gpu_window_us = gpu_window_us
        self.gpu_max_batch = gpu_max_batch
        self.gpu_direct = gpu_direct
        self.gpu_device = gpu_device
        self.gpu_cache = gpu_cache
        self.cache_capacity = cache_capacity
        self.cache_threshold = cache_threshold
        self.cache_seed = cache_seed
        self.cache_index_dtype = cache_index_dtype
        self.n_merges = n_merges
        self.tokenizer_seed = tokenizer_seed
        self._fallback_runner = None
        self.port = None

    async def start(self, host: str, port: int) -> int:
        app = self.py_server.make_app()
        self._fallback_runner = web.AppRunner(app, access_log=None)
        await self._fallback_runner.setup()
        site = web.TCPSite(self._fallback_runner, "127.0.0.1", 0)
        await site.start()
        fallback_port = self._fallback_runner.addresses[0][1]
        self.fast.set_fallback("127.0.0.1", fallback_port)
        if self.gpu_direct:
            # in-process HIP admission: the native server owns the BPE
            # kernels on its own stream; the merge table comes from the
            # same generator as the Python/CPU oracle, so counts are
            # bit-identical (tests/test_bpe_ref.py)
            from aigw.ops.bpe_ref import build_hash_table, make_merges

            keys, ranks = build_hash_table(make_merges(self.n_merges,
                                                       self.tokenizer_seed))
            self.fast.enable_gpu_direct(keys, ranks,
                                        max_batch=self.gpu_max_batch,
                                        device=self.gpu_device)
            if self.gpu_cache:
                # same random-init weights as aigw.ops.semcache (seeded),
                # bf16 passed as a uint16 view; the native server owns
                # the index ring and serves hits without Python
                import torch as _t

                g = _t.Generator().manual_seed(self.cache_seed)
                vocab = 256 + self.n_merges
                dim = 384
                emb = _t.randn(vocab, dim, generator=g).to(_t.bfloat16)
                proj = (_t.randn(dim, dim, generator=g) / dim ** 0.5).to(_t.bfloat16)
                self.fast.enable_gpu_direct_cache(
                    emb.view(_t.uint16).numpy(), proj.view(_t.uint16).numpy(),
                    dim=dim, capacity=self.cache_capacity,
                    threshold=self.cache_threshold,
                    fp8=self.cache_index_dtype == "fp8")
        sockets = ([self.gpu_socket] if isinstance(self.gpu_socket, str)
                   else list(self.gpu_socket or []))
        for s in sockets:
            if s:
                self.fast.enable_gpu(s, self.gpu_window_us, self.gpu_max_batch)
        if self.render_metrics_extra not in self.py_server.metrics_extra:
            # the fallback app's /metrics carries the native counters too
            self.py_server.metrics_extra.append(self.render_metrics_extra)
        self.port = self.fast.start(host, port)
        logger.info("fast front listening on %s:%d (fallback :%d)", host,
                    self.port, fallback_port)
        return self.port

    def reload(self, runtime: RuntimeConfig) -> None:
        """Hot reload: swap the native route table and the Python
        fallback's runtime; in-flight requests keep the tables they
        resolved (watcher.go swap semantics). Rate rules keep their
        windows — changing rule SHAPES needs a restart (documented)."""
        self.fast.swap_routes(route_specs(runtime))
        self.py_server.swap_runtime(runtime)
        self.runtime = runtime

    def stats(self) -> dict:
        return self.fast.stats()

    def render_metrics_extra(self) -> str:
        """Prometheus lines for the native counters, appended to the
        Python registry's render by the CLI's /metrics."""
        s = self.stats()
        lines = []
        for k in ("requests", "responses_2xx", "responses_4xx", "responses_5xx",
                  "local_429", "fallback", "retries", "gpu_tokens",
                  "cache_hits", "cache_misses",
                  "input_tokens", "output_tokens", "total_tokens",
                  "bytes_in", "bytes_out"):
            lines.append(f'aigw_fast_{k}_total {s[k]}')
        lines.append(f'aigw_fast_active_connections {s["active_connections"]}')
        hist = s["latency_us_log2"]
        cum = 0
        for i, n in enumerate(hist):
            cum += n
            lines.append(
                f'aigw_fast_latency_us_bucket{{le="{2 ** (i + 1)}"}} {cum}')
        return "\n".join(lines) + "\n"

    async def stop(self, drain_s: float = 0.0) -> None:
        if drain_s > 0:
            import asyncio as _asyncio

            left = await _asyncio.to_thread(self.fast.drain, drain_s)
            if left:
                logger.warning("fast front drain timeout with %d connections", left)
        self.fast.stop()
        if self._fallback_runner is not None:
            await self._fallback_runner.cleanup()
            self._fallback_runner = None
        await self.py_server.close()
