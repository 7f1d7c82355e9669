"""The ``aigw`` CLI: run / translate / healthcheck / version.

Parity with cmd/aigw (run.go:91, translate.go:41, healthcheck.go), minus
the Envoy orchestration the reference needs — here ``run`` starts the
in-process gateway shard(s) directly. ``--shards N`` forks one shard per
GPU and initializes RCCL-backed state sync between them (one process per
GPU, torch.distributed over nccl=RCCL).

Usage:
    python -m aigw run [--config FILE] [--port P] [--shards N] [--gpu]
    python -m aigw translate FILE...          # CRD YAML -> filter config
    python -m aigw healthcheck [--port P]
    python -m aigw version
"""

from __future__ import annotations

import argparse
import asyncio
import json
import logging
import os
import sys
import urllib.request

import aigw
from aigw import internalapi
from aigw.autoconfig import config_from_env
from aigw.controller import translate_yaml
from aigw.extproc.server import GatewayServer, run_server
from aigw.filterapi import ConfigWatcher, RuntimeConfig, load_config_file
from aigw.filterapi.config import dump_config_yaml


def _is_crd_bundle(text: str) -> bool:
    return "apiVersion" in text and "kind" in text


def _load_any_config(path: str | None):
    """Returns (config, watch_path, crd_mode). CRD bundles are watched
    too: a change re-runs the controller translation (the single-node
    reconcile loop; reference: controller/gateway.go watch -> push)."""
    if path is None:
        return config_from_env(), None, False
    import os as _os

    if _os.path.isdir(path):  # sharded config bundle dir (filterapi.bundle)
        from aigw.filterapi import bundle as _bundle

        return _bundle.load_bundle(path), path, False
    with open(path, "r", encoding="utf-8") as f:
        text = f.read()
    if _is_crd_bundle(text):
        from aigw.filterapi.config import _expand_env

        # ${ENV} expansion applies at load time on the run path (the
        # offline `aigw translate` keeps placeholders for GitOps)
        return translate_yaml(_expand_env(text)), path, True
    return load_config_file(path), path, False


async def _run_shard(args, rank: int = 0, world: int = 1) -> None:
    import torch

    from aigw.utils import tune_gc

    tune_gc()

    cfg, watch_path, crd_mode = _load_any_config(args.config)
    runtime = RuntimeConfig(cfg)

    gpu_services = None
    gpu_host = None
    if args.gpu and torch.cuda.is_available():
        gpu_socket = getattr(args, "_gpu_socket", "")
        if gpu_socket and getattr(args, "_is_worker", False):
            # worker process: RPC to the shard primary's GPU admission
            # service (one GPU context + one shared semantic cache per shard)
            from aigw.gpu.service import RemoteGPUClient

            gpu_services = RemoteGPUClient(
                gpu_socket, enable_cache=args.semantic_cache
            )
        else:
            from aigw.gpu import GPUServices

            gpu_services = GPUServices(
                device=f"cuda:{rank}", enable_cache=args.semantic_cache,
                cache_index_dtype=getattr(args, "cache_index_dtype", "bf16"),
            )
            if gpu_socket:
                from aigw.gpu.service import GPUServiceHost

                gpu_host = GPUServiceHost(gpu_services, gpu_socket)
                await gpu_host.start()
    server = GatewayServer(
        runtime, gpu_services=gpu_services,
        root_prefix=getattr(args, "root_prefix", ""),
        strict_schema=getattr(args, "strict_schema", False),
    )

    sync = None
    if world > 1:
        import torch.distributed as dist

        if dist.is_initialized():  # primary process of each shard only
            from aigw.parallel import StateSync

            sync = StateSync(server.limiter)
            await sync.start()

    watcher = None
    reload_cbs = [server.swap_runtime]

    def _on_reload(rc):
        for cb in reload_cbs:
            cb(rc)

    if watch_path:
        watcher = ConfigWatcher(watch_path, _on_reload, crd_mode=crd_mode)
        await watcher.start()

    # dynamic InferencePool membership (inferencepool.go analogue):
    # DNS / members-file re-resolution swapping the runtime per change
    pool_mgr = None
    if any(r.pool is not None for r in cfg.routes):
        from aigw.extproc.pool import PoolManager

        pool_mgr = PoolManager(server, swap_cb=_on_reload)
        await pool_mgr.start()

    # OIDC/token-exchange credential rotation (BSP rotator analogue):
    # refreshes the credential files the auth handlers read
    rotation_mgr = None
    if not getattr(args, "_is_worker", False):
        from aigw.backendauth.rotators import build_rotation_manager

        rotation_mgr = build_rotation_manager(cfg)
        if rotation_mgr is not None:
            await rotation_mgr.start()

    port = args.port + rank
    reuse = getattr(args, "workers", 1) > 1
    front_choice = getattr(args, "front", "lean")
    if front_choice == "fast":
        # native C++ data plane (csrc/fastpath.cpp); cold paths ride its
        # loopback fallback app. Configs the native server cannot express
        # fail loudly here — pick --front lean for those.
        if getattr(args, "strict_schema", False):
            raise SystemExit(
                "--strict-schema validates in the Python pipeline; the native "
                "front's hot paths would bypass it — use --front lean (or "
                "aiohttp) with --strict-schema")
        import torch as _torch

        from aigw.extproc.fast_front import FastFront

        fast = FastFront(
            server, runtime,
            gpu_direct=bool(args.gpu and _torch.cuda.is_available()),
            gpu_cache=bool(args.gpu and args.semantic_cache
                           and _torch.cuda.is_available()),
        )
        port = await fast.start(args.host, port)
        reload_cbs.append(fast.reload)

        async def _fast_cleanup():
            await fast.stop(
                drain_s=float(os.environ.get("AIGW_DRAIN_TIMEOUT", "30")))

        front_cleanup = _fast_cleanup
    elif front_choice == "aiohttp":
        runner = await run_server(server, host=args.host, port=port, reuse_port=reuse)
        front_cleanup = runner.cleanup
    else:
        # default: the raw-asyncio lean front the benchmarks measure; the
        # loopback aiohttp fallback keeps the cold paths (multipart, MCP)
        # on the same port
        from aigw.extproc.lean_front import serve_lean

        _, _, lean_cleanup = await serve_lean(
            server, args.host, port, with_fallback=True, reuse_port=reuse
        )
        front_cleanup = lean_cleanup
    admin_runner = None
    admin_port = getattr(args, "admin_port", 0)
    if admin_port:
        # separate admin server for /health + /metrics (mainlib --adminPort;
        # keeps scrapers and probes off the data-plane listener)
        from aiohttp import web as _web

        admin_app = _web.Application()
        admin_app.router.add_get("/health", server._handle_health)
        admin_app.router.add_get("/metrics", server._handle_metrics)
        admin_runner = _web.AppRunner(admin_app, access_log=None)
        await admin_runner.setup()
        await _web.TCPSite(admin_runner, "127.0.0.1", admin_port + rank).start()
    print(f"aigw shard {rank}/{world} listening on http://{args.host}:{port}", flush=True)
    stop = asyncio.Event()
    loop = asyncio.get_running_loop()
    import signal as _signal

    for sig in (_signal.SIGTERM, _signal.SIGINT):
        try:
            loop.add_signal_handler(sig, stop.set)
        except (NotImplementedError, RuntimeError):
            pass
    try:
        await stop.wait()
        # graceful drain: fail /health, 503 new work, finish in-flight
        left = await server.drain(float(os.environ.get("AIGW_DRAIN_TIMEOUT", "30")))
        if left:
            print(f"aigw shard {rank}: drain timeout with {left} in flight", flush=True)
    finally:
        if watcher:
            await watcher.stop()
        if pool_mgr is not None:
            await pool_mgr.stop()
        if rotation_mgr is not None:
            await rotation_mgr.stop()
        if sync:
            await sync.stop()
        if gpu_host is not None:
            await gpu_host.stop()
        if admin_runner is not None:
            await admin_runner.cleanup()
        await front_cleanup()


def cmd_run(args) -> int:
    world = int(os.environ.get("WORLD_SIZE", args.shards))
    rank = int(os.environ.get("RANK", 0))
    if args.workers > 1:
        # worker processes share the listen port via SO_REUSEPORT; GPU work
        # funnels through the primary's per-shard admission service (one GPU
        # context + one shared semantic cache per shard). Token-budget
        # buckets are per worker within a shard (cross-SHARD consistency is
        # the RCCL-synced path); divide limits by worker count when exact
        # in-shard budgets matter.
        import multiprocessing as _mp

        if args.gpu:
            args._gpu_socket = f"/tmp/aigw-gpu-{rank}-{os.getpid()}.sock"
        ctx = _mp.get_context("spawn")
        procs = [
            ctx.Process(target=_worker_run, args=(args, rank, world, i))
            for i in range(1, args.workers)
        ]
        for p in procs:
            p.start()
    if world > 1:
        import torch
        import torch.distributed as dist

        backend = "nccl" if (args.gpu and torch.cuda.is_available()) else "gloo"
        if backend == "nccl":
            torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", rank)))
        dist.init_process_group(backend)
    try:
        asyncio.run(_run_shard(args, rank, world))
    except KeyboardInterrupt:
        pass
    return 0


def _worker_run(args, rank: int, world: int, worker_idx: int) -> None:
    args._is_worker = True
    import time as _t

    _t.sleep(1.0)  # let the primary bind the GPU service socket first
    try:
        asyncio.run(_run_shard(args, rank, world))
    except KeyboardInterrupt:
        pass


def cmd_translate(args) -> int:
    docs = []
    for path in args.files:
        with open(path, "r", encoding="utf-8") as f:
            docs.append(f.read())
    cfg = translate_yaml("\n---\n".join(docs))
    sys.stdout.write(dump_config_yaml(cfg))
    return 0


def cmd_mcp_stdio(args) -> int:
    from aigw.mcp.stdio_bridge import serve_stdio_bridge

    command = [c for c in args.command if c != "--"]
    if not command:
        print("mcp-stdio: missing server command", file=sys.stderr)
        return 2

    async def run():
        _bridge, runner = await serve_stdio_bridge(command, args.host, args.port)
        print(f"mcp-stdio bridge on http://{args.host}:{args.port}/mcp -> {command}",
              flush=True)
        try:
            await asyncio.Event().wait()
        finally:
            await runner.cleanup()

    try:
        asyncio.run(run())
    except KeyboardInterrupt:
        pass
    return 0


def cmd_healthcheck(args) -> int:
    url = f"http://127.0.0.1:{args.port}/health"
    try:
        with urllib.request.urlopen(url, timeout=5) as r:
            body = json.loads(r.read())
            if body.get("status") == "ok":
                print("ok")
                return 0
    except Exception as e:
        print(f"unhealthy: {e}", file=sys.stderr)
    return 1


def main(argv=None) -> int:
    logging.basicConfig(level=os.environ.get("AIGW_LOG_LEVEL", "INFO"))
    if os.environ.get("AIGW_ACCESS_LOG", "").lower() in ("1", "true", "json"):
        logging.getLogger("aigw.access").setLevel(logging.INFO)
    ap = argparse.ArgumentParser(prog="aigw", description=__doc__)
    sub = ap.add_subparsers(dest="cmd", required=True)

    runp = sub.add_parser("run", help="run the gateway")
    runp.add_argument("--config", default=None, help="filter config YAML, CRD bundle YAML, or sharded bundle dir")
    runp.add_argument("--host", default="0.0.0.0")
    runp.add_argument("--port", type=int, default=internalapi.DEFAULT_LISTEN_PORT)
    runp.add_argument("--shards", type=int, default=1, help="shards (one per GPU)")
    runp.add_argument("--root-prefix", default="", dest="root_prefix",
                      help="global path prefix for every endpoint")
    runp.add_argument("--strict-schema", action="store_true", dest="strict_schema",
                      help="typed-union request validation at ingress "
                           "(aigw.apischema); wrong shapes answer 400")
    runp.add_argument("--front", choices=["lean", "aiohttp", "fast"], default="lean",
                      help="HTTP front: lean raw-asyncio (benchmarked default; aiohttp loopback serves multipart/MCP) or pure aiohttp")
    runp.add_argument("--admin-port", type=int, default=0, dest="admin_port",
                      help="separate localhost admin server for /health and /metrics (0 = serve on the data port only)")
    runp.add_argument("--workers", type=int, default=1,
                      help="HTTP worker processes per shard (SO_REUSEPORT; "
                           "CPython's GIL caps one loop near ~3k req/s)")
    runp.add_argument("--gpu", action="store_true", help="enable GPU services")
    runp.add_argument("--semantic-cache", action="store_true")
    runp.add_argument("--cache-index-dtype", choices=["bf16", "fp8"],
                      default="bf16", dest="cache_index_dtype",
                      help="semantic-cache index precision (fp8 = 2x rows per GB, faster lookups, ~1%% cosine error)")
    runp.set_defaults(fn=cmd_run)

    tr = sub.add_parser("translate", help="CRD bundle -> filter config YAML")
    tr.add_argument("files", nargs="+")
    tr.set_defaults(fn=cmd_translate)

    hc = sub.add_parser("healthcheck")
    hc.add_argument("--port", type=int, default=internalapi.DEFAULT_LISTEN_PORT)
    hc.set_defaults(fn=cmd_healthcheck)

    ver = sub.add_parser("version")
    ver.set_defaults(fn=lambda a: (print(f"aigw {aigw.__version__}"), 0)[1])

    br = sub.add_parser("mcp-stdio", help="expose a stdio MCP server over HTTP "
                                          "(cmd/aigw/stdio2http.go analogue)")
    br.add_argument("--host", default="127.0.0.1")
    br.add_argument("--port", type=int, default=internalapi.DEFAULT_MCP_PORT)
    br.add_argument("command", nargs=argparse.REMAINDER,
                    help="stdio MCP server command (after --)")
    br.set_defaults(fn=cmd_mcp_stdio)

    args = ap.parse_args(argv)
    return args.fn(args)


if __name__ == "__main__":
    raise SystemExit(main())
