"""Two `aigw run` shards (WORLD_SIZE=2, gloo) sharing a global token
budget through the RCCL/gloo StateSync tick — the end-to-end form of the
reference's Redis-backed global rate limit (examples/token_ratelimit),
exercised through the real CLI entrypoint."""

import asyncio
import json
import os
import subprocess
import sys
import threading
import time
import urllib.request

import pytest
from aiohttp import web

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

CFG = """\
version: v1
uuid: multishard
llmRequestCosts:
  - metadataKey: llm_total_token
    type: TotalToken
routes:
  - name: r
    backends:
      - name: b
        schema: OpenAI
        upstream: {host: 127.0.0.1, port: %d}
rateLimits:
  - name: node-budget
    metadataKey: llm_total_token
    limit: 35
    windowS: 3600
"""

OK = {
    "id": "x", "object": "chat.completion", "model": "m",
    "choices": [{"index": 0, "message": {"role": "assistant", "content": "hi"},
                 "finish_reason": "stop"}],
    "usage": {"prompt_tokens": 5, "completion_tokens": 5, "total_tokens": 10},
}


def _start_upstream():
    started = threading.Event()
    state = {}

    def run():
        async def amain():
            async def chat(request):
                return web.json_response(OK)

            app = web.Application()
            app.router.add_post("/v1/chat/completions", chat)
            runner = web.AppRunner(app)
            await runner.setup()
            site = web.TCPSite(runner, "127.0.0.1", 0)
            await site.start()
            state["port"] = site._server.sockets[0].getsockname()[1]
            started.set()
            await asyncio.Event().wait()

        asyncio.run(amain())

    t = threading.Thread(target=run, daemon=True)
    t.start()
    assert started.wait(timeout=30)
    return state["port"]


def _post(port, timeout=10):
    body = json.dumps(
        {"model": "m", "messages": [{"role": "user", "content": "q"}]}
    ).encode()
    req = urllib.request.Request(
        f"http://127.0.0.1:{port}/v1/chat/completions", data=body,
        headers={"content-type": "application/json"},
    )
    try:
        with urllib.request.urlopen(req, timeout=timeout) as r:
            return r.status
    except urllib.error.HTTPError as e:
        return e.code


def _wait_health(port, deadline_s=60):
    end = time.time() + deadline_s
    while time.time() < end:
        try:
            with urllib.request.urlopen(f"http://127.0.0.1:{port}/health", timeout=2):
                return True
        except Exception:
            time.sleep(0.3)
    return False


@pytest.mark.timeout(180)
def test_two_shards_share_token_budget(tmp_path):
    up_port = _start_upstream()
    cfg_path = tmp_path / "cfg.yaml"
    cfg_path.write_text(CFG % up_port)
    base_port = 19930
    env_common = dict(
        os.environ,
        PYTHONPATH=REPO,
        WORLD_SIZE="2",
        MASTER_ADDR="127.0.0.1",
        MASTER_PORT="29751",
    )
    procs = []
    try:
        for rank in range(2):
            env = dict(env_common, RANK=str(rank), LOCAL_RANK=str(rank))
            procs.append(subprocess.Popen(
                [sys.executable, "-m", "aigw", "run", "--config", str(cfg_path),
                 "--host", "127.0.0.1", "--port", str(base_port)],
                env=env, cwd=REPO,
                stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
            ))
        assert _wait_health(base_port), "shard 0 never became healthy"
        assert _wait_health(base_port + 1), "shard 1 never became healthy"

        # burn the whole global budget through shard 0 (3 x 10 tokens < 35,
        # 4th crosses it locally)
        statuses = [_post(base_port) for _ in range(4)]
        assert statuses[:3] == [200, 200, 200], statuses
        # give StateSync a few ticks (0.25 s cadence) to propagate spend
        deadline = time.time() + 20
        denied = False
        while time.time() < deadline:
            if _post(base_port + 1) == 429:
                denied = True
                break
            time.sleep(0.4)
        assert denied, "shard 1 never saw shard 0's token spend"
    finally:
        for p in procs:
            p.terminate()
        for p in procs:
            try:
                p.wait(timeout=15)
            except subprocess.TimeoutExpired:
                p.kill()
