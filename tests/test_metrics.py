"""Access-log enrichment: header->attribute mapping
with the requestheaderattrs session-tracking default."""

def test_access_log_session_id_attribute(caplog):
    """The access line carries header-mapped attributes (default
    agent-session-id -> session.id, requestheaderattrs parity)."""
    import asyncio
    import json as _json
    import logging

    import aiohttp

    from aigw.extproc.server import GatewayServer
    from aigw.filterapi import RuntimeConfig, load_config
    from aigw.testing.mockupstream import start_mock_upstream

    async def run():
        mock, runner, port_up = await start_mock_upstream()
        cfg = load_config({
            "routes": [{"name": "r", "backends": [
                {"name": "b", "schema": "OpenAI",
                 "upstream": {"host": "127.0.0.1", "port": port_up}}]}],
        })
        server = GatewayServer(RuntimeConfig(cfg))
        from aigw.extproc.lean_front import serve_lean

        _, port, cleanup = await serve_lean(server, "127.0.0.1", 0,
                                            with_fallback=False)
        with caplog.at_level(logging.INFO, logger="aigw.access"):
            async with aiohttp.ClientSession() as c:
                async with c.post(
                    f"http://127.0.0.1:{port}/v1/chat/completions",
                    json={"model": "m",
                          "messages": [{"role": "user", "content": "x"}]},
                    headers={"agent-session-id": "sess-42"},
                ) as r:
                    assert r.status == 200
        lines = [r.message for r in caplog.records
                 if r.name == "aigw.access"]
        assert lines, "no access line emitted"
        entry = _json.loads(lines[-1])
        assert entry["session.id"] == "sess-42"
        await cleanup()
        await server.close()
        await runner.cleanup()

    asyncio.run(run())
