"""Per-shard GPU admission service.

One GPU context per shard: the shard's primary process owns the device and
runs this unix-socket service; HTTP worker processes batch their requests
locally and RPC here. Measured motivation (profiles/r01): with a GPU
context per worker, 24 workers collapse to 11k req/s (device thrash from
24 processes' small kernel batches + syncs); without GPU work the same
topology does 95k req/s. Funneling admission through one service restores
worker scaling AND makes kernel batches larger (coalesced across workers),
and gives the shard ONE semantic cache instead of per-worker islands.

Wire protocol: 4-byte little-endian length + msgpack map. Ops:
  {id, op:"count_batch",  texts:[bytes]}          -> {id, counts:[int]}
  {id, op:"lookup_batch", texts:[bytes]}          -> {id, hits:[bytes|nil],
                                                      handles:[int|nil]}
  {id, op:"insert", handle:int, response:bytes}   -> {id, ok:true}
  {id, op:"tokenize", text:bytes}                 -> {id, ids:[int]}
  {id, op:"pick", stats:[[f,f,f,f]], predicted:f} -> {id, index:int}
Replies may arrive out of order (matched by id).
"""

from __future__ import annotations

import asyncio
import itertools
import logging
import struct
from typing import Optional

import msgpack

logger = logging.getLogger("aigw.gpu.service")

_LEN = struct.Struct("<I")


async def _read_msg(reader: asyncio.StreamReader) -> Optional[dict]:
    try:
        head = await reader.readexactly(4)
    except (asyncio.IncompleteReadError, ConnectionResetError):
        return None
    (n,) = _LEN.unpack(head)
    payload = await reader.readexactly(n)
    return msgpack.unpackb(payload, raw=False)


def _pack_msg(obj: dict) -> bytes:
    payload = msgpack.packb(obj, use_bin_type=True)
    return _LEN.pack(len(payload)) + payload


class GPUServiceHost:
    """Runs in the shard primary next to a local GPUServices instance."""

    def __init__(self, gpu_services, socket_path: str):
        self.gpu = gpu_services
        self.socket_path = socket_path
        self._server: Optional[asyncio.AbstractServer] = None
        self._handles = itertools.count(1)
        self._pending_vecs: dict[int, object] = {}

    async def start(self) -> None:
        self._server = await asyncio.start_unix_server(self._serve, path=self.socket_path)

    async def stop(self) -> None:
        if self._server is not None:
            self._server.close()
            await self._server.wait_closed()
            self._server = None

    async def _serve(self, reader: asyncio.StreamReader, writer: asyncio.StreamWriter):
        try:
            while True:
                msg = await _read_msg(reader)
                if msg is None:
                    return
                asyncio.ensure_future(self._dispatch(msg, writer))
        except asyncio.CancelledError:
            raise
        except Exception:
            logger.exception("gpu service connection error")
        finally:
            writer.close()

    async def _dispatch(self, msg: dict, writer: asyncio.StreamWriter) -> None:
        out: dict = {"id": msg.get("id")}
        try:
            op = msg.get("op")
            if op == "count_batch":
                out["counts"] = await self.gpu.count_texts_batch(msg["texts"])
            elif op == "lookup_batch":
                # ONE batched submit for the whole RPC (a sequential
                # per-text await would serialize a micro-batch window per
                # text)
                pairs = await self.gpu.lookup_texts_batch(msg["texts"])
                hits, handles = [], []
                for hit, vec in pairs:
                    hits.append(hit)
                    if vec is not None and hit is None:
                        h = next(self._handles)
                        self._pending_vecs[h] = vec
                        if len(self._pending_vecs) > 4096:  # drop oldest
                            self._pending_vecs.pop(next(iter(self._pending_vecs)))
                        handles.append(h)
                    else:
                        handles.append(None)
                out["hits"] = hits
                out["handles"] = handles
            elif op == "insert":
                vec = self._pending_vecs.pop(msg["handle"], None)
                if vec is not None:
                    await self.gpu.cache_insert(vec, msg["response"])
                out["ok"] = True
            elif op == "tokenize":
                out["ids"] = await self.gpu.tokenize(msg["text"])
            elif op == "pick":
                out["index"] = await self.gpu.pick_endpoint(
                    msg["stats"], float(msg["predicted"])
                )
            else:
                out["error"] = f"unknown op {op!r}"
        except Exception as e:  # pragma: no cover - defensive
            logger.exception("gpu service op failed")
            out["error"] = str(e)
        writer.write(_pack_msg(out))
        try:
            await writer.drain()
        except ConnectionResetError:
            pass


class RemoteGPUClient:
    """Worker-side stand-in for GPUServices: batches locally, RPCs to the
    shard's GPU service. Exposes the same interface GatewayServer uses."""

    def __init__(self, socket_path: str, *, enable_cache: bool = False,
                 window_ms: float = 0.1, max_batch: int = 128):
        self.socket_path = socket_path
        self.cache_enabled = enable_cache
        self.window_ms = window_ms
        self.max_batch = max_batch
        self._reader = None
        self._writer = None
        self._pump_task = None
        self._ids = itertools.count(1)
        self._waiters: dict[int, asyncio.Future] = {}
        self._count_queue: list = []  # (text, future)
        self._lookup_queue: list = []  # (text, future)
        self._flush_handle = None
        self._lock = asyncio.Lock()

    async def _ensure_conn(self):
        if self._writer is None:
            self._reader, self._writer = await asyncio.open_unix_connection(
                self.socket_path
            )
            self._pump_task = asyncio.create_task(self._pump(), name="gpu-rpc-pump")

    async def _pump(self):
        while True:
            msg = await _read_msg(self._reader)
            if msg is None:
                for fut in self._waiters.values():
                    if not fut.done():
                        fut.set_exception(ConnectionError("gpu service closed"))
                self._waiters.clear()
                return
            fut = self._waiters.pop(msg.get("id"), None)
            if fut is not None and not fut.done():
                if "error" in msg:
                    fut.set_exception(RuntimeError(msg["error"]))
                else:
                    fut.set_result(msg)

    async def _call(self, msg: dict) -> dict:
        await self._ensure_conn()
        mid = next(self._ids)
        msg["id"] = mid
        fut = asyncio.get_running_loop().create_future()
        self._waiters[mid] = fut
        async with self._lock:
            self._writer.write(_pack_msg(msg))
            await self._writer.drain()
        return await fut

    # ---- local micro-batching ----------------------------------------------

    def _schedule_flush(self):
        if (len(self._count_queue) + len(self._lookup_queue)) >= self.max_batch:
            if self._flush_handle:
                self._flush_handle.cancel()
                self._flush_handle = None
            asyncio.ensure_future(self._flush())
        elif self._flush_handle is None:
            loop = asyncio.get_running_loop()
            self._flush_handle = loop.call_later(
                self.window_ms / 1000.0, lambda: asyncio.ensure_future(self._flush())
            )

    async def _flush(self):
        self._flush_handle = None
        counts, self._count_queue = self._count_queue, []
        lookups, self._lookup_queue = self._lookup_queue, []
        if counts:
            asyncio.ensure_future(self._flush_counts(counts))
        if lookups:
            asyncio.ensure_future(self._flush_lookups(lookups))

    async def _flush_counts(self, batch):
        try:
            reply = await self._call(
                {"op": "count_batch", "texts": [t for t, _ in batch]}
            )
        except Exception as e:
            for _t, fut in batch:
                if not fut.done():
                    fut.set_exception(e)
            return
        for (_t, fut), c in zip(batch, reply["counts"]):
            if not fut.done():
                fut.set_result(c)

    async def _flush_lookups(self, batch):
        try:
            reply = await self._call(
                {"op": "lookup_batch", "texts": [t for t, _ in batch]}
            )
        except Exception as e:
            for _t, fut in batch:
                if not fut.done():
                    fut.set_exception(e)
            return
        for (_t, fut), hit, handle in zip(batch, reply["hits"], reply["handles"]):
            if not fut.done():
                fut.set_result((hit, handle))

    # ---- GPUServices-compatible interface ------------------------------------

    async def count_text_tokens(self, text: bytes) -> int:
        fut = asyncio.get_running_loop().create_future()
        self._count_queue.append((text or b" ", fut))
        self._schedule_flush()
        return await fut

    async def count_request_tokens(self, body: dict) -> int:
        from aigw.gpu.services import extract_chat_text

        return await self.count_text_tokens(extract_chat_text(body))

    async def cache_lookup_text(self, text: bytes):
        fut = asyncio.get_running_loop().create_future()
        self._lookup_queue.append((text or b" ", fut))
        self._schedule_flush()
        hit, handle = await fut
        return hit, handle  # handle plays the query-vec role for insert

    async def cache_insert(self, handle, response: bytes) -> None:
        if handle is None:
            return
        await self._call({"op": "insert", "handle": handle, "response": response})

    async def tokenize(self, text) -> list[int]:
        if isinstance(text, str):
            text = text.encode("utf-8", "replace")
        reply = await self._call({"op": "tokenize", "text": text or b" "})
        return reply["ids"]

    async def pick_endpoint(self, stats_rows, predicted: float) -> int:
        reply = await self._call(
            {"op": "pick", "stats": [list(map(float, r)) for r in stats_rows],
             "predicted": float(predicted)}
        )
        return reply["index"]

    def close(self):
        if self._pump_task:
            self._pump_task.cancel()
        if self._writer is not None:
            self._writer.close()
            self._writer = None
