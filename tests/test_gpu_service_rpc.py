"""Shard GPU admission service RPC: host + remote client over a unix
socket with a fake GPU backend (the real kernels behind the same interface
are covered by tests/test_gpu_gateway.py on hardware)."""

import asyncio

from aigw.gpu.service import GPUServiceHost, RemoteGPUClient


class FakeGPU:
    cache_enabled = True

    def __init__(self):
        self.cache = {}
        self.inserts = []

    async def count_texts_batch(self, texts):
        return [len(t.split()) for t in texts]

    async def cache_lookup_text(self, text):
        if text in self.cache:
            return self.cache[text], None
        return None, ("vec", text)

    async def lookup_texts_batch(self, texts):
        return [await self.cache_lookup_text(t) for t in texts]

    async def cache_insert(self, vec, response):
        _, text = vec
        self.cache[text] = response
        self.inserts.append(text)

    async def tokenize(self, text):
        return list(range(len(text.split())))

    async def pick_endpoint(self, stats, predicted):
        return int(max(range(len(stats)), key=lambda i: stats[i][1]))


def test_rpc_roundtrip(tmp_path):
    async def main():
        sock = str(tmp_path / "gpu.sock")
        fake = FakeGPU()
        host = GPUServiceHost(fake, sock)
        await host.start()
        client = RemoteGPUClient(sock, enable_cache=True, window_ms=0.5)

        # batched counting coalesces concurrent calls into one RPC
        counts = await asyncio.gather(
            client.count_text_tokens(b"one two three"),
            client.count_text_tokens(b"a b"),
            client.count_text_tokens(b"x"),
        )
        assert counts == [3, 2, 1]

        # cache miss -> handle -> insert -> hit (via the service's cache)
        hit, handle = await client.cache_lookup_text(b"what is rust")
        assert hit is None and handle is not None
        await client.cache_insert(handle, b"A LANGUAGE")
        hit2, _ = await client.cache_lookup_text(b"what is rust")
        assert hit2 == b"A LANGUAGE"
        assert fake.inserts == [b"what is rust"]

        assert await client.tokenize("a b c") == [0, 1, 2]
        assert await client.pick_endpoint([[0, 1, 0, 0], [0, 9, 0, 0]], 5.0) == 1

        # concurrent storm: ids demultiplex correctly
        vals = await asyncio.gather(
            *(client.count_text_tokens(("w " * (i % 7 + 1)).encode()) for i in range(200))
        )
        assert vals == [i % 7 + 1 for i in range(200)]

        client.close()
        await host.stop()

    asyncio.run(main())
