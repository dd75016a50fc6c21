"""TCP bus: pub/sub, KV TTL, hashes, disconnect announcement."""

import asyncio

import pytest

from gridllm_amd.bus.tcp import BusServer, TcpBus


async def _server(port):
    s = BusServer("127.0.0.1", port)
    await s.start()
    return s


@pytest.mark.asyncio
async def test_kv_hash_pubsub_roundtrip():
    srv = await _server(47101)
    try:
        a = TcpBus("127.0.0.1", 47101)
        b = TcpBus("127.0.0.1", 47101)
        await a.connect()
        await b.connect()
        # kv
        await a.kv_set("k1", "v1")
        assert await b.kv_get("k1") == "v1"
        await a.kv_del("k1")
        assert await b.kv_get("k1") is None
        # ttl
        await a.kv_set("k2", "v2", ttl_ms=50)
        assert await b.kv_get("k2") == "v2"
        await asyncio.sleep(0.1)
        assert await b.kv_get("k2") is None
        # hash
        await a.hset("h", "f1", "x")
        await a.hset("h", "f2", "y")
        assert await b.hget("h", "f1") == "x"
        assert await b.hgetall("h") == {"f1": "x", "f2": "y"}
        await a.hdel("h", "f1")
        assert await b.hgetall("h") == {"f2": "y"}
        # pub/sub
        got = asyncio.Queue()

        async def handler(ch, payload):
            await got.put((ch, payload))

        await b.subscribe("chan", handler)
        await a.publish("chan", {"n": 1})
        ch, payload = await asyncio.wait_for(got.get(), 2)
        assert ch == "chan" and payload == {"n": 1}
        # unsubscribe stops delivery
        await b.unsubscribe("chan", handler)
        await a.publish("chan", {"n": 2})
        await asyncio.sleep(0.1)
        assert got.empty()
        await a.close()
        await b.close()
    finally:
        await srv.stop()


@pytest.mark.asyncio
async def test_worker_disconnect_announced():
    srv = await _server(47102)
    try:
        watcher = TcpBus("127.0.0.1", 47102)
        await watcher.connect()
        got = asyncio.Queue()

        async def handler(ch, payload):
            await got.put(payload)

        await watcher.subscribe("worker:disconnected", handler)
        w = TcpBus("127.0.0.1", 47102)
        await w.connect()
        await w.identify("worker-x")
        await w.close()   # abrupt close -> server announces
        payload = await asyncio.wait_for(got.get(), 2)
        assert payload == {"workerId": "worker-x"}
        await watcher.close()
    finally:
        await srv.stop()


@pytest.mark.asyncio
async def test_client_disconnect_callback():
    srv = await _server(47103)
    flag = asyncio.Event()
    c = TcpBus("127.0.0.1", 47103, on_disconnect=lambda: flag.set())
    await c.connect()
    await srv.stop()   # server goes away
    await asyncio.wait_for(flag.wait(), 2)


@pytest.mark.asyncio
async def test_large_payload_roundtrip():
    """Regression: a >64 KiB payload (e.g. a persisted queue snapshot of a
    few hundred jobs) must not trip asyncio's readline limit and drop the
    connection (the >=192-way serving stall)."""
    from gridllm_amd.bus.tcp import BusServer, TcpBus
    import socket as _s
    srv_sock = _s.socket(); srv_sock.bind(("127.0.0.1", 0))
    port = srv_sock.getsockname()[1]; srv_sock.close()
    server = BusServer("127.0.0.1", port)
    await server.start()
    bus = TcpBus("127.0.0.1", port)
    await bus.connect()
    try:
        big = "x" * (300 * 1024)
        await bus.kv_set("bigkey", big)
        got = await bus.kv_get("bigkey")
        assert got == big
        seen = asyncio.Event()

        async def h(ch, p):
            assert len(p["blob"]) == 200 * 1024
            seen.set()
        await bus.subscribe("bigchan", h)
        await bus.publish("bigchan", {"blob": "y" * (200 * 1024)})
        await asyncio.wait_for(seen.wait(), 10)
        assert not bus.closed
    finally:
        await bus.close()
        await server.stop()
