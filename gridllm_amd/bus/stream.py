"""Direct token-stream plane: worker -> gateway chunk delivery over a
point-to-point TCP connection, bypassing the central bus.

In the reference EVERY token chunk crosses Redis (publish -> broker ->
subscriber; client/src/services/WorkerClientService.ts:613-626 ->
JobScheduler.ts:760-791) — a double hop through one broker process. At
MI355X serving rates (hundreds of thousands of tokens/s per node) the
broker becomes the bottleneck, so chunk traffic here flows DIRECTLY from
the worker to the gateway that owns the request:

  * the gateway runs a `StreamServer` and stamps its address into the
    job's metadata (`streamAddr`) at submit time;
  * the worker's flusher groups every active stream's pending chunks by
    destination each cycle and sends ONE batched frame per gateway
    (`StreamSink`), instead of one bus publish per job per cycle;
  * the bus keeps carrying control traffic (assignment, results,
    heartbeats) and remains the fallback when no streamAddr is present
    (e.g. cross-version workers), preserving the reference's at-most-once
    chunk semantics — the job:result path owns delivery either way.

Framing is the bus's length-prefixed msgpack (tcp.py): a batch frame is
{"pushes": [[channel, payload], ...]}.
"""

from __future__ import annotations

import asyncio
from collections import defaultdict
from typing import Callable, Optional

from gridllm_amd.bus.tcp import _pack, _read_frame
from gridllm_amd.utils.logging import get_logger

log = get_logger("stream")

Handler = Callable  # async (channel, payload) -> None


class StreamServer:
    """Gateway-side receiver: dispatches pushed (channel, payload) pairs
    to locally registered handlers. One instance per gateway process."""

    def __init__(self, host: str = "127.0.0.1", port: int = 0):
        self.host = host
        self.port = port
        self._handlers: dict[str, list[Handler]] = defaultdict(list)
        self._server: Optional[asyncio.AbstractServer] = None

    @property
    def addr(self) -> str:
        return f"{self.host}:{self.port}"

    async def start(self) -> None:
        self._server = await asyncio.start_server(
            self._on_conn, self.host, self.port, limit=32 * 1024 * 1024)
        if self.port == 0:  # ephemeral: record the bound port
            self.port = self._server.sockets[0].getsockname()[1]
        log.worker("stream server listening", addr=self.addr)

    async def stop(self) -> None:
        if self._server:
            self._server.close()
            await self._server.wait_closed()
            self._server = None

    def subscribe(self, channel: str, handler: Handler) -> None:
        self._handlers[channel].append(handler)

    def unsubscribe(self, channel: str,
                    handler: Optional[Handler] = None) -> None:
        if handler is None:
            self._handlers.pop(channel, None)
        else:
            hs = [h for h in self._handlers.get(channel, [])
                  if h is not handler]
            if hs:
                self._handlers[channel] = hs
            else:
                self._handlers.pop(channel, None)

    async def _on_conn(self, reader: asyncio.StreamReader,
                       writer: asyncio.StreamWriter) -> None:
        try:
            while True:
                msg = await _read_frame(reader)
                if msg is None:
                    break
                if "pushes" in msg:
                    for ch, payload in msg["pushes"]:
                        await self._dispatch(ch, payload)
                elif "push" in msg:
                    await self._dispatch(msg["push"],
                                         msg.get("payload") or {})
        except (ConnectionError, asyncio.IncompleteReadError, ValueError):
            pass
        finally:
            try:
                writer.close()
            except Exception:  # noqa: BLE001
                pass

    async def _dispatch(self, ch: str, payload: dict) -> None:
        for h in list(self._handlers.get(ch, ())):
            try:
                res = h(ch, payload)
                if asyncio.iscoroutine(res):
                    await res
            except Exception:  # noqa: BLE001
                log.logger.exception("stream handler failed")


class StreamSink:
    """Worker-side sender: pooled connections keyed by 'host:port'.
    send() writes one batched frame; a dead destination drops its
    connection and raises (callers count the drop — streaming stays
    at-most-once, the result path owns delivery)."""

    def __init__(self):
        self._conns: dict[str, asyncio.StreamWriter] = {}
        self._locks: dict[str, asyncio.Lock] = {}

    async def send(self, addr: str, items: list[tuple[str, dict]]) -> None:
        lock = self._locks.setdefault(addr, asyncio.Lock())
        async with lock:
            w = self._conns.get(addr)
            if w is None or w.is_closing():
                host, port = addr.rsplit(":", 1)
                _r, w = await asyncio.open_connection(host, int(port))
                self._conns[addr] = w
            try:
                w.write(_pack({"pushes": [[ch, p] for ch, p in items]}))
                await w.drain()
            except (ConnectionError, OSError):
                self._conns.pop(addr, None)
                try:
                    w.close()
                except Exception:  # noqa: BLE001
                    pass
                raise

    async def close(self) -> None:
        for w in self._conns.values():
            try:
                w.close()
            except Exception:  # noqa: BLE001
                pass
        self._conns.clear()
